"""StageTimer / torch_profile wrapper (SURVEY §5.1 helpers)."""
import time

from mi355x_scale.utils.profiling import StageTimer, torch_profile


def test_stage_timer_accumulates_and_reports():
    t = StageTimer()
    with t("decode"):
        time.sleep(0.01)
    with t("decode"):
        time.sleep(0.01)
    t.add("h2d", 0.5)
    rep = t.report()
    assert "decode" in rep and "h2d" in rep and "n=     2" in rep
    t.reset()
    assert t.report() == ""


def test_torch_profile_writes_trace(tmp_path):
    import torch
    with torch_profile(str(tmp_path), wait=0, warmup=0, active=1) as prof:
        torch.ones(8) @ torch.ones(8)
        prof.step()
    assert list(tmp_path.glob("**/*")), "no trace written"
