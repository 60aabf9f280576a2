"""bench.py driver-contract rehearsal (the round-end SCALE run shape).

The driver launches `python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...` and
parses ONE JSON line from rank 0. Rehearse that exact shape at N=2 on
CPU (gloo; bench.py's CPU smoke config) so contract regressions surface
here instead of in the driver's multi-GPU window.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(extra, env_extra=None, timeout=420):
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    if env_extra:
        env.update(env_extra)
    res = subprocess.run(extra, capture_output=True, text=True,
                         timeout=timeout, cwd=REPO, env=env)
    lines = [ln for ln in res.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert lines, (f"no JSON line. rc={res.returncode}\n"
                   f"stdout:\n{res.stdout[-1500:]}\n"
                   f"stderr:\n{res.stderr[-1500:]}")
    return json.loads(lines[-1])


def _check_contract(out, n_gpus):
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in out, f"missing {key}"
    assert out["n_gpus"] == n_gpus
    assert out["metric"] == "samples/sec"
    assert out["value"] > 0
    assert out["data"] == "synthetic"
    assert out["scaling"] == "weak"
    assert out["config"]["model"] == "resnet18"
    assert out["config"]["global_batch"] == 8 * n_gpus  # CPU smoke batch


def test_bench_single_process_contract():
    out = _run_bench([sys.executable, "bench.py", "--steps", "2",
                      "--warmup", "1"])
    _check_contract(out, 1)


@pytest.mark.timeout(600)
def test_bench_torchrun_ws2_contract():
    """The driver's exact launch shape at N=2 (gloo on CPU)."""
    out = _run_bench([
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", "29377", "bench.py", "--gpus", "2",
        "--steps", "2", "--warmup", "1",
    ], env_extra={"MI355X_BACKEND": "gloo"})
    _check_contract(out, 2)
