"""FlatAdam vs torch.optim.Adam — the flat fused optimizer must match
stock Adam step-for-step (fp32)."""
import pytest
import torch
import torch.nn as nn

from mi355x_scale.train.flat_adam import FlatAdam


def _tiny_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(10, 32), nn.ReLU(), nn.Linear(32, 4))


def _run_steps(model, opt, steps, seed=42):
    g = torch.Generator().manual_seed(seed)
    losses = []
    for _ in range(steps):
        x = torch.randn(16, 10, generator=g)
        y = torch.randn(16, 4, generator=g)
        if isinstance(opt, FlatAdam):
            opt.zero_grad()
        else:
            opt.zero_grad(set_to_none=False)
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def test_cpu_matches_torch_adam():
    m1, m2 = _tiny_model(), _tiny_model()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2)
    o1 = FlatAdam(m1.parameters(), lr=1e-2)
    o2 = torch.optim.Adam(m2.parameters(), lr=1e-2)
    _run_steps(m1, o1, 12)
    _run_steps(m2, o2, 12)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2, atol=1e-6, rtol=1e-5)


def test_params_alias_flat_buffer():
    m = _tiny_model()
    opt = FlatAdam(m.parameters(), lr=1e-3)
    with torch.no_grad():
        opt.flat_params.zero_()
    for p in m.parameters():
        assert p.abs().max().item() == 0.0  # p.data views the flat buffer


def test_state_dict_roundtrip():
    m = _tiny_model()
    opt = FlatAdam(m.parameters(), lr=1e-2)
    _run_steps(m, opt, 3)
    sd = {k: (v.clone() if torch.is_tensor(v) else v)
          for k, v in opt.state_dict().items()}
    _run_steps(m, opt, 2)
    opt.load_state_dict(sd)
    torch.testing.assert_close(opt.step_t,
                               torch.tensor([3], dtype=torch.int32))


@pytest.mark.gpu
def test_gpu_kernel_matches_cpu_reference():
    dev = torch.device("cuda:0")
    m_gpu = _tiny_model().to(dev)
    m_cpu = _tiny_model()
    o_gpu = FlatAdam(m_gpu.parameters(), lr=3e-3, weight_decay=0.01)
    o_cpu = FlatAdam(m_cpu.parameters(), lr=3e-3, weight_decay=0.01)
    g = torch.Generator().manual_seed(7)
    for _ in range(10):
        x = torch.randn(16, 10, generator=g)
        y = torch.randn(16, 4, generator=g)
        for model, opt, d in ((m_gpu, o_gpu, dev),
                              (m_cpu, o_cpu, torch.device("cpu"))):
            opt.zero_grad()
            loss = ((model(x.to(d)) - y.to(d)) ** 2).mean()
            loss.backward()
            opt.step()
    torch.testing.assert_close(o_gpu.flat_params.cpu(), o_cpu.flat_params,
                               atol=1e-4, rtol=1e-4)
    assert int(o_gpu.step_t.item()) == 10


@pytest.mark.gpu
def test_gpu_graph_capture_replay():
    """The fused step must capture into a hipGraph and advance the
    on-device step counter across replays."""
    dev = torch.device("cuda:0")
    m = _tiny_model().to(dev)
    opt = FlatAdam(m.parameters(), lr=1e-3)
    x = torch.randn(8, 10, device=dev)
    y = torch.randn(8, 4, device=dev)

    def one():
        opt.flat_grads.zero_()
        loss = ((m(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        return loss

    for _ in range(3):
        one()
    torch.cuda.synchronize()
    t0 = int(opt.step_t.item())
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph, capture_error_mode="thread_local"):
        one()
    for _ in range(5):
        graph.replay()
    torch.cuda.synchronize()
    # capture records without executing; only the 5 replays bump the
    # on-device counter
    assert int(opt.step_t.item()) == t0 + 5
    assert torch.isfinite(opt.flat_params).all()


def _conv_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8),
                         nn.ReLU(), nn.Flatten(), nn.Linear(8 * 16, 4))


def test_bf16_params_cpu_semantics():
    """bf16-param mode: matrix params become bf16 views of the flat bf16
    buffer, 1-D params stay fp32 master views; the bf16 copies always
    equal the rounded master."""
    m = _conv_model()
    opt = FlatAdam(m.parameters(), lr=1e-2, bf16_params=True)
    conv_w = m[0].weight
    bn_w = m[1].weight
    assert conv_w.dtype == torch.bfloat16
    assert bn_w.dtype == torch.float32
    g = torch.Generator().manual_seed(3)
    for _ in range(4):
        opt.zero_grad()
        x = torch.randn(2, 3, 4, 4, generator=g)
        # CPU conv wants uniform dtype: run the model in fp32 shadow by
        # exercising only the optimizer contract here
        for p in opt.params:
            p.grad.copy_(torch.randn(p.shape, generator=g).to(p.grad.dtype))
        opt.step()
    assert int(opt.step_t.item()) == 4
    torch.testing.assert_close(
        opt.flat_pb16, opt.flat_master[:opt.n_bf16].bfloat16())
    assert torch.isfinite(opt.flat_master).all()


def test_bf16_params_master_follows_adam_math():
    """The mixed-mode fp32 master must follow the exact Adam recurrence
    given the (bf16-rounded) gradients actually applied."""
    m = _conv_model()
    opt = FlatAdam(m.parameters(), lr=1e-2, bf16_params=True)
    master0 = opt.flat_master.clone()
    exp_m = torch.zeros_like(master0)
    exp_v = torch.zeros_like(master0)
    expect = master0.clone()
    g = torch.Generator().manual_seed(5)
    for t in range(1, 6):
        opt.zero_grad()
        flat_g = torch.empty_like(master0)
        off = 0
        for p in opt.params:
            gr = torch.randn(p.shape, generator=g).to(p.grad.dtype)
            p.grad.copy_(gr)
            flat_g[off:off + p.numel()].as_strided(
                p.shape, p.stride()).copy_(gr.float())
            off += p.numel()
        opt.step()
        exp_m.mul_(0.9).add_(flat_g, alpha=0.1)
        exp_v.mul_(0.999).addcmul_(flat_g, flat_g, value=0.001)
        bc1, bc2 = 1 - 0.9 ** t, 1 - 0.999 ** t
        expect.addcdiv_(exp_m, (exp_v / bc2).sqrt().add_(1e-8),
                        value=-1e-2 / bc1)
    torch.testing.assert_close(opt.flat_master, expect,
                               atol=1e-6, rtol=1e-5)
