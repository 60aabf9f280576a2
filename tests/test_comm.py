"""Comm bring-up guards."""
import pytest

from mi355x_scale.parallel.comm import assert_xgmi_mesh, topology_probe


def test_nccl_p2p_disable_is_rejected(monkeypatch):
    """The reference's NCCL_P2P_DISABLE=1 cloud workaround
    (deep_learning/2...py:362-363) must NOT be carried onto MI355X."""
    monkeypatch.setenv("NCCL_P2P_DISABLE", "1")
    with pytest.raises(RuntimeError, match="NCCL_P2P_DISABLE"):
        assert_xgmi_mesh()


def test_topology_probe_cpu_empty():
    import torch
    if not torch.cuda.is_available():
        assert topology_probe() == []


def test_assert_passes_without_gpus(monkeypatch):
    monkeypatch.delenv("NCCL_P2P_DISABLE", raising=False)
    assert_xgmi_mesh()  # no GPUs -> empty matrix -> nothing to flag
