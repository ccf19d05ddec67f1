"""CPU tests for the HipModel runtime's buffer plumbing: the k_pad GEMM
contract (ext_bind.hip:gemm) requires A operands to carry >= 128 bytes of
ZEROED storage slack past the tensor end — _slacked provisions that and
registers the storage so hip_ops can verify membership before padding."""

import torch

from zaremba_amd.ops.hip_model import _ks, _pad64, _slacked


def storage_view(t):
    full = torch.empty(0, dtype=t.dtype)
    full.set_(t.untyped_storage())
    return full


def test_pad64():
    assert _pad64(0) == 0
    assert _pad64(1) == 64
    assert _pad64(64) == 64
    assert _pad64(65) == 128
    assert _pad64(1500) == 1536
    assert _pad64(6000) == 6016
    assert _pad64(10000) == 10048


def test_ks():
    assert _ks(32) == 1
    assert _ks(33) == 2
    assert _ks(1500) == 47


def test_slacked_shape_and_registration():
    ptrs = set()
    t = _slacked((7, 13), torch.bfloat16, "cpu", ptrs)
    assert t.shape == (7, 13)
    assert t.is_contiguous()
    assert t.untyped_storage().data_ptr() in ptrs
    # >= 64 elements of storage slack past the tensor end
    assert t.untyped_storage().nbytes() >= (7 * 13 + 64) * t.element_size()


def test_slacked_slack_stays_zero():
    t = _slacked((5, 9), torch.float32, "cpu")
    t.fill_(3.0)  # writes through the view must not touch the slack
    full = storage_view(t)
    n = t.numel()
    assert torch.all(full[:n] == 3.0)
    assert torch.all(full[n:n + 64] == 0.0)


def test_slacked_without_registry():
    t = _slacked((4,), torch.bfloat16, "cpu", None)
    assert t.shape == (4,)


def test_layer_workspace_slack_provisioning():
    """h_all / dG (the k_pad GEMM A operands) come from _slacked; their
    storages land in the shared registry."""
    from zaremba_amd.ops.hip_model import _LayerWorkspace

    ptrs = set()
    ws = _LayerWorkspace(3, 4, 64, "cpu", ptrs)
    assert ws.h_all.untyped_storage().data_ptr() in ptrs
    assert ws.dG.untyped_storage().data_ptr() in ptrs
    assert ws.h_all.shape == (4, 4, 64)
    assert ws.dG.shape == (3, 4, 4 * 64)
    # barrier/counter state: padded-barrier words [0,513) + fused-bwd
    # pair counters at uint32[544..] (ext_bind.hip checks these bounds)
    assert ws.hgran.numel() * 8 >= 513 * 4
    assert ws.hgran.numel() * 8 >= 2176 + ((64 + 15) // 16) * 4
