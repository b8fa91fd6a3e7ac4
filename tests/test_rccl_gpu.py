"""Native RCCL communicator tests (single GPU: world_size=1 comm;
multi-GPU collectives are covered by the driver's scale runs and the
CPU gloo multi-process tests for protocol logic)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)


def test_rccl_ext_importable():
    from ray_lightning_amd.engine.rccl import rccl_available
    assert rccl_available(), "_rccl_comm extension must load on GPU box"


def test_rccl_world1_collectives():
    from ray_lightning_amd.ops import _rccl_comm as ext
    uid = ext.get_unique_id()
    assert len(bytes(uid)) == 128
    h = ext.comm_init(bytes(uid), 0, 1)
    assert ext.comm_rank(h) == 0
    assert ext.comm_world_size(h) == 1

    stream = torch.cuda.current_stream().cuda_stream
    t = torch.randn(4096, device="cuda")
    ref = t.clone()
    idx = ext.all_reduce(h, t, "sum", stream)
    ext.stream_wait_event(h, stream, idx)
    torch.cuda.synchronize()
    assert torch.equal(t, ref)  # world=1 sum is identity

    bt = torch.randn(1000, device="cuda", dtype=torch.bfloat16)
    bref = bt.clone()
    idx = ext.broadcast(h, bt, 0, stream)
    ext.stream_wait_event(h, stream, idx)
    torch.cuda.synchronize()
    assert torch.equal(bt, bref)

    out = torch.empty(512, device="cuda")
    inp = torch.randn(512, device="cuda")
    idx = ext.reduce_scatter(h, out, inp, stream)
    ext.stream_wait_event(h, stream, idx)
    torch.cuda.synchronize()
    assert torch.equal(out, inp)

    ag_out = torch.empty(512, device="cuda")
    idx = ext.all_gather(h, ag_out, inp, stream)
    ext.stream_wait_event(h, stream, idx)
    torch.cuda.synchronize()
    assert torch.equal(ag_out, inp)
    ext.comm_destroy(h)


def test_smoke_entry():
    import __graft_entry__
    __graft_entry__.smoke()
