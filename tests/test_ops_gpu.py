"""GPU numerics tests: HIP kernels vs plain PyTorch fp32 references."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from ray_lightning_amd import ops  # noqa: E402
from ray_lightning_amd.optim import (FusedAdamW, FusedSGD,  # noqa: E402
                                     ShardedFusedAdam)


@pytest.fixture(scope="module")
def ext():
    e = ops._load_ext()
    assert e is not None, "HIP extension must be built on the GPU box"
    return e


def _rand_segments(dtype, n_seg=7, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    sizes = [3, 17, 256, 1000, 4096, 65536, 100003][:n_seg]
    return [torch.randn(s, device="cuda", generator=g, dtype=torch.float32)
            .to(dtype) for s in sizes]


@pytest.mark.parametrize("src_dt,dst_dt", [
    (torch.float32, torch.float32),
    (torch.float32, torch.bfloat16),
    (torch.bfloat16, torch.bfloat16),
    (torch.bfloat16, torch.float32),
])
def test_multi_tensor_pack(ext, src_dt, dst_dt):
    srcs = _rand_segments(src_dt)
    # 8-element-aligned slots within one flat buffer (engine layout)
    offs, off = [], 0
    for s in srcs:
        offs.append(off)
        off += (s.numel() + 7) & ~7
    flat = torch.full((off,), -99.0, device="cuda", dtype=dst_dt)
    slots = [flat[o:o + s.numel()] for o, s in zip(offs, srcs)]
    ext.multi_tensor_pack(slots, [s.reshape(-1) for s in srcs])
    torch.cuda.synchronize()
    for slot, s in zip(slots, srcs):
        ref = s.reshape(-1).to(dst_dt)
        assert torch.equal(slot, ref), f"{src_dt}->{dst_dt} mismatch"


def test_scale_inplace(ext):
    for dt in (torch.float32, torch.bfloat16):
        x = torch.randn(100003, device="cuda").to(dt)
        ref = (x.float() * 0.125).to(dt)
        ext.scale_inplace(x, 0.125)
        torch.cuda.synchronize()
        assert torch.equal(x, ref)


def test_scale_cast(ext):
    src = torch.randn(65543, device="cuda").to(torch.bfloat16)
    dst = torch.empty(65543, device="cuda", dtype=torch.float32)
    ext.scale_cast(dst, src, 0.25)
    torch.cuda.synchronize()
    ref = src.float() * 0.25
    assert torch.allclose(dst, ref, atol=0, rtol=0)


def _clone_params(params):
    return [p.detach().clone() for p in params]


def test_fused_sgd_matches_torch():
    torch.manual_seed(0)
    shapes = [(64, 32), (128,), (1000, 3), (7,)]
    params_a = [torch.nn.Parameter(torch.randn(*s, device="cuda"))
                for s in shapes]
    params_b = [torch.nn.Parameter(p.detach().clone())
                for p in params_a]
    grads = [torch.randn_like(p) for p in params_a]

    opt_a = FusedSGD(params_a, lr=0.1, momentum=0.9, weight_decay=1e-4)
    opt_b = torch.optim.SGD(params_b, lr=0.1, momentum=0.9,
                            weight_decay=1e-4)
    for step in range(3):
        for p, g in zip(params_a, grads):
            p.grad = g.clone() * (step + 1)
        for p, g in zip(params_b, grads):
            p.grad = g.clone() * (step + 1)
        opt_a.step()
        opt_b.step()
    torch.cuda.synchronize()
    for a, b in zip(params_a, params_b):
        assert torch.allclose(a, b, atol=1e-6, rtol=1e-5), \
            f"max diff {(a - b).abs().max()}"


def test_fused_adamw_matches_torch():
    torch.manual_seed(1)
    shapes = [(33, 65), (4096,), (513,)]
    params_a = [torch.nn.Parameter(torch.randn(*s, device="cuda"))
                for s in shapes]
    params_b = [torch.nn.Parameter(p.detach().clone())
                for p in params_a]
    grads = [torch.randn_like(p) for p in params_a]

    opt_a = FusedAdamW(params_a, lr=1e-3, weight_decay=0.01)
    opt_b = torch.optim.AdamW(params_b, lr=1e-3, weight_decay=0.01)
    for step in range(4):
        for p, g in zip(params_a, grads):
            p.grad = g.clone() * (0.5 + step)
        for p, g in zip(params_b, grads):
            p.grad = g.clone() * (0.5 + step)
        opt_a.step()
        opt_b.step()
    torch.cuda.synchronize()
    for a, b in zip(params_a, params_b):
        assert torch.allclose(a, b, atol=2e-6, rtol=1e-5), \
            f"max diff {(a - b).abs().max()}"


def test_sharded_adam_bf16_vs_fp32_reference():
    torch.manual_seed(2)
    shapes = [(128, 64), (1001,)]
    params16 = [torch.nn.Parameter(
        torch.randn(*s, device="cuda").to(torch.bfloat16))
        for s in shapes]
    # fp32 reference starts from the same (bf16-rounded) values
    ref = [torch.nn.Parameter(p.detach().float().clone())
           for p in params16]
    grads = [torch.randn(*s, device="cuda") for s in shapes]

    opt = ShardedFusedAdam(params16, lr=1e-3, weight_decay=0.01)
    opt_ref = torch.optim.AdamW(ref, lr=1e-3, weight_decay=0.01)
    for step in range(3):
        for p, g in zip(params16, grads):
            p.grad = g.to(torch.bfloat16) * (step + 1)
        for p, g in zip(ref, grads):
            # reference sees the same bf16-rounded gradient
            p.grad = (g.to(torch.bfloat16) * (step + 1)).float()
        opt.step()
        opt_ref.step()
    torch.cuda.synchronize()
    for p, r, in zip(params16, ref):
        # bf16 params must equal the bf16-rounded fp32-master trajectory
        master = opt.state[p]["master"]
        assert torch.allclose(master, r.detach(), atol=3e-6, rtol=1e-5)
        assert torch.equal(p.detach(), master.to(torch.bfloat16))


def test_pack_unpack_scale_roundtrip_bf16_comm():
    """bf16-compressed comm path: pack f32->bf16, scale_cast back."""
    ext = ops._load_ext()
    g = torch.randn(50001, device="cuda")
    flat16 = torch.empty(50008, device="cuda", dtype=torch.bfloat16)
    ext.multi_tensor_pack([flat16[:50001]], [g])
    out = torch.empty(50008, device="cuda", dtype=torch.float32)
    ext.scale_cast(out, flat16, 0.5)
    torch.cuda.synchronize()
    ref = g.to(torch.bfloat16).float() * 0.5
    assert torch.equal(out[:50001], ref)
