"""MFMA layout probe + flash attention numerics vs fp32 reference,
plus a timing comparison against torch SDPA (AOTriton)."""
import math
import os
import time

os.environ["RLA_FLASH"] = "1"  # force the custom path under test

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

# the fused path must be the one under test — fail loudly if the HIP
# extension did not load on a GPU box (no silent eager fallback)
from ray_lightning_amd import ops as _ops
assert _ops._load_ext() is not None, "HIP extension failed to load"



def test_mfma_probe_layout():
    """Verifies the assumed 16x16x32 bf16 fragment maps against
    torch.matmul with asymmetric operands (guide §3 A=I-check rule)."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(0)
    A = (torch.randn(16, 32, device="cuda") * 0.5).bfloat16()
    B = (torch.randn(32, 16, device="cuda") * 0.5).bfloat16()
    C = ext.mfma_probe(A, B)
    ref = (A.float() @ B.float())
    assert torch.allclose(C, ref, atol=3e-2, rtol=3e-2), \
        f"MFMA layout mismatch: max err {(C - ref).abs().max()}"


def _ref_attention(q, k, v, scale):
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    T = q.shape[2]
    mask = torch.tril(torch.ones(T, T, device=q.device, dtype=torch.bool))
    s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return p @ v.float()


@pytest.mark.parametrize("B,H,T", [(2, 3, 256), (1, 2, 1024), (1, 1, 128)])
def test_flash_fwd_numerics(B, H, T):
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(0)
    hs = 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(hs)
    o, lse = ext.flash_attn_fwd(q, k, v, scale)
    ref = _ref_attention(q, k, v, scale)
    assert torch.allclose(o.float(), ref, atol=3e-2, rtol=3e-2), \
        f"max err {(o.float() - ref).abs().max()}"
    # LSE check
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    mask = torch.tril(torch.ones(T, T, device=q.device,
                                 dtype=torch.bool))
    s = s.masked_fill(~mask, float("-inf"))
    ref_lse = torch.logsumexp(s, dim=-1)
    assert torch.allclose(lse, ref_lse, atol=2e-2, rtol=2e-2)


def test_flash_bwd_numerics():
    from ray_lightning_amd.ops.flash_attn import flash_attention
    torch.manual_seed(1)
    B, H, T, hs = 2, 3, 256, 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    for t in (q, k, v):
        t.requires_grad_(True)
    o = flash_attention(q, k, v)
    dy = torch.randn_like(o)
    o.backward(dy)

    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ref = _ref_attention(q2, k2, v2, 1.0 / math.sqrt(hs))
    ref.backward(dy.float())

    assert torch.allclose(o.float(), ref, atol=3e-2, rtol=3e-2)
    for got, exp, name in ((q.grad, q2.grad, "dq"),
                           (k.grad, k2.grad, "dk"),
                           (v.grad, v2.grad, "dv")):
        assert torch.allclose(got.float(), exp, atol=5e-2, rtol=5e-2), \
            f"{name} max err {(got.float() - exp).abs().max()}"


def test_flash_vs_sdpa_speed():
    """Times custom kernels vs AOTriton SDPA on the GPT-2-XL shape;
    prints both (informational — custom must be at least correct)."""
    from ray_lightning_amd.ops.flash_attn import flash_attention
    torch.manual_seed(2)
    B, H, T, hs = 8, 25, 1024, 64
    q = torch.randn(B, H, T, hs, device="cuda",
                    dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    dy = torch.randn_like(q)

    def run(fn, n=10):
        for _ in range(3):
            o = fn()
            o.backward(dy)
            q.grad = k.grad = v.grad = None
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            o = fn()
            o.backward(dy)
            q.grad = k.grad = v.grad = None
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e3

    t_custom = run(lambda: flash_attention(q, k, v))
    t_sdpa = run(lambda: torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True))
    print(f"\n[flash] custom {t_custom:.3f} ms  sdpa {t_sdpa:.3f} ms "
          f"(fwd+bwd, B{B} H{H} T{T})")
    # v3 must stay at least on par with AOTriton (it wins by ~4% as of
    # r02: 0.68 vs 0.71 ms); 15% headroom for box-to-box variance
    assert t_custom < t_sdpa * 1.15


@pytest.mark.xfail(reason="single-bpermute-per-pair scheme is "
                   "structurally incomplete: a source lane contributes "
                   "one register per ds_bpermute but targets need "
                   "lane-varying (q,n) registers — see "
                   "profiles/r01_flash_attn_notes.md round-2 design",
                   strict=False)
def test_perm_redistribution_probe():
    """C-layout -> A-fragment in-register redistribution probe (the
    primitive the round-2 flash redesign needs instead of the P LDS
    round-trip). Currently xfail: records the measured state and the
    confirmed v_perm_b32 byte-pool order."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(3)
    M = torch.randn(16, 64, device="cuda") * 0.5
    B = (torch.randn(64, 16, device="cuda") * 0.5).bfloat16()
    ref = M.bfloat16().float() @ B.float()
    errs = []
    for variant in (0, 1):
        C2 = ext.perm_probe(M, B, variant)
        errs.append(float((C2 - ref).abs().max()))
    print(f"\n[perm_probe] variant errs: {errs}")
    assert min(errs) < 3e-2, f"neither perm variant matches: {errs}"


@pytest.mark.parametrize("use_permlane", [False, True])
@pytest.mark.parametrize("B,H,T", [(2, 3, 256), (1, 2, 1024),
                                   (1, 1, 64), (1, 2, 2048)])
def test_flash_fwd_v3_numerics(B, H, T, use_permlane):
    """v3 (swapped-operand S^T, in-register P redistribution) against
    the fp32 reference; both redistribution paths."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(0)
    hs = 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(hs)
    o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, use_permlane)
    ref = _ref_attention(q, k, v, scale)
    assert torch.allclose(o.float(), ref, atol=3e-2, rtol=3e-2), \
        f"max err {(o.float() - ref).abs().max()}"
    # LSE against the fp32 reference logsumexp
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    mask = torch.tril(
        torch.ones(T, T, device=q.device, dtype=torch.bool))
    s = s.masked_fill(~mask, float("-inf"))
    lse_ref = torch.logsumexp(s, dim=-1)
    assert torch.allclose(lse, lse_ref, atol=2e-3, rtol=1e-3), \
        f"LSE max err {(lse - lse_ref).abs().max()}"


def test_flash_fwd_v3_matches_v2_lse():
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(1)
    q = (torch.randn(2, 2, 512, 64, device="cuda")).bfloat16()
    k = torch.randn_like(q).bfloat16()
    v = torch.randn_like(q).bfloat16()
    o2, lse2 = ext.flash_attn_fwd(q, k, v, 0.125)
    o3, lse3 = ext.flash_attn_fwd_v3(q, k, v, 0.125, False)
    assert torch.allclose(lse2, lse3, atol=2e-3, rtol=1e-3)
    assert torch.allclose(o2.float(), o3.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("use_permlane", [False, True])
@pytest.mark.parametrize("B,H,T", [(2, 3, 256), (1, 2, 1024),
                                   (1, 1, 64), (1, 2, 2048)])
def test_flash_bwd_v3_numerics(B, H, T, use_permlane):
    """v3 backward (swapped C layouts, in-register dS/P redistribution)
    against torch autograd through the fp32 reference."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(3)
    hs = 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    dy = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(hs)

    o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, use_permlane)
    dq, dk, dv = ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, scale,
                                       use_permlane)

    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    ref = _ref_attention(qf, kf, vf, scale)
    ref.backward(dy.float())
    for got, want, name in ((dq, qf.grad, "dq"), (dk, kf.grad, "dk"),
                            (dv, vf.grad, "dv")):
        err = (got.float() - want).abs().max().item()
        # fwd runs T13 defer-max (P bounded by 2^8 instead of 1), which
        # costs ~3x max-abs error in O/D per the guide's measurement —
        # the bwd inherits it through D = rowsum(dO*O)
        assert err < 2e-2 * math.sqrt(T / 64), f"{name} max err {err}"


def test_flash_fwd_v3_defer_max_spike():
    """T13 hazard test (guide §5.4 rule 26): an input engineered to
    FORCE the defer-rescale branch (a late spiked key that jumps the
    running max far beyond the threshold) must match the fp32
    reference exactly as well as random data does."""
    from ray_lightning_amd.ops import _load_ext
    ext = _load_ext()
    torch.manual_seed(11)
    B, H, T, hs = 1, 2, 512, 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.3).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.3).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.3).bfloat16()
    # spike: key row 400 aligned with every query -> at its tile the
    # raw score jumps by >> 8/log2(e)*scale exponent units
    k[:, :, 400, :] = (q[:, :, -1, :] * 40.0).bfloat16()
    scale = 1.0 / math.sqrt(hs)
    o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, True)
    ref = _ref_attention(q, k, v, scale)
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"defer-max spike: max err {err}"


def test_mfma_probe32_layout():
    """Verifies the assumed 32x32x16 bf16 fragment maps against
    torch.matmul with asymmetric operands (guide A=I-check rule)."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(5)
    A = (torch.randn(32, 16, device="cuda") * 0.5).bfloat16()
    B = (torch.randn(16, 32, device="cuda") * 0.5).bfloat16()
    C = ext.mfma_probe32(A, B)
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=3e-2, rtol=3e-2), \
        f"32x32x16 layout mismatch: max err {(C - ref).abs().max()}"


@pytest.mark.parametrize("B,H,T", [(2, 3, 256), (1, 2, 1024),
                                   (1, 1, 128)])
def test_flash_fwd_v4_numerics(B, H, T):
    """v4 (32x32x16 MFMA shape) against the fp32 reference + LSE."""
    from ray_lightning_amd import ops
    ext = ops._load_ext()
    torch.manual_seed(4)
    hs = 64
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    scale = 1.0 / math.sqrt(hs)
    o, lse = ext.flash_attn_fwd_v4(q, k, v, scale)
    ref = _ref_attention(q, k, v, scale)
    assert torch.allclose(o.float(), ref, atol=3e-2, rtol=3e-2), \
        f"max err {(o.float() - ref).abs().max()}"
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    mask = torch.tril(
        torch.ones(T, T, device=q.device, dtype=torch.bool))
    s = s.masked_fill(~mask, float("-inf"))
    lse_ref = torch.logsumexp(s, dim=-1)
    assert torch.allclose(lse, lse_ref, atol=2e-3, rtol=1e-3), \
        f"LSE max err {(lse - lse_ref).abs().max()}"


def test_flash_v3_bitwise_deterministic():
    """Two identical fwd+bwd invocations must agree bitwise — guards
    against LDS double-buffer / tr_read races that would show up as
    run-to-run jitter rather than systematic error."""
    from ray_lightning_amd.ops import _load_ext
    ext = _load_ext()
    torch.manual_seed(9)
    q = torch.randn(2, 3, 256, 64, device="cuda").bfloat16()
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    dy = torch.randn_like(q)
    outs = []
    for _ in range(3):
        o, lse = ext.flash_attn_fwd_v3(q, k, v, 0.125, True)
        dq, dk, dv = ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, 0.125,
                                           True)
        outs.append((o, lse, dq, dk, dv))
    for a, b in zip(outs[0], outs[1]):
        assert torch.equal(a, b)
    for a, b in zip(outs[0], outs[2]):
        assert torch.equal(a, b)
