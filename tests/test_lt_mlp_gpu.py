"""Fused hipBLASLt-epilogue MLP vs the composed torch reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

import os

os.environ["RLA_LT_MLP"] = "1"  # experimental paths under test
os.environ["RLA_FUSED_MLP"] = "1"

from ray_lightning_amd.ops.lt_mlp import _load, fused_mlp

assert _load() is not None, "_lt_mlp extension failed to load"


@pytest.mark.parametrize("M,C,F", [(256, 256, 1024), (1024, 1600, 6400),
                                   (64, 128, 512)])
def test_lt_mlp_matches_reference(M, C, F):
    torch.manual_seed(0)
    x = (torch.randn(M, C, device="cuda") * 0.5).bfloat16()
    W1 = (torch.randn(F, C, device="cuda") * 0.02).bfloat16()
    b1 = torch.randn(F, device="cuda").bfloat16() * 0.1
    W2 = (torch.randn(C, F, device="cuda") * 0.02).bfloat16()
    b2 = torch.randn(C, device="cuda").bfloat16() * 0.1

    xs = [t.clone().requires_grad_(True) for t in (x, W1, b1, W2, b2)]
    y = fused_mlp(*xs)
    dy = torch.randn_like(y) * 0.5
    y.backward(dy)

    rs = [t.clone().float().requires_grad_(True)
          for t in (x, W1, b1, W2, b2)]
    ref = torch.nn.functional.linear(
        torch.nn.functional.gelu(
            torch.nn.functional.linear(rs[0], rs[1], rs[2]),
            approximate="tanh"), rs[3], rs[4])
    ref.backward(dy.float())

    assert torch.allclose(y.float(), ref, atol=8e-2, rtol=4e-2), \
        f"y max err {(y.float() - ref).abs().max()}"
    names = ["dx", "dW1", "db1", "dW2", "db2"]
    for name, got, want in zip(names, xs, rs):
        g, w = got.grad.float(), want.grad
        scale = w.abs().max().item() + 1e-6
        err = (g - w).abs().max().item()
        assert err < 5e-2 * max(scale, 1.0), \
            f"{name}: max err {err} (scale {scale})"


def test_lt_mlp_in_gpt2_block():
    """GPT-2 tiny model trains identically-shaped with the fused MLP
    on vs off (gradient agreement)."""
    import os

    from ray_lightning_amd.models.gpt2 import (GPT2, GPT2Config,
                                               to_bf16_training)
    torch.manual_seed(2)
    cfg = GPT2Config(vocab_size=512, n_positions=128, n_embd=128,
                     n_layer=2, n_head=2)
    model = to_bf16_training(GPT2(cfg).cuda())
    x = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda")
    t = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda")

    def run(flag):
        os.environ["RLA_LT_MLP"] = flag
        model.zero_grad(set_to_none=True)
        _, loss = model(x, t)
        loss.backward()
        return float(loss), {n: p.grad.float().clone()
                             for n, p in model.named_parameters()
                             if p.grad is not None}

    try:
        lf, gf = run("1")
        ls, gs = run("0")
    finally:
        os.environ["RLA_LT_MLP"] = "1"
    assert abs(lf - ls) < 3e-2
    for n in gf:
        scale = gs[n].abs().max().item() + 1e-6
        err = (gf[n] - gs[n]).abs().max().item()
        assert err < 7e-2 * max(scale, 1.0), f"{n}: {err}"


def test_lt_linear_matches_reference():
    from ray_lightning_amd.ops.lt_mlp import lt_linear
    torch.manual_seed(3)
    M, C, N = 512, 256, 768
    x = (torch.randn(M, C, device="cuda") * 0.5).bfloat16()
    W = (torch.randn(N, C, device="cuda") * 0.05).bfloat16()
    b = torch.randn(N, device="cuda").bfloat16() * 0.1
    xs = [t.clone().requires_grad_(True) for t in (x, W, b)]
    y = lt_linear(*xs)
    dy = torch.randn_like(y) * 0.5
    y.backward(dy)
    rs = [t.clone().float().requires_grad_(True) for t in (x, W, b)]
    ref = torch.nn.functional.linear(*rs)
    ref.backward(dy.float())
    assert torch.allclose(y.float(), ref, atol=5e-2, rtol=3e-2)
    for name, got, want in zip(("dx", "dW", "db"), xs, rs):
        scale = want.grad.abs().max().item() + 1e-6
        err = (got.grad.float() - want.grad).abs().max().item()
        assert err < 4e-2 * max(scale, 1.0), f"{name}: {err}"


def test_fused_dgelu_bgrad_numerics():
    """Hand-written fused dGELU+bias-grad vs the composed reference."""
    from ray_lightning_amd.ops import _load_ext
    ext = _load_ext()
    torch.manual_seed(5)
    M, F = 1024, 800
    dh = (torch.randn(M, F, device="cuda") * 0.5).bfloat16()
    z = (torch.randn(M, F, device="cuda")).bfloat16()
    dz, db = ext.fused_dgelu_bgrad(dh, z)
    zf = z.float().requires_grad_(True)
    hf = torch.nn.functional.gelu(zf, approximate="tanh")
    hf.backward(dh.float())
    ref_dz = zf.grad
    assert torch.allclose(dz.float(), ref_dz, atol=2e-2, rtol=2e-2), \
        f"dz max err {(dz.float() - ref_dz).abs().max()}"
    ref_db = ref_dz.sum(dim=0)
    scale = ref_db.abs().max().item() + 1e-6
    err = (db.float() - ref_db).abs().max().item()
    assert err < 3e-2 * scale, f"db max err {err} scale {scale}"


def test_fused_mlp_autograd_matches_reference():
    from ray_lightning_amd.ops.mlp import fused_mlp
    torch.manual_seed(6)
    M, C, F = 512, 256, 1024
    x = (torch.randn(M, C, device="cuda") * 0.5).bfloat16()
    W1 = (torch.randn(F, C, device="cuda") * 0.05).bfloat16()
    b1 = torch.randn(F, device="cuda").bfloat16() * 0.1
    W2 = (torch.randn(C, F, device="cuda") * 0.05).bfloat16()
    b2 = torch.randn(C, device="cuda").bfloat16() * 0.1
    xs = [t.clone().requires_grad_(True) for t in (x, W1, b1, W2, b2)]
    y = fused_mlp(*xs)
    dy = torch.randn_like(y) * 0.5
    y.backward(dy)
    rs = [t.clone().float().requires_grad_(True)
          for t in (x, W1, b1, W2, b2)]
    ref = torch.nn.functional.linear(
        torch.nn.functional.gelu(
            torch.nn.functional.linear(rs[0], rs[1], rs[2]),
            approximate="tanh"), rs[3], rs[4])
    ref.backward(dy.float())
    assert torch.allclose(y.float(), ref, atol=8e-2, rtol=4e-2)
    for name, got, want in zip(("dx", "dW1", "db1", "dW2", "db2"),
                               xs, rs):
        scale = want.grad.abs().max().item() + 1e-6
        err = (got.grad.float() - want.grad).abs().max().item()
        assert err < 5e-2 * max(scale, 1.0), f"{name}: {err}"


def teardown_module():
    # don't leak the experimental opt-ins into later test modules
    os.environ.pop("RLA_LT_MLP", None)
    os.environ.pop("RLA_FUSED_MLP", None)
