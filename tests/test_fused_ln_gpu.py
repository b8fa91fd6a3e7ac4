"""Numerics for the fused residual-add+LayerNorm CDNA4 kernels vs
plain fp32 torch reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

# the fused path must be the one under test — fail loudly if the HIP
# extension did not load on a GPU box (no silent eager fallback)
from ray_lightning_amd import ops as _ops
assert _ops._load_ext() is not None, "HIP extension failed to load"



@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("C", [1600, 768, 2048])
def test_fused_ln_plain(dtype, C):
    from ray_lightning_amd.ops.fused_ln import FusedLayerNorm
    torch.manual_seed(0)
    R = 512
    x = (torch.randn(4, R // 4, C, device="cuda") * 2).to(
        dtype).requires_grad_(True)
    ln = FusedLayerNorm(C).to("cuda").float()
    with torch.no_grad():
        ln.weight.uniform_(0.5, 1.5)
        ln.bias.uniform_(-0.5, 0.5)
    y = ln(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    x32 = x.detach().float().requires_grad_(True)
    w32 = ln.weight.detach().clone().requires_grad_(True)
    b32 = ln.bias.detach().clone().requires_grad_(True)
    y32 = torch.nn.functional.layer_norm(x32, (C,), w32, b32, ln.eps)
    y32.backward(dy.float())

    tol = dict(atol=5e-2, rtol=5e-2) if dtype == torch.bfloat16 else \
        dict(atol=1e-4, rtol=1e-4)
    assert torch.allclose(y.float(), y32, **tol)
    assert torch.allclose(x.grad.float(), x32.grad, **tol)
    rtol = dict(atol=5e-1, rtol=3e-2) if dtype == torch.bfloat16 else \
        dict(atol=1e-2, rtol=1e-4)
    assert torch.allclose(ln.weight.grad, w32.grad, **rtol)
    assert torch.allclose(ln.bias.grad, b32.grad, **rtol)


def test_fused_add_ln_chain():
    """x_new = residual + sub; y = ln(x_new); grads flow into both
    inputs and into x_new's other consumer."""
    from ray_lightning_amd.ops.fused_ln import FusedLayerNorm
    torch.manual_seed(1)
    C, R = 1600, 256
    res = torch.randn(R, C, device="cuda", dtype=torch.bfloat16
                      ).requires_grad_(True)
    sub = torch.randn(R, C, device="cuda", dtype=torch.bfloat16
                      ).requires_grad_(True)
    ln = FusedLayerNorm(C).to("cuda").float()
    x_new, y = ln(sub, residual=res)
    # x_new feeds a second consumer (like the next residual chain)
    loss = y.float().pow(2).mean() + x_new.float().mean()
    loss.backward()

    res32 = res.detach().float().requires_grad_(True)
    sub32 = sub.detach().float().requires_grad_(True)
    xn32 = res32 + sub32
    y32 = torch.nn.functional.layer_norm(
        xn32, (C,), ln.weight.detach().double().float(),
        ln.bias.detach().clone(), ln.eps)
    (y32.pow(2).mean() + xn32.mean()).backward()

    assert torch.allclose(x_new.float(), xn32, atol=5e-2, rtol=5e-2)
    assert torch.allclose(y.float(), y32, atol=5e-2, rtol=5e-2)
    assert torch.allclose(res.grad.float(), res32.grad, atol=5e-2,
                          rtol=5e-2)
    assert torch.allclose(sub.grad.float(), sub32.grad, atol=5e-2,
                          rtol=5e-2)


def test_gpt2_fused_chain_matches_eager_blocks():
    """GPT2.forward's cross-block fused chain == the unfused Block
    composition on the same weights."""
    from ray_lightning_amd.models.gpt2 import GPT2, GPT2Config
    torch.manual_seed(2)
    cfg = GPT2Config(vocab_size=503, n_positions=64, n_embd=256,
                     n_layer=3, n_head=4)
    m = GPT2(cfg).to("cuda")
    idx = torch.randint(0, 503, (2, 32), device="cuda")
    logits, _ = m(idx)

    # eager reference: plain Block.forward composition
    x = m.wte(idx) + m.wpe(torch.arange(32, device="cuda"))
    for block in m.h:
        x = x + block.attn(torch.nn.functional.layer_norm(
            x, (256,), block.ln_1.weight, block.ln_1.bias,
            block.ln_1.eps))
        x = x + block.mlp(torch.nn.functional.layer_norm(
            x, (256,), block.ln_2.weight, block.ln_2.bias,
            block.ln_2.eps))
    x = torch.nn.functional.layer_norm(
        x, (256,), m.ln_f.weight, m.ln_f.bias, m.ln_f.eps)
    ref = m.lm_head(x)
    assert torch.allclose(logits, ref, atol=1e-3, rtol=1e-3)
