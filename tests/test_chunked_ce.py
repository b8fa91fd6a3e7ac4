"""Chunked linear+CE vs the naive full-logits reference (CPU, fp32 and
bf16-input paths)."""
import pytest
import torch

from ray_lightning_amd.ops.chunked_ce import chunked_cross_entropy


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_matches_full_cross_entropy(dtype):
    torch.manual_seed(0)
    n, c, v = 64, 32, 97
    x = (torch.randn(n, c) * 0.5).to(dtype).requires_grad_(True)
    w = (torch.randn(v, c) * 0.1).to(dtype).requires_grad_(True)
    t = torch.randint(0, v, (n,))

    loss = chunked_cross_entropy(x, w, t, chunk_rows=17)
    loss.backward()

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy((x2 @ w2.t()).float(), t)
    ref.backward()

    tol = dict(atol=1e-5, rtol=1e-5) if dtype == torch.float32 else \
        dict(atol=2e-2, rtol=2e-2)
    assert torch.allclose(loss, ref, **tol)
    assert torch.allclose(x.grad.float(), x2.grad.float(), **tol)
    assert torch.allclose(w.grad.float(), w2.grad.float(), **tol)


def test_gpt2_forward_uses_chunked_path():
    from ray_lightning_amd.models.gpt2 import GPT2, GPT2Config
    torch.manual_seed(1)
    cfg = GPT2Config(vocab_size=211, n_positions=32, n_embd=32,
                     n_layer=2, n_head=2)
    m = GPT2(cfg)
    x = torch.randint(0, 211, (2, 16))
    y = torch.randint(0, 211, (2, 16))
    logits, loss = m(x, y)
    assert logits is None and loss.dim() == 0
    loss.backward()
    assert m.wte.weight.grad is not None
    # inference path still produces logits
    logits, loss = m(x)
    assert loss is None and logits.shape == (2, 16, 211)
