"""Numerics for the hand-written fused NHWC BN(+ReLU)(+residual)
CDNA4 kernels vs a plain fp32 torch reference of the same op."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

# the fused path must be the one under test — fail loudly if the HIP
# extension did not load on a GPU box (no silent eager fallback)
from ray_lightning_amd import ops as _ops
assert _ops._load_ext() is not None, "HIP extension failed to load"



def _ref(x32, w, b, rm, rv, res32, relu, training, momentum, eps):
    """fp32 reference: torch BN + add + relu (keeps grads)."""
    y = torch.nn.functional.batch_norm(
        x32, rm, rv, w, b, training=training, momentum=momentum, eps=eps)
    if res32 is not None:
        y = y + res32
    if relu:
        y = torch.relu(y)
    return y


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize("relu,with_res", [(True, False), (False, False),
                                           (True, True)])
def test_fused_bn_train_fwd_bwd(dtype, relu, with_res):
    from ray_lightning_amd.ops.fused_bn import FusedBatchNorm2d
    torch.manual_seed(0)
    N, C, H, W = 8, 64, 14, 14
    dev = "cuda"
    x = torch.randn(N, C, H, W, device=dev).to(dtype).to(
        memory_format=torch.channels_last).requires_grad_(True)
    res = None
    if with_res:
        res = torch.randn(N, C, H, W, device=dev).to(dtype).to(
            memory_format=torch.channels_last).requires_grad_(True)

    bn = FusedBatchNorm2d(C, relu=relu).to(dev)
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)
    bn.train()

    y = bn(x, res) if with_res else bn(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    # fp32 reference with identical init
    x32 = x.detach().float().requires_grad_(True)
    res32 = res.detach().float().requires_grad_(True) if with_res else None
    w32 = bn.weight.detach().clone().requires_grad_(True)
    b32 = bn.bias.detach().clone().requires_grad_(True)
    rm = torch.zeros(C, device=dev)
    rv = torch.ones(C, device=dev)
    y32 = _ref(x32, w32, b32, rm, rv, res32, relu, True, bn.momentum,
               bn.eps)
    y32.backward(dy.float())

    tol = dict(atol=5e-2, rtol=5e-2) if dtype == torch.bfloat16 else \
        dict(atol=1e-4, rtol=1e-4)
    assert torch.allclose(y.float(), y32, **tol), "forward mismatch"
    assert torch.allclose(x.grad.float(), x32.grad, **tol), "dx mismatch"
    # channel reductions: compare with slightly looser atol (N*H*W sums)
    rtol = dict(atol=2e-1, rtol=2e-2) if dtype == torch.bfloat16 else \
        dict(atol=1e-3, rtol=1e-4)
    assert torch.allclose(bn.weight.grad, w32.grad, **rtol), "dgamma"
    assert torch.allclose(bn.bias.grad, b32.grad, **rtol), "dbeta"
    if with_res:
        assert torch.allclose(res.grad.float(), res32.grad, **tol)
    # running stats updated to batch stats
    assert torch.allclose(bn.running_mean, rm, atol=5e-2, rtol=5e-2)
    assert torch.allclose(bn.running_var, rv, atol=5e-2, rtol=5e-2)


def test_fused_bn_eval_matches_running_stats():
    from ray_lightning_amd.ops.fused_bn import FusedBatchNorm2d
    torch.manual_seed(1)
    C = 128
    bn = FusedBatchNorm2d(C, relu=True).to("cuda")
    with torch.no_grad():
        bn.running_mean.normal_()
        bn.running_var.uniform_(0.5, 2.0)
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)
    bn.eval()
    x = torch.randn(4, C, 7, 7, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    with torch.no_grad():
        y = bn(x)
        ref = torch.relu(torch.nn.functional.batch_norm(
            x.float(), bn.running_mean, bn.running_var, bn.weight,
            bn.bias, training=False, eps=bn.eps))
    assert torch.allclose(y.float(), ref, atol=5e-2, rtol=5e-2)


def test_fused_bn_big_channels():
    """C=2048 (ResNet-50 layer4): one full row per wavefront-block."""
    from ray_lightning_amd.ops.fused_bn import FusedBatchNorm2d
    torch.manual_seed(2)
    C = 2048
    bn = FusedBatchNorm2d(C, relu=True).to("cuda").train()
    x = torch.randn(4, C, 7, 7, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    y = bn(x)
    y.mean().backward()
    x32 = x.detach().float().requires_grad_(True)
    rm, rv = torch.zeros(C, device="cuda"), torch.ones(C, device="cuda")
    y32 = torch.relu(torch.nn.functional.batch_norm(
        x32, rm, rv, bn.weight.detach(), bn.bias.detach(), training=True,
        momentum=bn.momentum, eps=bn.eps))
    y32.mean().backward()
    assert torch.allclose(y.float(), y32, atol=5e-2, rtol=5e-2)
    assert torch.allclose(x.grad.float(), x32.grad, atol=5e-2, rtol=5e-2)
