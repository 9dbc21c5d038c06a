"""GroupNorm + MSE/MAE/Huber HIP kernels vs fp32 torch references
(reference groupnorm_ops.cu:46-171, loss_ops.cu:308-390)."""


import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tnn_amd import _C
    ext = _C.ext()

from tnn_amd import ops

DEV = "cuda"


def maxerr(a, b):
    return (a.float() - b.float()).abs().max().item()


def ref_gn(x, gamma, beta, groups, eps=1e-5):
    # torch group_norm is NCHW; permute our NHWC input
    xf = x.float().permute(0, 3, 1, 2) if x.dim() == 4 else x.float().transpose(-1, -2)
    y = F.group_norm(xf, groups, gamma.float(), beta.float(), eps)
    return y.permute(0, 2, 3, 1) if x.dim() == 4 else y.transpose(-1, -2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape,groups", [((4, 8, 8, 32), 8),
                                          ((2, 16, 16, 64), 4),
                                          ((3, 5, 7, 48), 6)])
def test_gn_fwd(dtype, shape, groups):
    torch.manual_seed(0)
    x = torch.randn(*shape, dtype=dtype, device=DEV) * 2 + 0.5
    gamma = torch.randn(shape[-1], device=DEV).abs() + 0.5
    beta = torch.randn(shape[-1], device=DEV)
    y = ops.group_norm(x, gamma, beta, groups)
    ref = ref_gn(x, gamma, beta, groups)
    assert maxerr(y, ref) < (1e-4 if dtype == torch.float32 else 5e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_gn_bwd(dtype):
    torch.manual_seed(1)
    N, H, W, C, G = 3, 6, 6, 32, 8
    x = torch.randn(N, H, W, C, dtype=dtype, device=DEV, requires_grad=True)
    gamma = (torch.randn(C, device=DEV).abs() + 0.5).requires_grad_()
    beta = torch.randn(C, device=DEV, requires_grad=True)
    y = ops.group_norm(x, gamma, beta, G)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().float().requires_grad_()
    g2 = gamma.detach().clone().requires_grad_()
    b2 = beta.detach().clone().requires_grad_()
    ref_gn(x2, g2, b2, G).backward(g.float())
    tol = 1e-3 if dtype == torch.float32 else 6e-2
    assert maxerr(x.grad, x2.grad) < tol
    assert maxerr(gamma.grad, g2.grad) < tol * 10
    assert maxerr(beta.grad, b2.grad) < tol * 10


def test_gn_layer_dispatches_hip():
    """GroupNorm layer on GPU must run the HIP kernel, not torch eager."""
    from tnn_amd.nn.layers import GroupNorm
    torch.manual_seed(2)
    layer = GroupNorm(4, 32, dtype=torch.bfloat16).to(DEV)
    x = torch.randn(2, 4, 4, 32, dtype=torch.bfloat16, device=DEV)
    y = layer(x)
    ref = ref_gn(x, layer.gamma, layer.beta, 4)
    assert maxerr(y, ref) < 5e-2


@pytest.mark.parametrize("kind,ref_fn", [
    ("mse", lambda p, t: F.mse_loss(p, t)),
    ("mae", lambda p, t: F.l1_loss(p, t)),
    ("huber", lambda p, t: F.huber_loss(p, t, delta=0.7)),
])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_pointwise_losses(kind, ref_fn, dtype):
    torch.manual_seed(3)
    pred = torch.randn(1000, 37, dtype=dtype, device=DEV, requires_grad=True)
    tgt = torch.randn(1000, 37, dtype=dtype, device=DEV)
    loss = ops.pointwise_loss(pred, tgt, kind, 0.7)
    loss.backward()

    p2 = pred.detach().float().requires_grad_()
    ref = ref_fn(p2, tgt.float())
    ref.backward()
    tol = 1e-4 if dtype == torch.float32 else 1e-2
    assert abs(loss.item() - ref.item()) < tol * 10
    assert maxerr(pred.grad, p2.grad) < tol
