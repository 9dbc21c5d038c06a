"""Batched MFMA GEMM + fused scaled/causal softmax + fused decode
attention vs fp32 torch references (reference gemm_strided_batched_ex,
src/math/cuda/gemm.cu:84; softmax.cu:47-79; causal_mask.cu:13)."""


import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tnn_amd import _C
    ext = _C.ext()

from tnn_amd import ops

DEV = "cuda"


def maxerr(a, b):
    return (a.float() - b.float()).abs().max().item()


def tol(dtype, K):
    return 3e-6 * K if dtype == torch.float32 else 2e-2 * (K ** 0.5)


# ---------------------------------------------------------------------------
# bmm: all four operand layouts, both dtypes, ragged shapes
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(6, 64, 64, 64), (4, 128, 96, 64),
                                   (3, 57, 130, 33), (2, 512, 512, 64),
                                   (12, 1, 64, 64)])
def test_bmm_layouts(dtype, shape):
    torch.manual_seed(0)
    B, M, N, K = shape
    a = torch.randn(B, M, K, dtype=dtype, device=DEV)
    b = torch.randn(B, K, N, dtype=dtype, device=DEV)
    ref = torch.bmm(a.float(), b.float())
    t = tol(dtype, K)
    # NN
    assert maxerr(ext.bmm(a, b), ref) < t
    # NT: b passed as a transpose view of its [N, K] storage
    bt = b.transpose(-1, -2).contiguous().transpose(-1, -2)
    assert maxerr(ext.bmm(a, bt), ref) < t
    # TN: a passed as a transpose view
    at = a.transpose(-1, -2).contiguous().transpose(-1, -2)
    assert maxerr(ext.bmm(at, b), ref) < t
    # TT
    assert maxerr(ext.bmm(at, bt), ref) < t


def test_bmm_noncontig_fallback():
    torch.manual_seed(1)
    a = torch.randn(2, 64, 128, dtype=torch.bfloat16, device=DEV)[:, :, ::2]
    b = torch.randn(2, 64, 32, dtype=torch.bfloat16, device=DEV)
    out = ext.bmm(a, b)
    assert maxerr(out, torch.bmm(a.float(), b.float())) < tol(torch.bfloat16, 64)


def test_bmm_autograd_matches_torch():
    torch.manual_seed(2)
    B, M, N, K = 3, 96, 80, 64
    a = torch.randn(B, M, K, dtype=torch.bfloat16, device=DEV, requires_grad=True)
    b = torch.randn(B, K, N, dtype=torch.bfloat16, device=DEV, requires_grad=True)
    c = ops.matmul(a, b)
    g = torch.randn_like(c)
    c.backward(g)
    a2 = a.detach().float().requires_grad_()
    b2 = b.detach().float().requires_grad_()
    torch.bmm(a2, b2).backward(g.float())
    assert maxerr(a.grad, a2.grad) < tol(torch.bfloat16, N)
    assert maxerr(b.grad, b2.grad) < tol(torch.bfloat16, M)


# ---------------------------------------------------------------------------
# scaled / causal softmax
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("cols", [64, 500, 1024, 4096, 8192])
def test_softmax_fwd(dtype, cols):
    torch.manual_seed(0)
    x = torch.randn(3, 7, cols, dtype=dtype, device=DEV) * 4
    y = ext.smax_fwd(x, 7, 0, 0.5, False)
    ref = torch.softmax(x.float() * 0.5, dim=-1)
    assert maxerr(y, ref) < 2e-3


@pytest.mark.parametrize("qoff", [0, 5])
def test_softmax_causal(qoff):
    torch.manual_seed(0)
    B, M = 4, 96
    C = M + qoff
    x = torch.randn(B, M, C, dtype=torch.bfloat16, device=DEV) * 3
    y = ext.smax_fwd(x, M, qoff, 0.25, True)
    pos_q = torch.arange(M, device=DEV).unsqueeze(-1) + qoff
    pos_k = torch.arange(C, device=DEV)
    xf = (x.float() * 0.25).masked_fill(pos_k > pos_q, float("-inf"))
    ref = torch.softmax(xf, dim=-1)
    assert maxerr(y, ref) < 2e-3
    # masked tail exactly zero
    assert (y.float() * (pos_k > pos_q)).abs().max().item() == 0.0


def test_softmax_fully_masked_row_is_zero():
    x = torch.randn(1, 4, 10, dtype=torch.float32, device=DEV)
    # qoff = -2: rows 0 and 1 fully masked
    y = ext.smax_fwd(x, 4, -2, 1.0, True)
    assert y[0, 0].abs().sum().item() == 0.0
    assert y[0, 1].abs().sum().item() == 0.0
    assert abs(y[0, 2].sum().item() - 1.0) < 1e-5


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_softmax_bwd(dtype):
    torch.manual_seed(3)
    R, C = 64, 768
    x = torch.randn(R, C, dtype=dtype, device=DEV, requires_grad=False) * 2
    scale = 0.125
    p = ext.smax_fwd(x, R, 0, scale, False)
    dy = torch.randn(R, C, dtype=dtype, device=DEV)
    dx = ext.smax_bwd(p, dy, scale)
    x2 = x.detach().float().requires_grad_()
    torch.softmax(x2 * scale, dim=-1).backward(dy.float())
    assert maxerr(dx, x2.grad) < 2e-3


def test_scaled_softmax_autograd():
    torch.manual_seed(4)
    x = torch.randn(6, 32, 32, dtype=torch.bfloat16, device=DEV,
                    requires_grad=True)
    y = ops.scaled_softmax(x, 0.2, causal=True)
    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().float().requires_grad_()
    y2 = ops.scaled_softmax(x2.cpu(), 0.2, causal=True)  # CPU oracle path
    y2.backward(g.float().cpu())
    assert maxerr(y.cpu(), y2) < 2e-3
    assert maxerr(x.grad, x2.grad) < 2e-3


# ---------------------------------------------------------------------------
# sdpa_materialized: end-to-end fwd+bwd vs fp32 reference
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("S,D", [(128, 64), (192, 128), (96, 64)])
def test_sdpa_materialized(S, D, causal):
    torch.manual_seed(5)
    B, H = 2, 3
    mk = lambda: torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV,
                             requires_grad=True)
    q, k, v = mk(), mk(), mk()
    o = ops.sdpa_materialized(q, k, v, causal=causal)
    g = torch.randn_like(o)
    o.backward(g)

    scale = D ** -0.5
    q2 = q.detach().float().requires_grad_()
    k2 = k.detach().float().requires_grad_()
    v2 = v.detach().float().requires_grad_()
    s = (q2 @ k2.transpose(-1, -2)) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=DEV).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    (torch.softmax(s, dim=-1) @ v2).backward(g.float())
    ref = (torch.softmax(s, dim=-1) @ v2).detach()
    assert maxerr(o, ref) < 5e-2
    assert maxerr(q.grad, q2.grad) < 6e-2
    assert maxerr(k.grad, k2.grad) < 6e-2
    assert maxerr(v.grad, v2.grad) < 6e-2


# ---------------------------------------------------------------------------
# fused decode attention vs full-recompute reference
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("cap,length", [(64, 1), (1024, 700), (2048, 2048)])
def test_attn_decode(D, cap, length):
    torch.manual_seed(6)
    BH = 7
    q = torch.randn(BH, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(BH, cap, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(BH, cap, D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    o = ext.attn_decode(q, k, v, None, length, scale)
    s = (q.float().unsqueeze(1) @ k[:, :length].float().transpose(-1, -2)) * scale
    ref = (torch.softmax(s, dim=-1) @ v[:, :length].float()).squeeze(1)
    assert maxerr(o, ref) < 3e-2


def test_attn_decode_pos_tensor():
    torch.manual_seed(7)
    BH, D, cap = 12, 64, 512
    q = torch.randn(BH, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(BH, cap, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(BH, cap, D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    pos = torch.tensor([299], dtype=torch.int64, device=DEV)  # len = 300
    o = ext.attn_decode(q, k, v, pos, 0, scale)
    ref_o = ext.attn_decode(q, k, v, None, 300, scale)
    assert torch.equal(o, ref_o)


# ---------------------------------------------------------------------------
# fused-residual linear + single-launch decode GEMV
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("M", [1, 2, 128])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_linear_residual_fused(M, dtype):
    torch.manual_seed(8)
    K, N = 768, 768
    x = torch.randn(M, K, dtype=dtype, device=DEV, requires_grad=True)
    w = (torch.randn(K, N, dtype=dtype, device=DEV) * 0.02).requires_grad_()
    b = torch.randn(N, dtype=dtype, device=DEV, requires_grad=True)
    r = torch.randn(M, N, dtype=dtype, device=DEV, requires_grad=True)
    y = ops.linear(x, w, b, residual=r)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().float().requires_grad_()
    w2 = w.detach().float().requires_grad_()
    b2 = b.detach().float().requires_grad_()
    r2 = r.detach().float().requires_grad_()
    (x2 @ w2 + b2 + r2).backward(g.float())
    t = tol(dtype, K)
    assert maxerr(y, (x.detach().float() @ w.detach().float()
                      + b.detach().float() + r.detach().float())) < t
    assert maxerr(r.grad, r2.grad) < 1e-5
    assert maxerr(x.grad, x2.grad) < t
    assert maxerr(w.grad, w2.grad) < t * 2


@pytest.mark.parametrize("N,K", [(768, 768), (3072, 768), (768, 3072),
                                 (50257, 768), (65, 100)])
def test_gemv_single_launch_m1(N, K):
    torch.manual_seed(9)
    x = torch.randn(1, K, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(K, N, dtype=torch.bfloat16, device=DEV) * 0.05
    b = torch.randn(N, dtype=torch.bfloat16, device=DEV)
    y = ext.gemm(x, w, b, 0)
    ref = x.float() @ w.float() + b.float()
    assert maxerr(y, ref) < tol(torch.bfloat16, K)


@pytest.mark.parametrize("act", ["linear", "relu", "tanh"])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_add_act_fused(act, dtype):
    torch.manual_seed(10)
    a = torch.randn(1000, 37, dtype=dtype, device=DEV, requires_grad=True)
    b = torch.randn(1000, 37, dtype=dtype, device=DEV, requires_grad=True)
    y = ops.add_act(a, b, act)
    g = torch.randn_like(y)
    y.backward(g)
    a2 = a.detach().float().requires_grad_()
    b2 = b.detach().float().requires_grad_()
    z = a2 + b2
    z = {"linear": z, "relu": torch.relu(z), "tanh": torch.tanh(z)}[act]
    z.backward(g.float())
    t = 1e-5 if dtype == torch.float32 else 2e-2
    assert maxerr(y, z.detach()) < t
    assert maxerr(a.grad, a2.grad) < t
    assert maxerr(b.grad, b2.grad) < t


def test_gemm_nt_huge_k_split():
    """vocab-head dgrad shape: K-split fp32 slabs path"""
    torch.manual_seed(11)
    M, K, N = 256, 16384, 128
    a = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
    b = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
    c = ext.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    assert maxerr(c, ref) < tol(torch.bfloat16, K)
