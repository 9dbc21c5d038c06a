"""HIP kernel numerics vs the fp32 PyTorch reference (reference test
strategy: kernel fixtures with tolerance + CPU/GPU parity suites,
unit_tests/cuda_*_test.cpp, layer_device_agnosticity_test.cpp:25).

Everything here needs an MI355X; the suite is the main payload of
`pytest -m gpu` on the GPU box.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tnn_amd import _C
    ext = _C.ext()

DEV = "cuda"


def relerr(a, b):
    a, b = a.float(), b.float()
    return ((a - b).abs() / (b.abs().clamp_min(1.0))).max().item()


TOL = {torch.float32: 2e-4, torch.bfloat16: 3e-2}
DTYPES = [torch.float32, torch.bfloat16]


# ---------------------------------------------------------------------------
# MFMA fragment-layout ground truth (asymmetric operands; guide §3)
# ---------------------------------------------------------------------------

def test_mfma_selftest_bf16():
    torch.manual_seed(0)
    a = (torch.randn(16, 32) * 0.5).bfloat16().to(DEV)
    b = (torch.arange(32 * 16).reshape(32, 16) % 7 - 3).bfloat16().to(DEV)
    d = ext.mfma_selftest(a.contiguous(), b.contiguous())
    ref = a.float().cpu() @ b.float().cpu()
    assert torch.allclose(d.cpu(), ref, atol=1e-2, rtol=1e-2), \
        f"bf16 MFMA fragment layout wrong:\n{d.cpu()[:4,:4]}\nvs\n{ref[:4,:4]}"


def test_mfma_selftest_f32():
    torch.manual_seed(0)
    a = torch.randn(16, 4, device=DEV)
    b = torch.randn(4, 16, device=DEV)
    d = ext.mfma_selftest_f32(a.contiguous(), b.contiguous())
    ref = a.cpu() @ b.cpu()
    assert torch.allclose(d.cpu(), ref, atol=1e-5), "f32 MFMA layout wrong"


# ---------------------------------------------------------------------------
# GEMM family
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("cols", [16, 100, 128, 2304, 50264])
def test_colsum(dtype, cols):
    # 2304 spans >1 channel-block in the vectorized kernel (regression:
    # the final per-channel loop wrote past cols without a bound check)
    torch.manual_seed(11)
    x = torch.randn(777, cols, dtype=dtype, device=DEV)
    out = ext.colsum(x)
    ref = x.float().sum(0)
    assert relerr(out, ref) < 1e-2, cols


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("M,N,K", [(128, 64, 32), (256, 128, 512),
                                   (100, 10, 27), (64, 100, 512),
                                   (130, 70, 33),
                                   (1, 3072, 768), (1, 97, 33)])
def test_gemm_nn(dtype, M, N, K):
    torch.manual_seed(1)
    a = torch.randn(M, K, dtype=dtype, device=DEV)
    b = torch.randn(K, N, dtype=dtype, device=DEV)
    bias = torch.randn(N, dtype=dtype, device=DEV)
    c = ext.gemm(a, b, bias, 0)
    ref = a.float() @ b.float() + bias.float()
    assert relerr(c, ref) < TOL[dtype] * max(1, K // 128), (M, N, K)


@pytest.mark.parametrize("dtype", DTYPES)
def test_gemm_relu_epilogue(dtype):
    a = torch.randn(64, 32, dtype=dtype, device=DEV)
    b = torch.randn(32, 48, dtype=dtype, device=DEV)
    c = ext.gemm(a, b, None, 1)  # relu
    ref = (a.float() @ b.float()).clamp_min(0)
    assert relerr(c, ref) < TOL[dtype]
    assert (c >= 0).all()


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("M,N,K", [(128, 64, 32), (100, 512, 100), (64, 27, 130),
                                   (1, 3072, 768), (1, 100, 33)])
def test_gemm_nt(dtype, M, N, K):
    torch.manual_seed(2)
    a = torch.randn(M, K, dtype=dtype, device=DEV)
    b = torch.randn(N, K, dtype=dtype, device=DEV)
    c = ext.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    assert relerr(c, ref) < TOL[dtype] * max(1, K // 128)


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("M,N,K", [(512, 64, 128), (10000, 100, 512),
                                   (130, 33, 27)])
def test_gemm_tn(dtype, M, N, K):
    torch.manual_seed(3)
    a = torch.randn(M, K, dtype=dtype, device=DEV)
    b = torch.randn(M, N, dtype=dtype, device=DEV)
    c = ext.gemm_tn(a, b, False)  # fp32 out
    ref = a.float().t() @ b.float()
    tol = TOL[dtype] * max(1, M // 256)
    assert relerr(c, ref) < tol, relerr(c, ref)


# ---------------------------------------------------------------------------
# conv2d implicit GEMM (shapes from WRN-16-8 / ResNet-9)
# ---------------------------------------------------------------------------

CONV_CASES = [
    # N, H, W, Cin, Cout, KH, KW, S, P
    (4, 32, 32, 3, 16, 3, 3, 1, 1),     # stem (scalar gather path)
    (4, 32, 32, 16, 128, 3, 3, 1, 1),   # group1 entry
    (4, 32, 32, 128, 128, 3, 3, 1, 1),  # main body
    (4, 32, 32, 128, 256, 3, 3, 2, 1),  # stride-2 downsample
    (4, 32, 32, 128, 256, 1, 1, 2, 0),  # 1x1 shortcut
    (2, 8, 8, 512, 512, 3, 3, 1, 1),    # deep layer
    (2, 9, 9, 24, 40, 3, 3, 2, 1),      # odd sizes
    (2, 56, 56, 64, 64, 3, 3, 1, 1),    # non-pow2 spatial (imagenet-style)
    (2, 28, 28, 96, 192, 3, 3, 2, 1),   # non-pow2 channels + stride
]


def _conv_ref(x, w, bias, s, p):
    y = torch.nn.functional.conv2d(
        x.permute(0, 3, 1, 2).float(), w.permute(3, 2, 0, 1).float(),
        None if bias is None else bias.float(), stride=s, padding=p)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_fwd(dtype, case):
    N, H, W, Ci, Co, KH, KW, S, P = case
    torch.manual_seed(4)
    x = torch.randn(N, H, W, Ci, dtype=dtype, device=DEV)
    w = torch.randn(KH, KW, Ci, Co, dtype=dtype, device=DEV) * 0.1
    bias = torch.randn(Co, dtype=dtype, device=DEV)
    y = ext.conv2d_fwd(x, w, bias, S, S, P, P, False)
    ref = _conv_ref(x, w, bias, S, P)
    tol = TOL[dtype] * max(1, (KH * KW * Ci) // 256)
    assert relerr(y, ref.to(DEV)) < tol, (case, relerr(y, ref.to(DEV)))


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_dgrad(dtype, case):
    N, H, W, Ci, Co, KH, KW, S, P = case
    torch.manual_seed(5)
    OH = (H + 2 * P - KH) // S + 1
    OW = (W + 2 * P - KW) // S + 1
    dy = torch.randn(N, OH, OW, Co, dtype=dtype, device=DEV)
    w = torch.randn(KH, KW, Ci, Co, dtype=dtype, device=DEV) * 0.1
    dx = ext.conv2d_dgrad(dy, w, H, W, S, S, P, P)
    ref = torch.nn.grad.conv2d_input(
        (N, Ci, H, W), w.permute(3, 2, 0, 1).float().cpu(),
        dy.permute(0, 3, 1, 2).float().cpu(), stride=S, padding=P)
    ref = ref.permute(0, 2, 3, 1)
    tol = TOL[dtype] * max(1, (KH * KW * Co) // 256)
    assert relerr(dx.cpu(), ref) < tol, (case, relerr(dx.cpu(), ref))


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_wgrad(dtype, case):
    N, H, W, Ci, Co, KH, KW, S, P = case
    torch.manual_seed(6)
    OH = (H + 2 * P - KH) // S + 1
    OW = (W + 2 * P - KW) // S + 1
    x = torch.randn(N, H, W, Ci, dtype=dtype, device=DEV)
    dy = torch.randn(N, OH, OW, Co, dtype=dtype, device=DEV)
    dw = ext.conv2d_wgrad(x, dy, KH, KW, S, S, P, P, False)  # [KH,KW,Ci,Co] fp32
    ref = torch.nn.grad.conv2d_weight(
        x.permute(0, 3, 1, 2).float().cpu(), (Co, Ci, KH, KW),
        dy.permute(0, 3, 1, 2).float().cpu(), stride=S, padding=P)
    ref = ref.permute(2, 3, 1, 0)  # -> [KH,KW,Ci,Co]
    tol = TOL[dtype] * max(1, (N * OH * OW) // 256)
    assert relerr(dw.cpu(), ref) < tol, (case, relerr(dw.cpu(), ref))


# ---------------------------------------------------------------------------
# batch norm
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("C", [16, 100, 128, 512])
def test_bn_fwd_train(dtype, C):
    torch.manual_seed(7)
    x = torch.randn(8, 6, 6, C, dtype=dtype, device=DEV) * 2 + 0.5
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)
    y, mean, invstd = ext.bn_fwd_train(x, gamma, beta, None, None, 0.1, 1e-5, False, 0.0, 0, None)
    xf = x.float().reshape(-1, C)
    rmean = xf.mean(0)
    rvar = xf.var(0, unbiased=False)
    ref = ((xf - rmean) / (rvar + 1e-5).sqrt() * gamma + beta).reshape(x.shape)
    assert relerr(mean, rmean) < 1e-3
    assert relerr(y, ref) < TOL[dtype] * 2


@pytest.mark.parametrize("dtype", DTYPES)
def test_bn_relu_and_infer(dtype):
    C = 64
    x = torch.randn(4, 5, 5, C, dtype=dtype, device=DEV)
    gamma = torch.ones(C, device=DEV)
    beta = torch.zeros(C, device=DEV)
    y, _, _ = ext.bn_fwd_train(x, gamma, beta, None, None, 0.1, 1e-5, True, 0.0, 0, None)
    assert (y.float() >= 0).all()
    rm = torch.randn(C, device=DEV) * 0.1
    rv = torch.rand(C, device=DEV) + 0.5
    y2 = ext.bn_fwd_infer(x, gamma, beta, rm, rv, 1e-5, False)
    ref = (x.float() - rm) / (rv + 1e-5).sqrt()
    assert relerr(y2, ref) < TOL[dtype] * 2


@pytest.mark.parametrize("dtype", DTYPES)
def test_bn_bwd_matches_autograd(dtype):
    torch.manual_seed(8)
    C = 32
    x = torch.randn(6, 4, 4, C, device=DEV)
    gamma = (torch.rand(C, device=DEV) + 0.5).requires_grad_(True)
    beta = torch.randn(C, device=DEV).requires_grad_(True)
    xg = x.clone().requires_grad_(True)
    xf = xg.reshape(-1, C)
    mean = xf.mean(0)
    var = xf.var(0, unbiased=False)
    y = ((xf - mean) / (var + 1e-5).sqrt() * gamma + beta).reshape(x.shape)
    dy = torch.randn_like(y)
    y.backward(dy)
    xdt = x.to(dtype)
    y_k, mean_k, invstd_k = ext.bn_fwd_train(xdt, gamma.detach(), beta.detach(),
                                             None, None, 0.1, 1e-5, False, 0.0,
                                             0, None)
    dx, dgamma, dbeta = ext.bn_bwd(xdt, dy.to(dtype), gamma.detach(), mean_k,
                                   invstd_k, None, 1.0)
    tol = TOL[dtype] * 4
    assert relerr(dx, xg.grad) < tol
    assert relerr(dgamma, gamma.grad) < tol
    assert relerr(dbeta, beta.grad) < tol


# ---------------------------------------------------------------------------
# pooling / dropout / activations
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("cfg", [(2, 2, 2, 2, 0, 0), (3, 3, 2, 2, 1, 1),
                                 (3, 3, 3, 3, 0, 0)])
def test_maxpool(dtype, cfg):
    kh, kw, sh, sw, ph, pw = cfg
    torch.manual_seed(9)
    x = torch.randn(3, 12, 12, 24, dtype=dtype, device=DEV)
    y, idx = ext.maxpool_fwd(x, kh, kw, sh, sw, ph, pw)
    xn = x.permute(0, 3, 1, 2).float()
    ref = torch.nn.functional.max_pool2d(xn, (kh, kw), (sh, sw), (ph, pw))
    assert relerr(y, ref.permute(0, 2, 3, 1)) < TOL[dtype]
    # backward
    dy = torch.randn_like(y)
    dx = ext.maxpool_bwd(dy, idx, 12, 12)
    xr = xn.clone().requires_grad_(True)
    torch.nn.functional.max_pool2d(xr, (kh, kw), (sh, sw), (ph, pw)).backward(
        dy.permute(0, 3, 1, 2).float())
    assert relerr(dx, xr.grad.permute(0, 2, 3, 1)) < TOL[dtype]


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("cfg", [(2, 2, 2, 2, 0, 0), (8, 8, 1, 1, 0, 0)])
def test_avgpool(dtype, cfg):
    kh, kw, sh, sw, ph, pw = cfg
    x = torch.randn(3, 8, 8, 20, dtype=dtype, device=DEV)
    y = ext.avgpool_fwd(x, kh, kw, sh, sw, ph, pw)
    xn = x.permute(0, 3, 1, 2).float()
    ref = torch.nn.functional.avg_pool2d(xn, (kh, kw), (sh, sw), (ph, pw))
    assert relerr(y, ref.permute(0, 2, 3, 1)) < TOL[dtype]
    dy = torch.randn_like(y)
    dx = ext.avgpool_bwd(dy, 8, 8, kh, kw, sh, sw, ph, pw)
    xr = xn.clone().requires_grad_(True)
    torch.nn.functional.avg_pool2d(xr, (kh, kw), (sh, sw), (ph, pw)).backward(
        dy.permute(0, 3, 1, 2).float())
    assert relerr(dx, xr.grad.permute(0, 2, 3, 1)) < TOL[dtype]


def test_dropout_stats_and_replay():
    x = torch.ones(1 << 20, device=DEV)
    y, mask = ext.dropout_fwd(x, 0.3, 12345)
    keep = mask.float().mean().item()
    assert abs(keep - 0.7) < 0.01
    assert relerr(y.sum() / x.numel(), torch.tensor(1.0)) < 0.02
    dy = torch.ones_like(x)
    dx = ext.dropout_bwd(dy, mask, 0.3)
    assert torch.equal(dx != 0, y != 0)
    # same seed reproduces the mask
    _, mask2 = ext.dropout_fwd(x, 0.3, 12345)
    assert torch.equal(mask, mask2)


@pytest.mark.parametrize("kind,name", [(1, "relu"), (2, "gelu"), (3, "sigmoid"),
                                       (4, "tanh"), (5, "elu"),
                                       (6, "leaky_relu"), (7, "silu")])
def test_activations(kind, name):
    torch.manual_seed(10)
    x = torch.randn(4096, device=DEV)
    y = ext.act_fwd(x, kind)
    import torch.nn.functional as F
    refs = {1: F.relu, 2: lambda t: F.gelu(t, approximate="tanh"),
            3: torch.sigmoid, 4: torch.tanh, 5: F.elu,
            6: lambda t: F.leaky_relu(t, 0.01), 7: F.silu}
    ref = refs[kind](x.float())
    assert relerr(y, ref) < 1e-4, name
    xr = x.clone().requires_grad_(True)
    refs[kind](xr).backward(torch.ones_like(x))
    dx = ext.act_bwd(torch.ones_like(x), x, y, kind)
    assert relerr(dx, xr.grad) < 1e-4, name


# ---------------------------------------------------------------------------
# loss / layernorm / embedding
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("C", [10, 100, 50257])
def test_ce(dtype, C):
    torch.manual_seed(11)
    logits = (torch.randn(32, C, dtype=dtype, device=DEV) * 2)
    targets = torch.randint(0, C, (32,), device=DEV)
    loss, lse = ext.ce_fwd(logits, targets)
    lr = logits.float().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lr, targets, reduction="none")
    assert relerr(loss, ref) < TOL[dtype]
    dloss = torch.rand(32, device=DEV)
    (ref * dloss).sum().backward()
    dl = ext.ce_bwd(logits, targets, lse, dloss)
    assert relerr(dl, lr.grad) < TOL[dtype]


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("D", [64, 768, 1280])
def test_layernorm(dtype, D):
    torch.manual_seed(12)
    x = torch.randn(64, D, dtype=dtype, device=DEV)
    gamma = torch.rand(D, device=DEV) + 0.5
    beta = torch.randn(D, device=DEV)
    y, mean, invstd = ext.ln_fwd(x, gamma, beta, 1e-5)
    xr = x.float().clone().requires_grad_(True)
    gr = gamma.clone().requires_grad_(True)
    br = beta.clone().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xr, (D,), gr, br, 1e-5)
    assert relerr(y, ref) < TOL[dtype] * 2
    dy = torch.randn_like(ref)
    ref.backward(dy)
    dx, dgamma, dbeta = ext.ln_bwd(x, dy.to(dtype), gamma, mean, invstd)
    assert relerr(dx, xr.grad) < TOL[dtype] * 4
    assert relerr(dgamma, gr.grad) < TOL[dtype] * 4
    assert relerr(dbeta, br.grad) < TOL[dtype] * 4


@pytest.mark.parametrize("dtype", DTYPES)
def test_embedding(dtype):
    table = torch.randn(1000, 64, dtype=dtype, device=DEV)
    ids = torch.randint(0, 1000, (8, 32), device=DEV)
    y = ext.embedding_fwd(ids, table)
    assert torch.equal(y, table[ids])
    dy = torch.randn(8, 32, 64, dtype=dtype, device=DEV)
    dt = ext.embedding_bwd(ids, dy, 1000)
    ref = torch.zeros(1000, 64, device=DEV)
    ref.index_add_(0, ids.reshape(-1), dy.reshape(-1, 64).float())
    assert relerr(dt, ref) < TOL[dtype]


# ---------------------------------------------------------------------------
# optimizers (GPU kernels vs our CPU implementation)
# ---------------------------------------------------------------------------

def test_adamw_gpu_matches_cpu():
    from tnn_amd.nn import optim
    torch.manual_seed(13)
    init = torch.randn(1000)
    grads = [torch.randn(1000) for _ in range(5)]
    p_cpu = torch.nn.Parameter(init.clone())
    o_cpu = optim.AdamW([p_cpu], lr=0.01, weight_decay=0.01)
    p_gpu = torch.nn.Parameter(init.clone().to(DEV))
    o_gpu = optim.AdamW([p_gpu], lr=0.01, weight_decay=0.01)
    for g in grads:
        p_cpu.grad = g.clone()
        p_gpu.grad = g.clone().to(DEV)
        o_cpu.step()
        o_gpu.step()
    assert relerr(p_gpu.cpu(), p_cpu) < 1e-5


def test_adamw_gpu_bf16_master():
    from tnn_amd.nn import optim
    torch.manual_seed(14)
    init = torch.randn(512)
    p_cpu = torch.nn.Parameter(init.bfloat16())
    o_cpu = optim.AdamW([p_cpu], lr=0.05)
    p_gpu = torch.nn.Parameter(init.bfloat16().to(DEV))
    o_gpu = optim.AdamW([p_gpu], lr=0.05)
    for _ in range(5):
        g = torch.randn(512).bfloat16()
        p_cpu.grad = g.clone()
        p_gpu.grad = g.clone().to(DEV)
        o_cpu.step()
        o_gpu.step()
    assert relerr(o_gpu.state[0]["master"].cpu(), o_cpu.state[0]["master"]) < 1e-5
    assert torch.equal(p_gpu.cpu(), p_cpu)


def test_sgd_gpu_matches_cpu():
    from tnn_amd.nn import optim
    torch.manual_seed(15)
    init = torch.randn(777)
    p_cpu = torch.nn.Parameter(init.clone())
    o_cpu = optim.SGD([p_cpu], lr=0.1, momentum=0.9, weight_decay=1e-4,
                      nesterov=True)
    p_gpu = torch.nn.Parameter(init.clone().to(DEV))
    o_gpu = optim.SGD([p_gpu], lr=0.1, momentum=0.9, weight_decay=1e-4,
                      nesterov=True)
    for _ in range(5):
        g = torch.randn(777)
        p_cpu.grad = g.clone()
        p_gpu.grad = g.clone().to(DEV)
        o_cpu.step()
        o_gpu.step()
    assert relerr(p_gpu.cpu(), p_cpu) < 1e-5


# ---------------------------------------------------------------------------
# end-to-end: full layers on GPU vs CPU (device-agnosticity oracle)
# ---------------------------------------------------------------------------

def test_wrn_block_gpu_vs_cpu():
    from tnn_amd.nn import LayerBuilder
    torch.manual_seed(16)
    model = (LayerBuilder((16, 16, 32))
             .wide_residual_block(32, 64, 2, 0.0, "wb")
             .batchnorm(relu=True, name="bnf")
             .avgpool2d(8, 8)
             .flatten()
             .dense(10, True, "fc")
             .build("slice"))
    model.eval()
    x = torch.randn(4, 16, 16, 32)
    y_cpu = model(x)
    model_gpu = model.to(DEV)
    y_gpu = model_gpu(x.to(DEV))
    assert relerr(y_gpu.cpu(), y_cpu) < 5e-3


def test_training_step_reduces_loss_gpu():
    from tnn_amd import models
    from tnn_amd.nn import CrossEntropyLoss, AdamW
    torch.manual_seed(17)
    model = models.create_model("cifar10_resnet9").to(DEV).train()
    opt = AdamW(model.parameters(), lr=3e-4)
    crit = CrossEntropyLoss()
    x = torch.randn(32, 32, 32, 3, device=DEV)
    y = torch.randint(0, 10, (32,), device=DEV)
    losses = []
    for _ in range(10):
        out = model(x)
        loss = crit(out, y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, losses


def test_wrn_converges_on_learnable_task():
    """End-to-end convergence on GPU: quadrant-energy classification must
    reach high accuracy within a few hundred steps (the reference's
    convergence evidence is CIFAR logs; no datasets exist in this image)."""
    from tnn_amd.nn import LayerBuilder, CrossEntropyLoss, AdamW
    from tnn_amd.nn.layer import cast_compute_dtype
    torch.manual_seed(42)
    model = (LayerBuilder((16, 16, 3))
             .conv2d(32, 3, 3, 1, 1, 1, 1, True, "c1")
             .batchnorm(relu=True, name="b1")
             .wide_residual_block(32, 64, 2, 0.0, "wb1")
             .batchnorm(relu=True, name="bf")
             .avgpool2d(8, 8)
             .flatten()
             .dense(4, True, "fc")
             .build("quadnet"))
    cast_compute_dtype(model, torch.bfloat16)
    model.to(DEV).train()
    opt = AdamW(model.parameters(), lr=2e-3)
    crit = CrossEntropyLoss()
    n = 1024
    x = torch.randn(n, 16, 16, 3).abs()
    y = torch.stack([x[:, :8, :8].sum((1, 2, 3)), x[:, :8, 8:].sum((1, 2, 3)),
                     x[:, 8:, :8].sum((1, 2, 3)), x[:, 8:, 8:].sum((1, 2, 3))],
                    1).argmax(1)
    xg, yg = x.bfloat16().to(DEV), y.to(DEV)
    acc = 0.0
    for epoch in range(30):
        for i in range(0, n, 256):
            xb, yb = xg[i:i + 256], yg[i:i + 256]
            out = model(xb)
            loss = crit(out, yb)
            opt.zero_grad()
            loss.backward()
            opt.step()
    model.eval()
    with torch.no_grad():
        pred = model(xg).argmax(-1)
        acc = (pred == yg).float().mean().item()
    assert acc > 0.9, f"did not converge: acc={acc}"


def test_graphed_inference_matches_eager():
    """hipGraph capture of an eval forward (utils/graphstep.py): the
    replayed graph must match eager outputs for fresh inputs."""
    from tnn_amd import models
    from tnn_amd.nn.layer import cast_compute_dtype
    from tnn_amd.utils.graphstep import GraphedInference
    torch.manual_seed(21)
    m = models.create_model("cifar10_resnet9")
    cast_compute_dtype(m, torch.bfloat16)
    m.to(DEV).eval()
    x0 = torch.randn(16, 32, 32, 3, dtype=torch.bfloat16, device=DEV)
    g = GraphedInference(m, x0)
    x1 = torch.randn_like(x0)
    with torch.no_grad():
        ref = m(x1)
    out = g(x1)
    assert torch.equal(out, ref)


def test_generate_graphed_gpu_matches_recompute():
    # real hipGraph capture of the decode step vs full recompute
    from tnn_amd.nn import LayerBuilder
    from tnn_amd.models.generate import generate, generate_graphed
    torch.manual_seed(3)
    m = (LayerBuilder((32,))
         .embedding(96, 64, "tok")
         .positional_embedding(32, "pos")
         .gpt_block(8, 2, flash=False, name="b0")
         .gpt_block(8, 2, flash=False, name="b1")
         .layernorm(name="ln_f")
         .dense(96, True, "head")
         .build("tiny_gpt")).to(DEV)
    ref = generate(m, [1, 2, 3], max_new_tokens=12, seq_len=32,
                   eot_token=None, device=torch.device(DEV))
    fast = generate_graphed(m, [1, 2, 3], max_new_tokens=12, seq_len=32,
                            eot_token=None, device=torch.device(DEV))
    assert fast == ref


@pytest.mark.parametrize("dtype", DTYPES)
def test_bn_relu_dropout_fused(dtype):
    # fused BN+ReLU+dropout: dropped positions are exactly 0, kept positions
    # are relu(bn(x))/keep; backward = autograd with the realized mask
    torch.manual_seed(9)
    p = 0.4
    N, C = 4096, 128
    x = torch.randn(N, C, dtype=dtype, device=DEV)
    gamma = torch.rand(C, device=DEV, dtype=torch.float32) + 0.5
    beta = torch.randn(C, device=DEV, dtype=torch.float32)
    dy = torch.randn(N, C, dtype=dtype, device=DEV)

    xr = x.detach().clone().requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    from tnn_amd.ops.functional import _BatchNormAct
    y, _, _ = _BatchNormAct.apply(xr, gr, br, None, None, 0.1, 1e-5, True,
                                  p, 1234)
    # reference bn+relu
    xf = x.float()
    mean = xf.mean(0)
    var = xf.var(0, unbiased=False)
    ref = ((xf - mean) / torch.sqrt(var + 1e-5)) * gamma + beta
    ref = torch.relu(ref)
    kept = y.float() != 0.0
    frac = kept.float().mean().item()
    pos_frac = (ref > 0).float().mean().item()
    assert abs(frac - pos_frac * (1 - p)) < 0.02
    assert relerr(y.float()[kept], (ref / (1 - p))[kept]) < TOL[dtype] * 2

    # backward vs autograd using the realized mask
    y.backward(dy)
    xa = x.detach().clone().float().requires_grad_(True)
    ga = gamma.detach().clone().requires_grad_(True)
    ba = beta.detach().clone().requires_grad_(True)
    m2 = xa.mean(0)
    v2 = xa.var(0, unbiased=False)
    ya = torch.relu(((xa - m2) / torch.sqrt(v2 + 1e-5)) * ga + ba)
    ya = ya * kept.float() / (1 - p)
    ya.backward(dy.float())
    tol = TOL[dtype] * 8
    assert relerr(xr.grad, xa.grad) < tol
    assert relerr(gr.grad, ga.grad) < tol
    assert relerr(br.grad, ba.grad) < tol


@pytest.mark.parametrize("cfg", [(256, 16, 16, 128, 128), (256, 8, 8, 256, 512)])
def test_conv_fwd_fused_bn_stats(cfg):
    # conv epilogue per-channel sum/sumsq vs reductions of the conv output
    N, H, W, Ci, Co = cfg
    torch.manual_seed(12)
    x = torch.randn(N, H, W, Ci, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(3, 3, Ci, Co, dtype=torch.bfloat16, device=DEV) * 0.05
    bias = torch.randn(Co, dtype=torch.bfloat16, device=DEV)
    y, stats = ext.conv2d_fwd_stats(x, w, bias, 1, 1, 1, 1)
    assert stats.numel() == 2 * Co, "DB path expected for this shape"
    yf = y.float().reshape(-1, Co)
    assert relerr(stats[0], yf.sum(0)) < 2e-2
    assert relerr(stats[1], (yf * yf).sum(0)) < 2e-2


def test_wrn_block_fused_peepholes_match_unfused():
    # Sequential.forward (peepholes active: conv->BN stats fusion) vs
    # calling each layer directly (no peepholes). Stats come from fp32
    # atomics in a different order, so compare with bf16-level tolerance.
    from tnn_amd.nn import LayerBuilder
    torch.manual_seed(13)
    m = (LayerBuilder((16, 16, 128))
         .batchnorm(relu=True, name="bn1")
         .conv2d(128, 3, 3, 1, 1, 1, 1, True, "conv1")
         .batchnorm(relu=True, name="bn2")
         .conv2d(128, 3, 3, 1, 1, 1, 1, True, "conv2")
         .build("seq_net"))
    from tnn_amd.nn.layer import cast_compute_dtype
    cast_compute_dtype(m, torch.bfloat16)
    m = m.to(DEV)
    m.train()
    x = torch.randn(32, 16, 16, 128, dtype=torch.bfloat16, device=DEV)

    y_fused = m(x)
    loss = y_fused.float().square().mean()
    loss.backward()
    g_fused = [p.grad.clone() for p in m.parameters() if p.grad is not None]
    for p in m.parameters():
        p.grad = None

    h = x
    for layer in m.layers:   # direct layer calls skip the peepholes
        h = layer(h)
    loss2 = h.float().square().mean()
    loss2.backward()
    g_plain = [p.grad.clone() for p in m.parameters() if p.grad is not None]

    assert relerr(y_fused, h) < 2e-2
    assert len(g_fused) == len(g_plain)
    for a, b in zip(g_fused, g_plain):
        assert relerr(a, b) < 5e-2
