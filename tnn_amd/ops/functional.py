"""Functional ops with CPU(torch-composition) / GPU(HIP kernel) dispatch.

GPU autograd Functions call into the in-tree HIP extension; CPU paths are
pure differentiable torch compositions used as the numerics oracle
(reference unit_tests/layer_device_agnosticity_test.cpp:25 pattern).

All image ops are NHWC: x is a contiguous ``[N, H, W, C]`` tensor and conv
weights are ``[KH, KW, Cin, Cout]`` (the implicit-GEMM B-operand layout).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from .. import _C

# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

ACT_KINDS = {
    "linear": 0,
    "relu": 1,
    "gelu": 2,       # tanh approximation (reference gelu_kernels.cu:18)
    "sigmoid": 3,
    "tanh": 4,
    "elu": 5,
    "leaky_relu": 6,
    "silu": 7,
}


def _nhwc_to_nchw(x: torch.Tensor) -> torch.Tensor:
    return x.permute(0, 3, 1, 2)


def _nchw_to_nhwc(x: torch.Tensor) -> torch.Tensor:
    return x.permute(0, 2, 3, 1).contiguous()


def _use_hip(*tensors: torch.Tensor) -> bool:
    return any(t is not None and t.is_cuda for t in tensors)


# hipGraph-captured training steps: RNG consumers combine a per-call salt
# (baked at capture) with this device counter (incremented in-graph), so
# every replay draws a fresh Philox stream with zero host involvement.
_graph_ctr: Optional[torch.Tensor] = None


def set_graph_seed_ctr(t: Optional[torch.Tensor]):
    global _graph_ctr
    _graph_ctr = t


def _draw_seed() -> int:
    if _graph_ctr is not None:
        import random
        return random.getrandbits(62)  # salt only; no GPU sync in capture
    return int(torch.randint(0, 2 ** 62, (1,)).item())


# ---------------------------------------------------------------------------
# conv2d NHWC (implicit GEMM on GPU; reference legacy_conv2d_layer.cpp:137)
# ---------------------------------------------------------------------------


class _Conv2dNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, padding, fuse_relu, want_stats=False):
        ext = _C.ext()
        # narrow-channel inputs (the RGB stem) fall off every vector/glds
        # path (measured 192us vs ~35 for the padded form): zero-pad Cin to
        # the 16B vector width once -- zero channels contribute nothing
        Ci = x.shape[-1]
        V = 8 if x.dtype == torch.bfloat16 else 4
        ctx.cin = Ci
        if Ci % V != 0 and Ci < V:
            x = F.pad(x, (0, V - Ci))
            w = F.pad(w, (0, 0, 0, V - Ci))
        ctx.want_stats = want_stats
        if want_stats:
            # fused per-channel sum/sumsq for a following train-mode BN;
            # stats is empty when the fused kernel is not eligible
            y, stats = ext.conv2d_fwd_stats(x, w, bias, stride[0], stride[1],
                                            padding[0], padding[1])
        else:
            y = ext.conv2d_fwd(x, w, bias, stride[0], stride[1], padding[0],
                               padding[1], fuse_relu)
        ctx.save_for_backward(x, w, y if fuse_relu else None)
        ctx.stride, ctx.padding, ctx.fuse_relu = stride, padding, fuse_relu
        ctx.has_bias = bias is not None
        if want_stats:
            ctx.mark_non_differentiable(stats)
            return y, stats
        return y

    @staticmethod
    def backward(ctx, dy, *unused_stats_grad):
        x, w, y = ctx.saved_tensors
        ext = _C.ext()
        dy = dy.contiguous()
        if ctx.fuse_relu:
            dy = ext.relu_bwd_mask(dy, y)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = ext.conv2d_dgrad(dy, w, x.shape[1], x.shape[2],
                                  ctx.stride[0], ctx.stride[1], ctx.padding[0], ctx.padding[1])
            if x.shape[-1] != ctx.cin:
                dx = dx[..., :ctx.cin].contiguous()
        if ctx.needs_input_grad[1]:
            dw = ext.conv2d_wgrad(x, dy, w.shape[0], w.shape[1],
                                  ctx.stride[0], ctx.stride[1], ctx.padding[0],
                                  ctx.padding[1], w.dtype == x.dtype)
            if x.shape[-1] != ctx.cin:
                dw = dw[:, :, :ctx.cin].contiguous()
            dw = dw.to(w.dtype)  # no-op when the reduce already cast
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = ext.colsum(dy.reshape(-1, dy.shape[-1]))
        return dx, dw, db, None, None, None, None


def conv2d_nhwc(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    stride: Tuple[int, int] = (1, 1),
    padding: Tuple[int, int] = (0, 0),
    fuse_relu: bool = False,
    want_stats: bool = False,
):
    """2-D convolution, NHWC activations, ``[KH,KW,Cin,Cout]`` weights.

    ``want_stats=True`` (GPU, linear act) additionally returns the fused
    per-channel (sum, sumsq) for a following train-mode BatchNorm, or None
    when the fused kernel is not eligible."""
    if _use_hip(x):
        if want_stats:
            y, stats = _Conv2dNHWC.apply(
                x.contiguous(), w.contiguous(),
                None if bias is None else bias.contiguous(), stride, padding,
                False, True)
            return y, (stats if stats.numel() else None)
        return _Conv2dNHWC.apply(x.contiguous(), w.contiguous(),
                                 None if bias is None else bias.contiguous(),
                                 stride, padding, fuse_relu)
    wn = w.permute(3, 2, 0, 1)  # -> [Cout, Cin, KH, KW]
    y = F.conv2d(_nhwc_to_nchw(x), wn, bias, stride=stride, padding=padding)
    if want_stats:
        return _nchw_to_nhwc(y), None
    y = _nchw_to_nhwc(y)
    if fuse_relu:
        y = F.relu(y)
    return y


# ---------------------------------------------------------------------------
# batch norm (+ fused ReLU) NHWC; stats always fp32
# (reference src/nn/layers_impl/batchnorm_layer.cpp:133-149)
# ---------------------------------------------------------------------------


class _BatchNormAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum,
                eps, relu, dropout_p=0.0, seed=0, precomp=None, ctr=None):
        ext = _C.ext()
        # running stats update fused into the finalize kernel; dropout (if
        # any) fused into the apply kernel -- dropped positions write 0, so
        # the saved output doubles as relu mask and dropout mask and the
        # backward is a constant 1/(1-p) scale on the kept positions
        y, mean, invstd = ext.bn_fwd_train(x, gamma, beta, running_mean,
                                           running_var, momentum, eps, relu,
                                           dropout_p, seed, precomp, ctr)
        save_y = y if (relu or dropout_p > 0.0) else None
        ctx.save_for_backward(x, gamma, mean, invstd, save_y)
        ctx.relu = relu or dropout_p > 0.0
        ctx.dy_scale = 1.0 / (1.0 - dropout_p) if dropout_p > 0.0 else 1.0
        # without this autograd materializes zero grads for mean/invstd on
        # every backward (2 fill launches per BN per step)
        ctx.mark_non_differentiable(mean, invstd)
        return y, mean, invstd

    @staticmethod
    def backward(ctx, dy, _dmean, _dinvstd):
        x, gamma, mean, invstd, y = ctx.saved_tensors
        ext = _C.ext()
        dx, dgamma, dbeta = ext.bn_bwd(x, dy.contiguous(), gamma, mean, invstd,
                                       y if ctx.relu else None, ctx.dy_scale)
        return (dx, dgamma, dbeta, None, None, None, None, None, None, None,
                None, None)


class _BatchNormActRes(torch.autograd.Function):
    """_BatchNormAct plus a residual PASSTHROUGH output (pre-activation
    residual blocks: the block input's junction grad join d_x = bn_grad +
    d_residual runs inside the bn_bwd apply kernel — no autograd fan-in
    add pass; mirrors _LayerNormRes)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum,
                eps, relu, dropout_p, seed, precomp, ctr):
        ext = _C.ext()
        y, mean, invstd = ext.bn_fwd_train(x, gamma, beta, running_mean,
                                           running_var, momentum, eps, relu,
                                           dropout_p, seed, precomp, ctr)
        save_y = y if (relu or dropout_p > 0.0) else None
        ctx.save_for_backward(x, gamma, mean, invstd, save_y)
        ctx.relu = relu or dropout_p > 0.0
        ctx.dy_scale = 1.0 / (1.0 - dropout_p) if dropout_p > 0.0 else 1.0
        ctx.mark_non_differentiable(mean, invstd)
        ctx.set_materialize_grads(False)
        return y, mean, invstd, x.view_as(x)

    @staticmethod
    def backward(ctx, dy, _dm, _di, dres):
        x, gamma, mean, invstd, y = ctx.saved_tensors
        ext = _C.ext()
        if dy is None:
            return (dres, None, None, None, None, None, None, None, None,
                    None, None, None)
        if dres is not None:
            dres = dres.contiguous()
        dx, dgamma, dbeta = ext.bn_bwd(x, dy.contiguous(), gamma, mean,
                                       invstd, y if ctx.relu else None,
                                       ctx.dy_scale, resid=dres)
        return (dx, dgamma, dbeta, None, None, None, None, None, None, None,
                None, None)


def batch_norm_act(
    x: torch.Tensor,
    gamma: torch.Tensor,
    beta: torch.Tensor,
    running_mean: torch.Tensor,
    running_var: torch.Tensor,
    training: bool,
    momentum: float = 0.1,
    eps: float = 1e-5,
    relu: bool = False,
    dropout_p: float = 0.0,
    precomputed: Optional[torch.Tensor] = None,
    passthrough: bool = False,
):
    """BatchNorm over NHWC channels-last with optional fused ReLU and
    (train-time) fused dropout.

    Returns y. Updates running stats in-place when training (fp32, unbiased
    variance, matching torch semantics). gamma/beta/running stats are fp32
    regardless of x dtype (reference batchnorm_layer.cpp:140-148).
    dropout_p > 0 (requires relu) folds the dropout mask + 1/(1-p) scale
    into the BN apply kernel -- one memory pass instead of three.
    ``passthrough=True`` additionally returns x as a residual-branch
    alias whose grad joins inside bn_bwd (see _BatchNormActRes).
    """
    C = x.shape[-1]
    n = x.numel() // C
    if _use_hip(x):
        if training:
            seed = _draw_seed() if dropout_p > 0.0 else 0
            if passthrough:
                y, _, _, xr = _BatchNormActRes.apply(
                    x.contiguous(), gamma, beta, running_mean, running_var,
                    momentum, eps, relu, dropout_p, seed, precomputed,
                    _graph_ctr)
                return y, xr
            y, _, _ = _BatchNormAct.apply(x.contiguous(), gamma, beta,
                                          running_mean, running_var, momentum,
                                          eps, relu, dropout_p, seed,
                                          precomputed, _graph_ctr)
            return y
        ext = _C.ext()
        return ext.bn_fwd_infer(x.contiguous(), gamma, beta, running_mean, running_var,
                                eps, relu)
    # CPU reference path
    xf = x.reshape(-1, C).float()
    if training:
        mean = xf.mean(0)
        var = xf.var(0, unbiased=False)
        with torch.no_grad():
            running_mean.mul_(1 - momentum).add_(mean.detach(), alpha=momentum)
            running_var.mul_(1 - momentum).add_(var.detach() * (n / max(n - 1, 1)),
                                                alpha=momentum)
    else:
        mean, var = running_mean, running_var
    xhat = (x.float() - mean) / torch.sqrt(var + eps)
    y = xhat * gamma + beta
    if relu:
        y = F.relu(y)
    if dropout_p > 0.0 and training:
        y = F.dropout(y, dropout_p, training=True)
    y = y.to(x.dtype)
    return (y, x) if passthrough else y


# ---------------------------------------------------------------------------
# dense / GEMM (reference src/nn/layers_impl/cuda/dense_ops.cu:17-117)
# ---------------------------------------------------------------------------


class _Linear(torch.autograd.Function):
    """y = act(x @ w + b) [+ residual] — the residual add rides the GEMM
    epilogue (one fused kernel instead of a separate at::native
    elementwise pass; reference dense_ops.cu has no analog)."""

    @staticmethod
    def forward(ctx, x, w, bias, act, residual=None):
        ext = _C.ext()
        assert residual is None or act == "linear", \
            "fused residual requires act='linear' (relu mask would include it)"
        y = ext.gemm(x, w, bias, ACT_KINDS[act], residual)
        ctx.save_for_backward(x, w, y if act != "linear" else None)
        ctx.act = act
        ctx.has_bias = bias is not None
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        ext = _C.ext()
        dy = dy.contiguous()
        dres = dy if ctx.has_res else None
        if ctx.act == "relu":
            dy = ext.relu_bwd_mask(dy, y)
        elif ctx.act != "linear":
            raise NotImplementedError(f"fused act bwd for {ctx.act}")
        dx = dw = db = None
        N = dy.shape[-1]
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = ext.colsum(dy)
        # ragged N (e.g. the gpt2 vocab head, N=50257) forces every backward
        # GEMM onto the scalar-gather fallback (~75 TF vs ~450). Zero-pad dy
        # and w along N once: identical contraction, vector/glds kernels
        # throughout (measured 5.7x on the head's dgrad at seq 512).
        V = 8 if dy.dtype == torch.bfloat16 else 4
        pad = (-N) % V
        if pad and dy.shape[0] >= 64:
            dy_p = F.pad(dy, (0, pad))
            if ctx.needs_input_grad[0]:
                dx = ext.gemm_nt(dy_p, F.pad(w, (0, pad)))
            if ctx.needs_input_grad[1]:
                dw = ext.gemm_tn(x, dy_p, w.dtype == x.dtype)
                dw = dw[:, :N].contiguous().to(w.dtype)
            return dx, dw, db, None, dres
        if ctx.needs_input_grad[0]:
            dx = ext.gemm_nt(dy, w)       # [M,N] @ [K,N]^T -> [M,K]
        if ctx.needs_input_grad[1]:
            dw = ext.gemm_tn(x, dy, w.dtype == x.dtype)
            dw = dw.to(w.dtype)  # no-op when the reduce already cast
        return dx, dw, db, None, dres


class _LinearQKV(torch.autograd.Function):
    """The three attention projections as ONE GEMM: wqkv = cat(wq,wk,wv)
    rebuilt per call (a weight-sized copy, ~1% of the GEMM it replaces);
    3x fewer GEMM/colsum launches and the shared input x keeps a single
    consumer (no autograd fan-in adds). Weight/bias grads come back as
    contiguous slices of the single dW/db reductions."""

    @staticmethod
    def forward(ctx, x2, wq, wk, wv, bq, bk, bv):
        ext = _C.ext()
        wqkv = torch.cat([wq, wk, wv], dim=1)
        has_b = bq is not None
        bqkv = torch.cat([bq, bk, bv]) if has_b else None
        qkv = ext.gemm(x2, wqkv, bqkv, ACT_KINDS["linear"])
        ctx.save_for_backward(x2, wqkv)
        ctx.has_b = has_b
        return qkv

    @staticmethod
    def backward(ctx, dy):
        x2, wqkv = ctx.saved_tensors
        ext = _C.ext()
        dy = dy.contiguous()
        N = wqkv.shape[1] // 3
        dx = dwq = dwk = dwv = dbq = dbk = dbv = None
        if ctx.has_b and ctx.needs_input_grad[4]:
            db = ext.colsum(dy)
            dbq, dbk, dbv = db[:N], db[N:2 * N], db[2 * N:]
        if ctx.needs_input_grad[0]:
            dx = ext.gemm_nt(dy, wqkv)
        if ctx.needs_input_grad[1]:
            dw = ext.gemm_tn(x2, dy, wqkv.dtype == x2.dtype).to(wqkv.dtype)
            # contiguous slices: the fused multi-tensor optimizer reads
            # grads as flat buffers
            dwq = dw[:, :N].contiguous()
            dwk = dw[:, N:2 * N].contiguous()
            dwv = dw[:, 2 * N:].contiguous()
        return dx, dwq, dwk, dwv, dbq, dbk, dbv


def linear_qkv(x: torch.Tensor, wq, wk, wv, bq=None, bk=None, bv=None):
    """Merged attention projection: [.., K] -> [.., 3N] (see _LinearQKV)."""
    if not _use_hip(x):
        outs = [x @ w + (b if b is not None else 0)
                for w, b in ((wq, bq), (wk, bk), (wv, bv))]
        return torch.cat(outs, dim=-1)
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    y = _LinearQKV.apply(x2.contiguous(), wq, wk, wv, bq, bk, bv)
    return y.reshape(*lead, y.shape[-1])


def linear(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor] = None,
           act: str = "linear",
           residual: Optional[torch.Tensor] = None,
           ln: Optional[tuple] = None) -> torch.Tensor:
    """y = act(ln?(x) @ w + bias) [+ residual]; x ``[..., K]``, w ``[K, N]``
    row-major. ``residual`` (shape of y) fuses the add into the GEMM
    epilogue on GPU (act must be 'linear'). ``ln=(gamma, beta, eps)``
    fuses a LayerNorm of x into the decode GEMV's staging (M=1,
    inference only)."""
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.shape[-1])
    if ln is not None:
        gamma, beta, eps = ln
        if (_use_hip(x) and x2.shape[0] == 1
                and not torch.is_grad_enabled()):
            ext = _C.ext()
            y = ext.gemm(x2.contiguous(), w, bias, ACT_KINDS[act],
                         residual.reshape(1, -1).contiguous()
                         if residual is not None else None,
                         gamma, beta, eps)
            return y.reshape(*lead, w.shape[1])
        x = layer_norm(x, gamma, beta, eps)
        x2 = x.reshape(-1, x.shape[-1])
    if _use_hip(x):
        # relu fuses into the GEMM epilogue (mask-replay backward); other
        # activations need the pre-activation saved, so run them unfused —
        # except under no_grad (decode), where any activation fuses
        fused = act if (act in ("linear", "relu")
                        or not torch.is_grad_enabled()) else "linear"
        r2 = (residual.reshape(-1, residual.shape[-1]).contiguous()
              if residual is not None else None)
        y = _Linear.apply(x2.contiguous(), w.contiguous(),
                          None if bias is None else bias.contiguous(), fused,
                          r2 if fused == act else None)
        if fused != act:
            y = _Activation.apply(y, act)
            if residual is not None:
                y = y + r2
    else:
        y = x2 @ w
        if bias is not None:
            y = y + bias
        if act == "relu":
            y = F.relu(y)
        elif act == "gelu":
            y = F.gelu(y, approximate="tanh")
        elif act != "linear":
            y = activation(y, act)
        if residual is not None:
            y = y + residual.reshape(-1, residual.shape[-1])
    return y.reshape(*lead, w.shape[1])


class _Bmm(torch.autograd.Function):
    """Strided-batched MFMA GEMM (reference gemm_strided_batched_ex,
    src/math/cuda/gemm.cu:84-105). Transpose views pass through as
    strides — no materialized transposes anywhere in fwd or bwd."""

    @staticmethod
    def forward(ctx, a, b):
        ext = _C.ext()
        ctx.save_for_backward(a, b)
        return ext.bmm(a, b)

    @staticmethod
    def backward(ctx, dc):
        a, b = ctx.saved_tensors
        ext = _C.ext()
        da = db = None
        if ctx.needs_input_grad[0]:
            da = ext.bmm(dc, b.transpose(-1, -2))
        if ctx.needs_input_grad[1]:
            db = ext.bmm(a.transpose(-1, -2), dc)
        return da, db


def matmul(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Plain 2-D/batched matmul through the HIP MFMA GEMMs on GPU."""
    if _use_hip(a, b):
        if a.dim() == 2 and b.dim() == 2:
            return _Linear.apply(a.contiguous(), b.contiguous(), None, "linear")
        lead = a.shape[:-2]
        a3 = a.reshape(-1, a.shape[-2], a.shape[-1])
        b3 = b.reshape(-1, b.shape[-2], b.shape[-1])
        return _Bmm.apply(a3, b3).reshape(*lead, a.shape[-2], b.shape[-1])
    return a @ b


# ---------------------------------------------------------------------------
# pooling NHWC (reference maxpool_ops.cu / avgpool_ops.cu)
# ---------------------------------------------------------------------------


class _MaxPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ext = _C.ext()
        y, idx = ext.maxpool_fwd(x, kernel[0], kernel[1], stride[0], stride[1],
                                 padding[0], padding[1])
        ctx.save_for_backward(idx)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        ext = _C.ext()
        dx = ext.maxpool_bwd(dy.contiguous(), idx, ctx.in_shape[1], ctx.in_shape[2])
        return dx, None, None, None


class _AvgPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ext = _C.ext()
        y = ext.avgpool_fwd(x, kernel[0], kernel[1], stride[0], stride[1],
                            padding[0], padding[1])
        ctx.in_shape = x.shape
        ctx.kernel, ctx.stride, ctx.padding = kernel, stride, padding
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _C.ext()
        dx = ext.avgpool_bwd(dy.contiguous(), ctx.in_shape[1], ctx.in_shape[2],
                             ctx.kernel[0], ctx.kernel[1], ctx.stride[0], ctx.stride[1],
                             ctx.padding[0], ctx.padding[1])
        return dx, None, None, None


def max_pool2d_nhwc(x, kernel, stride=None, padding=(0, 0)):
    stride = stride or kernel
    if _use_hip(x):
        return _MaxPool2d.apply(x.contiguous(), kernel, stride, padding)
    y = F.max_pool2d(_nhwc_to_nchw(x), kernel, stride, padding)
    return _nchw_to_nhwc(y)


def avg_pool2d_nhwc(x, kernel, stride=None, padding=(0, 0)):
    stride = stride or kernel
    if _use_hip(x):
        return _AvgPool2d.apply(x.contiguous(), kernel, stride, padding)
    y = F.avg_pool2d(_nhwc_to_nchw(x), kernel, stride, padding)
    return _nchw_to_nhwc(y)


# ---------------------------------------------------------------------------
# dropout (Philox mask on GPU; reference dropout.cu:33)
# ---------------------------------------------------------------------------


class _Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, seed, ctr=None):
        ext = _C.ext()
        y, mask = ext.dropout_fwd(x, p, seed, ctr)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        ext = _C.ext()
        return (ext.dropout_bwd(dy.contiguous(), mask, ctx.p), None, None,
                None)


def dropout(x: torch.Tensor, p: float, training: bool) -> torch.Tensor:
    if not training or p <= 0.0:
        return x
    if _use_hip(x):
        return _Dropout.apply(x.contiguous(), p, _draw_seed(), _graph_ctr)
    return F.dropout(x, p, training=True)


# ---------------------------------------------------------------------------
# standalone activations (reference src/nn/activations_impl/)
# ---------------------------------------------------------------------------


class _Activation(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kind):
        ext = _C.ext()
        y = ext.act_fwd(x, ACT_KINDS[kind])
        ctx.save_for_backward(x, y)
        ctx.kind = kind
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y = ctx.saved_tensors
        ext = _C.ext()
        return ext.act_bwd(dy.contiguous(), x, y, ACT_KINDS[ctx.kind]), None


class _ScaledSoftmax(torch.autograd.Function):
    """Row softmax with fused scale + causal mask (replaces the reference's
    causal_mask.cu fill + cudnnSoftmaxForward/Backward pair and the
    softmax activation kernels, softmax_kernels.cu:11-346)."""

    @staticmethod
    def forward(ctx, x, scale, causal, mrows, qoff):
        ext = _C.ext()
        y = ext.smax_fwd(x, mrows, qoff, scale, causal)
        ctx.save_for_backward(y)
        ctx.scale = scale
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        ext = _C.ext()
        dx = ext.smax_bwd(y, dy.contiguous(), ctx.scale)
        return dx, None, None, None, None


def scaled_softmax(x: torch.Tensor, scale: float = 1.0, causal: bool = False,
                   qoff: int = 0) -> torch.Tensor:
    """softmax(x * scale) over the last dim; with ``causal``, row r of each
    trailing [M, C] matrix may attend to columns <= r + qoff (suffix-causal
    decode prefill passes qoff = kv_len - M). Fully masked rows are 0."""
    if _use_hip(x):
        mrows = x.shape[-2] if x.dim() >= 2 else 1
        return _ScaledSoftmax.apply(x.contiguous(), scale, causal, mrows, qoff)
    xf = x.float() * scale
    if causal:
        M, C = x.shape[-2], x.shape[-1]
        pos_q = torch.arange(M, device=x.device).unsqueeze(-1) + qoff
        pos_k = torch.arange(C, device=x.device)
        xf = xf.masked_fill(pos_k > pos_q, float("-inf"))
    y = F.softmax(xf, dim=-1)
    y = torch.nan_to_num(y, nan=0.0)  # fully-masked rows -> 0 (kernel parity)
    return y.to(x.dtype)


class _AddAct(torch.autograd.Function):
    """y = act(a + b) in one pass (ResidualBlock join; backward is one
    shared mask pass for both inputs). Only y-derivable activations
    (relu/sigmoid/tanh/linear)."""

    @staticmethod
    def forward(ctx, a, b, kind):
        ext = _C.ext()
        y = ext.add_act_fwd(a, b, ACT_KINDS[kind])
        ctx.save_for_backward(y)
        ctx.kind = kind
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        ext = _C.ext()
        g = ext.add_act_bwd(dy.contiguous(), y, ACT_KINDS[ctx.kind])
        return g, g, None


def add_act(a: torch.Tensor, b: torch.Tensor,
            act: str = "linear") -> torch.Tensor:
    """Fused act(a + b) (residual joins)."""
    if act == "none":
        act = "linear"
    if (_use_hip(a, b) and a.shape == b.shape and a.dtype == b.dtype
            and act in ("linear", "relu", "sigmoid", "tanh")):
        return _AddAct.apply(a.contiguous(), b.contiguous(), act)
    y = a + b
    return activation(y, act) if act != "linear" else y


def activation(x: torch.Tensor, kind: str) -> torch.Tensor:
    if kind == "linear":
        return x
    if kind == "softmax":
        if _use_hip(x):
            return scaled_softmax(x, 1.0)
        return F.softmax(x, dim=-1)
    if _use_hip(x):
        return _Activation.apply(x.contiguous(), kind)
    if kind == "relu":
        return F.relu(x)
    if kind == "gelu":
        return F.gelu(x, approximate="tanh")
    if kind == "sigmoid":
        return torch.sigmoid(x)
    if kind == "tanh":
        return torch.tanh(x)
    if kind == "elu":
        return F.elu(x)
    if kind == "leaky_relu":
        return F.leaky_relu(x, 0.01)
    if kind == "silu":
        return F.silu(x)
    raise ValueError(f"unknown activation {kind!r}")


# ---------------------------------------------------------------------------
# fused logsoftmax + cross-entropy (reference loss_ops.cu:76,150)
# ---------------------------------------------------------------------------


class _SoftmaxXent(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        ext = _C.ext()
        loss, lse = ext.ce_fwd(logits, targets)
        ctx.save_for_backward(logits, targets, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        ext = _C.ext()
        dlogits = ext.ce_bwd(logits, targets, lse, dloss.contiguous())
        return dlogits, None


def softmax_cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy from raw logits; ``targets`` are int64 class ids."""
    l2 = logits.reshape(-1, logits.shape[-1])
    t = targets.reshape(-1)
    if _use_hip(logits):
        per = _SoftmaxXent.apply(l2.contiguous(), t.contiguous())
        return per.mean()
    return F.cross_entropy(l2.float(), t)


# ---------------------------------------------------------------------------
# layer norm (reference layer_norm_ops.cu:21,51)
# ---------------------------------------------------------------------------


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ext = _C.ext()
        y, mean, invstd = ext.ln_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd = ctx.saved_tensors
        ext = _C.ext()
        dx, dgamma, dbeta = ext.ln_bwd(x, dy.contiguous(), gamma, mean, invstd)
        return dx, dgamma, dbeta, None


class _LayerNormRes(torch.autograd.Function):
    """LayerNorm with a residual PASSTHROUGH output: ``y, xr = f(x)`` where
    xr aliases x. A pre-LN transformer block routes its residual branch
    through xr, so x has ONE consumer and the junction's grad join
    (d_x = ln_grad + d_residual) happens inside the ln_bwd kernel instead
    of an autograd fan-in add pass (2 activation-sized adds per block per
    step without this)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ext = _C.ext()
        y, mean, invstd = ext.ln_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, invstd)
        ctx.set_materialize_grads(False)
        return y, x.view_as(x)

    @staticmethod
    def backward(ctx, dy, dres):
        x, gamma, mean, invstd = ctx.saved_tensors
        ext = _C.ext()
        if dy is None:  # residual-only consumer (degenerate use)
            return dres, None, None, None
        if dres is not None:
            dres = dres.contiguous()
        dx, dgamma, dbeta = ext.ln_bwd(x, dy.contiguous(), gamma, mean,
                                       invstd, resid=dres)
        return dx, dgamma, dbeta, None


def layer_norm_res(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                   eps: float = 1e-5):
    """(layer_norm(x), residual passthrough of x) — see _LayerNormRes."""
    if _use_hip(x):
        lead = x.shape[:-1]
        y, xr = _LayerNormRes.apply(
            x.reshape(-1, x.shape[-1]).contiguous(), gamma, beta, eps)
        return (y.reshape(*lead, x.shape[-1]),
                xr.reshape(*lead, x.shape[-1]))
    return layer_norm(x, gamma, beta, eps), x


def layer_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    if _use_hip(x):
        lead = x.shape[:-1]
        y = _LayerNorm.apply(x.reshape(-1, x.shape[-1]).contiguous(), gamma, beta, eps)
        return y.reshape(*lead, x.shape[-1])
    xf = x.float()
    y = F.layer_norm(xf, (x.shape[-1],), gamma.float(), beta.float(), eps)
    return y.to(x.dtype)


# ---------------------------------------------------------------------------
# group norm NHWC (reference groupnorm_ops.cu:46-171)
# ---------------------------------------------------------------------------


class _GroupNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, groups, eps):
        ext = _C.ext()
        y, mean, invstd = ext.gn_fwd(x, gamma, beta, groups, eps)
        ctx.save_for_backward(x, gamma, mean, invstd)
        ctx.groups = groups
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd = ctx.saved_tensors
        ext = _C.ext()
        dx, dgamma, dbeta = ext.gn_bwd(x, dy.contiguous(), mean, invstd,
                                       gamma, ctx.groups)
        return dx, dgamma, dbeta, None, None


def group_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               groups: int, eps: float = 1e-5) -> torch.Tensor:
    """GroupNorm over channels-last [N, ..., C]; fp32 stats, fp32 gamma/beta."""
    if _use_hip(x):
        return _GroupNorm.apply(x.contiguous(), gamma, beta, groups, eps)
    n, c = x.shape[0], x.shape[-1]
    xf = x.float().reshape(n, -1, groups, c // groups)
    mean = xf.mean(dim=(1, 3), keepdim=True)
    var = xf.var(dim=(1, 3), unbiased=False, keepdim=True)
    xhat = ((xf - mean) / torch.sqrt(var + eps)).reshape(*x.shape)
    return (xhat * gamma.float() + beta.float()).to(x.dtype)


# ---------------------------------------------------------------------------
# pointwise regression losses (reference loss_ops.cu:308-390)
# ---------------------------------------------------------------------------

PT_LOSS_KINDS = {"mse": 0, "mae": 1, "huber": 2}


class _PtLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pred, target, kind, delta):
        ext = _C.ext()
        loss = ext.ptloss_fwd(pred, target, PT_LOSS_KINDS[kind], delta)
        ctx.save_for_backward(pred, target)
        ctx.kind, ctx.delta = kind, delta
        return loss

    @staticmethod
    def backward(ctx, dloss):
        pred, target = ctx.saved_tensors
        ext = _C.ext()
        dpred = ext.ptloss_bwd(pred, target, dloss.contiguous().float(),
                               PT_LOSS_KINDS[ctx.kind], ctx.delta)
        return dpred, None, None, None


def pointwise_loss(pred: torch.Tensor, target: torch.Tensor, kind: str,
                   delta: float = 1.0) -> torch.Tensor:
    """Mean MSE / MAE / Huber loss with a fused reduce on GPU."""
    if _use_hip(pred):
        return _PtLoss.apply(pred.contiguous(),
                             target.to(pred.dtype).contiguous(), kind, delta)
    if kind == "mse":
        return F.mse_loss(pred.float(), target.float())
    if kind == "mae":
        return F.l1_loss(pred.float(), target.float())
    return F.huber_loss(pred.float(), target.float(), delta=delta)


# ---------------------------------------------------------------------------
# embedding (reference embedding_ops.cu:17,48)
# ---------------------------------------------------------------------------


class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, table):
        ext = _C.ext()
        y = ext.embedding_fwd(ids, table)
        ctx.save_for_backward(ids)
        ctx.rows = table.shape[0]
        ctx.dtype = table.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        ext = _C.ext()
        dtab = ext.embedding_bwd(ids, dy.contiguous(), ctx.rows)
        return None, dtab.to(ctx.dtype)


def embedding(ids: torch.Tensor, table: torch.Tensor) -> torch.Tensor:
    if _use_hip(table):
        return _Embedding.apply(ids.contiguous(), table.contiguous())
    return F.embedding(ids, table)


# ---------------------------------------------------------------------------
# scaled-dot-product attention (flash kernel on GPU, Phase 3;
# reference src/nn/blocks_impl/cuda/cudnn_flash_attention_ops.cu:81)
# ---------------------------------------------------------------------------


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = True) -> torch.Tensor:
    """q/k/v: [B, H, S, D] -> [B, H, S, D]."""
    if _use_hip(q):
        from .flash import flash_attention
        return flash_attention(q, k, v, causal=causal)
    scale = q.shape[-1] ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = F.softmax(s, dim=-1)
    return (p @ v.float()).to(q.dtype)


def sdpa_materialized(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      causal: bool = True, qoff: int = 0) -> torch.Tensor:
    """Reference-AttentionBlock math with the S x S scores materialized:
    batched MFMA GEMMs + one fused scale/causal-mask softmax pass on GPU
    (reference attention_block.cpp:144-147, cuda/softmax.cu:47-79,
    causal_mask.cu:13). q [*, M, D], k/v [*, T, D]; with ``qoff`` the
    suffix-causal offset (kv_len - M) for KV-cache prefill."""
    scale = q.shape[-1] ** -0.5
    if _use_hip(q):
        scores = matmul(q, k.transpose(-1, -2))
        p = scaled_softmax(scores, scale, causal=causal, qoff=qoff)
        return matmul(p, v)
    s = (q.float() @ k.float().transpose(-1, -2))
    p = scaled_softmax(s, scale, causal=causal, qoff=qoff)
    return (p @ v.float()).to(q.dtype)


def attention_decode(q: torch.Tensor, k_cache: torch.Tensor,
                     v_cache: torch.Tensor, pos: Optional[torch.Tensor],
                     length: int, scale: float) -> torch.Tensor:
    """Fused single-token decode attention (inference only): q [BH, D]
    against the full KV cache buffers [BH, cap, D]; live length from the
    device ``pos`` tensor (len = pos+1, hipGraph-capturable) or ``length``."""
    ext = _C.ext()
    return ext.attn_decode(q.contiguous(), k_cache, v_cache, pos,
                           length, scale)
