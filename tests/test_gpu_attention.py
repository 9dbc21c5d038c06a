"""Flash attention kernels vs fp32 composed reference (full-tensor check,
asymmetric random operands — cdna guide §5.4 rules 16/25)."""


import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tnn_amd import _C
    ext = _C.ext()

DEV = "cuda"


def ref_attention(q, k, v, causal):
    scale = q.shape[-1] ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return p @ v.float()


def maxerr(a, b):
    return (a.float() - b.float()).abs().max().item()


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("S", [64, 128, 192, 1000])
@pytest.mark.parametrize("D", [64, 128])
def test_attn_fwd(S, D, causal):
    torch.manual_seed(0)
    B, H = 2, 3
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    o, lse = ext.attn_fwd(q, k, v, causal)
    ref = ref_attention(q, k, v, causal)
    assert maxerr(o, ref) < 3e-2, (S, D, causal, maxerr(o, ref))
    # lse check
    scale = D ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=DEV).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    ref_lse = torch.logsumexp(s, dim=-1)
    assert maxerr(lse, ref_lse) < 2e-2


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("S,D", [(128, 64), (256, 64), (192, 128)])
def test_attn_bwd(S, D, causal):
    torch.manual_seed(1)
    B, H = 2, 2
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    do = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)

    o, lse = ext.attn_fwd(q, k, v, causal)
    dq, dk, dv = ext.attn_bwd(q, k, v, o, do, lse, causal)

    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    ref_attention(qf, kf, vf, causal).backward(do.float())
    tol = 6e-2
    assert maxerr(dq, qf.grad) < tol, ("dq", maxerr(dq, qf.grad))
    assert maxerr(dk, kf.grad) < tol, ("dk", maxerr(dk, kf.grad))
    assert maxerr(dv, vf.grad) < tol, ("dv", maxerr(dv, vf.grad))


def test_attn_spiked_rescale_path():
    """Force large per-tile max jumps (guide §5.4 rule 26): one K row
    spiked so the running max moves far past earlier tiles."""
    torch.manual_seed(2)
    S, D = 512, 64
    q = torch.randn(1, 1, S, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(1, 1, S, D, dtype=torch.bfloat16, device=DEV)
    k[0, 0, 400] *= 30  # spike in a late tile
    v = torch.randn(1, 1, S, D, dtype=torch.bfloat16, device=DEV)
    o, _ = ext.attn_fwd(q, k, v, False)
    ref = ref_attention(q, k, v, False)
    assert maxerr(o, ref) < 5e-2


def test_flash_autograd_function():
    from tnn_amd.ops.flash import flash_attention
    torch.manual_seed(3)
    q = torch.randn(1, 2, 128, 64, dtype=torch.bfloat16, device=DEV,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    o = flash_attention(q, k, v, causal=True)
    o.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad.float()).all()


def test_gpt_block_gpu():
    from tnn_amd.nn.blocks import GPTBlock
    from tnn_amd.nn.layer import cast_compute_dtype
    torch.manual_seed(4)
    blk = GPTBlock(256, 4, flash=True, dtype=torch.float32)
    blk_ref = GPTBlock(256, 4, flash=False, dtype=torch.float32)
    blk_ref.load_state_dict(blk.state_dict())
    x = torch.randn(2, 128, 256)
    ref = blk_ref(x)  # CPU fp32 naive attention
    cast_compute_dtype(blk, torch.bfloat16)
    blk.to(DEV)
    y = blk(x.to(DEV).bfloat16())
    err = (y.float().cpu() - ref).abs().max().item()
    assert err < 0.15, err


@pytest.mark.gpu
@pytest.mark.parametrize("causal", [False, True])
def test_attn_strided_views(causal):
    """BSHD transposed views (the projection layout) and merged-QKV head
    slices must match the contiguous path bit-for-bit concern-free: the
    kernels read rows through stride tuples instead of forcing copies."""
    torch.manual_seed(4)
    B, H, S, D = 2, 4, 192, 64
    # three slices of one merged [B, S, 3*H*D] buffer
    qkv = torch.randn(B, S, 3 * H * D, dtype=torch.bfloat16, device=DEV)
    q, k, v = (t.unflatten(-1, (H, D)).permute(0, 2, 1, 3)
               for t in qkv.split(H * D, dim=-1))
    assert not q.is_contiguous()
    o_v, lse_v = ext.attn_fwd(q, k, v, causal)
    o_c, lse_c = ext.attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(),
                              causal)
    assert maxerr(o_v, o_c) == 0.0
    assert maxerr(lse_v, lse_c) == 0.0

    do = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
    # strided grads written into one merged dQKV buffer
    dqkv = torch.empty_like(qkv)
    dq_o, dk_o, dv_o = (t.unflatten(-1, (H, D)).permute(0, 2, 1, 3)
                        for t in dqkv.split(H * D, dim=-1))
    dq, dk, dv = ext.attn_bwd(q, k, v, o_v, do.transpose(1, 2).contiguous()
                              .transpose(1, 2), lse_v, causal,
                              dq_out=dq_o, dk_out=dk_o, dv_out=dv_o)
    rq, rk, rv = ext.attn_bwd(q.contiguous(), k.contiguous(), v.contiguous(),
                              o_c.contiguous(), do, lse_c, causal)
    # dq is fp32-atomic accumulated: ordering differs run to run, so the
    # bf16 cast can move by an ulp; dk/dv are single-writer and exact
    assert maxerr(dq, rq) < 2e-3
    assert maxerr(dk, rk) == 0.0
    assert maxerr(dv, rv) == 0.0


@pytest.mark.gpu
def test_ln_passthrough_residual_grads():
    """layer_norm_res: the residual junction grad join happens inside
    ln_bwd; grads must match the unfused two-consumer formulation."""
    from tnn_amd import ops
    torch.manual_seed(7)
    x = torch.randn(4, 96, 768, dtype=torch.bfloat16, device=DEV)
    g = torch.randn(768, device=DEV)
    b = torch.randn(768, device=DEV)
    w = (torch.randn(768, 768, dtype=torch.bfloat16, device=DEV) * 0.02)

    xa = x.clone().requires_grad_()
    y, xr = ops.layer_norm_res(xa, g, b, 1e-5)
    out = ops.linear(y, w, None, residual=xr)
    out.float().square().mean().backward()

    xb = x.clone().requires_grad_()
    y2 = ops.layer_norm(xb, g, b, 1e-5)
    out2 = ops.linear(y2, w, None, residual=xb)
    out2.float().square().mean().backward()

    assert maxerr(out, out2) == 0.0
    assert maxerr(xa.grad, xb.grad) < 2e-3


@pytest.mark.gpu
def test_gpt_block_train_matches_cpu_path():
    """GPU GPTBlock (passthrough-residual structure) vs the CPU
    composition on the same weights."""
    from tnn_amd.nn.blocks import GPTBlock
    torch.manual_seed(8)
    blk = GPTBlock(128, 4, flash=True, dropout=0.0)
    x = torch.randn(2, 64, 128)
    ref = blk(x)
    blk_g = blk.to(DEV)
    y = blk_g(x.to(DEV))
    assert maxerr(y.cpu(), ref) < 5e-2
    y.square().mean().backward()
    for n, p in blk_g.named_parameters():
        assert p.grad is None or torch.isfinite(p.grad.float()).all(), n


@pytest.mark.gpu
def test_merged_qkv_training_path_matches_separate():
    """linear_qkv + flash_attention_qkv (one GEMM, one dQKV buffer) vs
    three separate projections + flash_attention: same fwd and grads."""
    from tnn_amd import ops
    torch.manual_seed(9)
    B, S, H, D = 2, 128, 4, 64
    dim = H * D
    x = torch.randn(B, S, dim, dtype=torch.bfloat16, device=DEV)
    ws = [torch.randn(dim, dim, dtype=torch.bfloat16, device=DEV) * 0.02
          for _ in range(3)]
    bs = [torch.randn(dim, dtype=torch.bfloat16, device=DEV) * 0.01
          for _ in range(3)]

    def run(merged):
        wq, wk, wv = [w.clone().requires_grad_() for w in ws]
        bq, bk, bv = [b.clone().requires_grad_() for b in bs]
        xa = x.clone().requires_grad_()
        if merged:
            qkv = ops.linear_qkv(xa, wq, wk, wv, bq, bk, bv)
            o = ops.flash_attention_qkv(qkv, H, True)
        else:
            q = ops.linear(xa, wq, bq).view(B, S, H, D).transpose(1, 2)
            k = ops.linear(xa, wk, bk).view(B, S, H, D).transpose(1, 2)
            v = ops.linear(xa, wv, bv).view(B, S, H, D).transpose(1, 2)
            o = ops.attention(q, k, v, causal=True)
            o = o.transpose(1, 2).reshape(B, S, dim)
        o.float().square().mean().backward()
        return o, xa.grad, wq.grad, bq.grad

    o1, dx1, dwq1, dbq1 = run(True)
    o2, dx2, dwq2, dbq2 = run(False)
    assert maxerr(o1, o2) < 3e-3
    assert maxerr(dx1, dx2) < 3e-3
    assert maxerr(dwq1, dwq2) < 3e-3
    assert maxerr(dbq1.float(), dbq2.float()) < 3e-3
