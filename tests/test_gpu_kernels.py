"""Kernel numerics tests: every HIP kernel against the
plain PyTorch fp32 reference in ops/reference.py, on the same bf16-rounded
inputs.  All @pytest.mark.gpu — run on an MI355X via gpurun."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from transformer_amd.ops import ext
    E = None


def _ext():
    global E
    if E is None:
        from transformer_amd.ops import ext as _e
        E = _e()
    return E


def assert_close(got, ref, tol=0.03, name=""):
    got = got.float().cpu()
    ref = ref.float().cpu()
    assert got.shape == ref.shape, (name, got.shape, ref.shape)
    scale = ref.abs().max().clamp(min=1.0)
    err = (got - ref).abs().max() / scale
    assert err < tol, f"{name}: rel-max err {err:.4f} (tol {tol})"


# ---------------------------------------------------------------------------
# GEMM family
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 512, 512),
                                   (100, 130, 72), (64, 32770 // 10, 512),
                                   (33, 17, 24)])
def test_gemm_nt(m, n, k):
    torch.manual_seed(0)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_nt(a, w, b, 0)
    ref = a.float() @ w.float().T + b.float()
    assert_close(c, ref, 0.03, "gemm_nt")


@pytest.mark.parametrize("m,n,k,epi", [
    (512, 512, 128, 0),       # minimum-depth pipeline (2 K-tiles)
    (777, 300, 512, 0),       # ragged M/N edge clamping
    (2048, 1536, 512, 1),     # QKV-like + ReLU epilogue
    (300, 1000, 1024, 0),     # partial tail tiles both dims
])
def test_gemm256_nt(m, n, k, epi):
    """Deep-pipelined 256x256 kernel (gemm256.hip) against fp32 reference,
    invoked directly so small shapes exercise it regardless of the
    dispatch heuristic in gemm_nt."""
    torch.manual_seed(2)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm256_nt(a, w, b, epi)
    ref = a.float() @ w.float().T + b.float()
    if epi == 1:
        ref = torch.relu(ref)
    assert_close(c, ref, 0.03, "gemm256_nt")


@pytest.mark.parametrize("m,n,k", [
    (16384, 512, 512),    # split-M accumulation path (128-tile)
    (4096, 1536, 512),    # 256-tile path
    (1000, 2049, 300),    # 256-tile path, ragged n/k/m edges
    (200, 130, 70),       # ragged every dim, single-slice path
    (64, 512, 512),
])
def test_gemm_dw(m, n, k):
    """tr16 transpose-read dW kernel (gemm_dw.hip) vs fp32 reference."""
    torch.manual_seed(4)
    dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_dw(dy, x)
    ref = dy.float().t() @ x.float()
    assert_close(c, ref, 0.03, "gemm_dw")


@pytest.mark.parametrize("m,n,k", [(16384, 512, 512), (200, 130, 70)])
def test_gemm_dw_fused_bias_grad(m, n, k):
    torch.manual_seed(6)
    dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    db = torch.empty(n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_dw(dy, x, None, db)
    assert_close(c, dy.float().t() @ x.float(), 0.03, "gemm_dw+db dW")
    assert_close(db, dy.float().sum(0), 0.03, "gemm_dw+db db")


def test_gemm_dw_out_destination():
    torch.manual_seed(5)
    dy = torch.randn(1024, 256, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(1024, 384, device="cuda", dtype=torch.bfloat16)
    out = torch.empty(256, 384, device="cuda", dtype=torch.bfloat16)
    r = _ext().gemm_dw(dy, x, out)
    assert r.data_ptr() == out.data_ptr()
    assert_close(out, dy.float().t() @ x.float(), 0.03, "gemm_dw_out")


def test_gemm256_nt_out_destination():
    torch.manual_seed(3)
    a = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16)
    out = torch.empty(512, 512, device="cuda", dtype=torch.bfloat16)
    r = _ext().gemm256_nt(a, w, torch.Tensor(), 0, out)
    assert r.data_ptr() == out.data_ptr()
    assert_close(out, a.float() @ w.float().T, 0.03, "gemm256_out")


def test_gemm_nt_relu():
    torch.manual_seed(1)
    a = torch.randn(200, 512, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(300, 512, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(300, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_nt(a, w, b, 1)
    ref = torch.relu(a.float() @ w.float().T + b.float())
    assert_close(c, ref, 0.03, "gemm_relu")
    assert (c.float() >= 0).all()


def test_gemm_no_bias():
    a = torch.randn(64, 128, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(96, 128, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_nt(a, w, torch.Tensor(), 0)
    assert_close(c, a.float() @ w.float().T, 0.03, "gemm_nobias")


@pytest.mark.parametrize("m,n,k", [(256, 512, 512), (256, 512, 1536),
                                   (100, 130, 72), (64, 512, 32770),
                                   (33, 17, 24)])
def test_gemm_nn(m, n, k):
    # dX = dY @ W: contraction over W's row dim, no physical transpose
    torch.manual_seed(2)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_nn(a, b)
    assert_close(c, a.float() @ b.float(), 0.03, "gemm_nn")


@pytest.mark.parametrize("m,n,k", [(512, 512, 4096), (1536, 512, 1000),
                                   (130, 70, 100), (2048, 512, 16384),
                                   (17, 33, 50)])
def test_gemm_tn(m, n, k):
    # dW = dY^T @ X: both operands contraction(row)-major
    torch.manual_seed(3)
    a = torch.randn(k, m, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_tn(a, b)
    assert_close(c, a.float().T @ b.float(), 0.03, "gemm_tn")


def test_gemm_tn_out_destination():
    # dW written straight into a provided (flat-grad view) buffer
    torch.manual_seed(4)
    a = torch.randn(777, 96, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(777, 64, device="cuda", dtype=torch.bfloat16)
    out = torch.empty(96, 64, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_tn(a, b, out)
    assert c.data_ptr() == out.data_ptr()
    assert_close(out, a.float().T @ b.float(), 0.03, "gemm_tn_out")


def test_transpose2d():
    a = torch.randn(130, 70, device="cuda", dtype=torch.bfloat16)
    t = _ext().transpose2d(a)
    assert torch.equal(t.cpu(), a.cpu().T.contiguous())


def test_colsum():
    a = torch.randn(500, 300, device="cuda", dtype=torch.bfloat16)
    s = _ext().colsum(a)
    assert_close(s, a.float().sum(0), 0.03, "colsum")


def test_relu_bwd():
    dy = torch.randn(1000, device="cuda", dtype=torch.bfloat16)
    y = torch.randn(1000, device="cuda", dtype=torch.bfloat16)
    dz = _ext().relu_bwd(dy, y)
    ref = dy.float() * (y.float() > 0)
    assert_close(dz, ref, 1e-3, "relu_bwd")


# ---------------------------------------------------------------------------
# LayerNorm
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("r,d", [(256, 512), (100, 1024), (64, 128)])
def test_ln_fwd_bwd(r, d):
    from transformer_amd.ops import reference as R
    torch.manual_seed(0)
    x = torch.randn(r, d, device="cuda", dtype=torch.bfloat16)
    res = torch.randn(r, d, device="cuda", dtype=torch.bfloat16)
    gamma = (torch.randn(d, device="cuda") * 0.1 + 1).bfloat16()
    beta = (torch.randn(d, device="cuda") * 0.1).bfloat16()
    y, s, mean, rstd = _ext().ln_fwd(x, res, gamma, beta, 1e-6)
    xf = x.float().requires_grad_(False)
    ref_in = (x.float() + res.float()).requires_grad_(True)
    gf = gamma.float().requires_grad_(True)
    bf = beta.float().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(ref_in, (d,), gf, bf, 1e-6)
    assert_close(y, ref, 0.03, "ln_fwd")
    assert_close(s, x.float() + res.float(), 0.02, "ln_s")

    dy = torch.randn(r, d, device="cuda", dtype=torch.bfloat16)
    ref.backward(dy.float())
    dx, dgamma, dbeta = _ext().ln_bwd(dy, s, gamma, mean, rstd)
    assert_close(dx, ref_in.grad, 0.05, "ln_dx")
    assert_close(dgamma, gf.grad, 0.05, "ln_dgamma")
    assert_close(dbeta, bf.grad, 0.05, "ln_dbeta")


# ---------------------------------------------------------------------------
# Embedding + PE
# ---------------------------------------------------------------------------

def test_embed_pe_fwd_bwd():
    from transformer_amd.ops import reference as R
    torch.manual_seed(0)
    V, d, B, S = 1000, 512, 4, 37
    w = torch.randn(V, d, device="cuda", dtype=torch.bfloat16)
    pe = R.positional_encoding(64, d).squeeze(0).to("cuda", torch.bfloat16)
    toks = torch.randint(0, V, (B, S), device="cuda")
    y = _ext().embed_pe_fwd(toks, w, pe)
    ref = R.embedding_scale_pe(toks, w.float(), pe.float()[None])
    assert_close(y, ref, 0.03, "embed_fwd")

    dy = torch.randn(B, S, d, device="cuda", dtype=torch.bfloat16)
    dw = _ext().embed_pe_bwd(dy, toks, V)
    ref_w = w.float().requires_grad_(True)
    out = R.embedding_scale_pe(toks, ref_w, pe.float()[None])
    out.backward(dy.float())
    assert_close(dw, ref_w.grad, 0.05, "embed_bwd")


# ---------------------------------------------------------------------------
# Dropout
# ---------------------------------------------------------------------------

def test_dropout_stats_and_bwd():
    x = torch.ones(100000, device="cuda", dtype=torch.bfloat16)
    y, mask = _ext().dropout_fwd(x, 0.1, 1234)
    keep = mask.float().mean().item()
    assert abs(keep - 0.9) < 0.01
    # kept elements scaled by 1/(1-p)
    yv = y.float()
    assert ((yv - mask.float() * (1 / 0.9)).abs() < 0.01).all()
    dy = torch.randn(100000, device="cuda", dtype=torch.bfloat16)
    dx = _ext().dropout_bwd(dy, mask, 0.1)
    ref = dy.float() * mask.float() / 0.9
    assert_close(dx, ref, 0.01, "dropout_bwd")
    # deterministic for same seed
    y2, mask2 = _ext().dropout_fwd(x, 0.1, 1234)
    assert torch.equal(mask, mask2)


# ---------------------------------------------------------------------------
# Cross entropy
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("ls", [0.0, 0.1])
def test_ce_fwd_bwd(ls):
    from transformer_amd.ops import reference as R
    torch.manual_seed(0)
    Rn, V = 64, 32770
    logits = (torch.randn(Rn, V, device="cuda") * 2).bfloat16()
    targets = torch.randint(0, V, (Rn,), device="cuda")
    targets[::7] = 0  # some padding
    loss, lse = _ext().ce_fwd(logits, targets, 8.0, ls)
    lref = R.masked_cross_entropy(logits.float().view(1, Rn, V),
                                  targets.view(1, Rn), 8, ls)
    assert abs(loss.item() - lref.item()) / max(abs(lref.item()), 1) < 0.01

    lg = logits.float().requires_grad_(True)
    lr2 = R.masked_cross_entropy(lg.view(1, Rn, V), targets.view(1, Rn), 8, ls)
    lr2.backward()
    one = torch.ones(1, device="cuda", dtype=torch.float32)
    # ce_bwd pads rows to roundup(V, 256) with zero columns (gemm_uni
    # contraction alignment); the [:, :V] slice is the gradient
    dl = _ext().ce_bwd(logits, targets, lse, one, 8.0, ls)[:, :V]
    assert_close(dl, lg.grad.view(Rn, V), 0.02, "ce_bwd")


# ---------------------------------------------------------------------------
# Adam
# ---------------------------------------------------------------------------

def test_adam_fused():
    from transformer_amd.ops import reference as R
    torch.manual_seed(0)
    n = 10000
    master = torch.randn(n, device="cuda")
    m = torch.randn(n, device="cuda").abs() * 0.01
    v = torch.randn(n, device="cuda").abs() * 0.01
    grad = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    param = master.bfloat16()
    mm, vv, mast = m.clone(), v.clone(), master.clone()
    _ext().adam_fused(master, m, v, grad, param, 1e-3, 0.9, 0.98, 1e-9, 3)
    R.adam_step_reference(mast, grad.float(), mm, vv, 3, 1e-3, 0.9, 0.98, 1e-9)
    assert_close(master, mast, 1e-4, "adam_master")
    assert_close(m, mm, 1e-4, "adam_m")
    assert_close(v, vv, 1e-4, "adam_v")
    assert_close(param, mast.bfloat16(), 0.01, "adam_param")


# ---------------------------------------------------------------------------
# argmax / accuracy
# ---------------------------------------------------------------------------

def test_argmax_accuracy():
    torch.manual_seed(0)
    logits = torch.randn(257, 1003, device="cuda", dtype=torch.bfloat16)
    am = _ext().argmax_lastdim(logits)
    ref = logits.float().argmax(-1)
    assert torch.equal(am.cpu(), ref.cpu())
    targets = torch.randint(0, 1003, (257,), device="cuda")
    targets[:40] = 0
    correct, total = _ext().accuracy(logits, targets)
    mask = targets != 0
    assert total == int(mask.sum())
    assert correct == int(((ref == targets) & mask).sum())


# ---------------------------------------------------------------------------
# Attention (3 mask variants), fwd + bwd vs fp32 reference
# ---------------------------------------------------------------------------

def _attn_ref(q, k, v, kv_pad, causal):
    """fp32 reference in (B,S,H,dh); mirrors ops.fused_attention eager path."""
    import transformer_amd.ops.reference as R
    qt, kt, vt = (t.float().permute(0, 2, 1, 3) for t in (q, k, v))
    mask = None
    if kv_pad is not None:
        mask = kv_pad.float()[:, None, None, :]
    if causal:
        la = R.create_look_ahead_mask(q.shape[1], device=q.device)
        mask = la[None, None] if mask is None else torch.maximum(mask, la[None, None])
    out = R.scaled_dot_product_attention(qt, kt, vt, mask)
    return out.permute(0, 2, 1, 3)


@pytest.mark.parametrize("dh", [32, 64, 128])
@pytest.mark.parametrize("causal", [False, True])
def test_attn_fwd(dh, causal):
    torch.manual_seed(0)
    B, S, H = 2, 100, 3
    q = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    pad = torch.zeros(B, S, device="cuda", dtype=torch.uint8)
    pad[0, 80:] = 1
    o, lse = _ext().attn_fwd(q, k, v, pad, causal, 1.0 / math.sqrt(dh))
    ref = _attn_ref(q, k, v, pad, causal)
    assert_close(o, ref, 0.04, f"attn_fwd dh{dh} causal{causal}")


@pytest.mark.parametrize("causal", [False, True])
def test_attn_fwd_long_seq_rf2(causal):
    """Sk >= 2048 dispatches the RF=2 (32 q-rows/wave) forward variant;
    odd length exercises its edge guards."""
    torch.manual_seed(11)
    B, S, H, dh = 1, 2051, 2, 64
    q = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    pad = torch.zeros(B, S, device="cuda", dtype=torch.uint8)
    pad[0, 2000:] = 1
    o, lse = _ext().attn_fwd(q, k, v, pad, causal, 1.0 / math.sqrt(dh))
    ref = _attn_ref(q, k, v, pad, causal)
    assert_close(o, ref, 0.05, f"attn_fwd_rf2 causal{causal}")


def test_attn_fwd_cross_shapes():
    torch.manual_seed(1)
    B, Sq, Sk, H, dh = 2, 37, 75, 4, 64
    q = torch.randn(B, Sq, H, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
    pad = torch.zeros(B, Sk, device="cuda", dtype=torch.uint8)
    pad[1, 60:] = 1
    o, _ = _ext().attn_fwd(q, k, v, pad, False, 1.0 / math.sqrt(dh))
    ref = _attn_ref(q, k, v, pad, False)
    assert_close(o, ref, 0.04, "attn_cross")


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("dh", [64, 128])
def test_attn_bwd(causal, dh):
    torch.manual_seed(0)
    B, S, H = 2, 64, 2
    mk = lambda: torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    q, k, v = mk(), mk(), mk()
    pad = torch.zeros(B, S, device="cuda", dtype=torch.uint8)
    pad[0, 50:] = 1
    scale = 1.0 / math.sqrt(dh)
    o, lse = _ext().attn_fwd(q, k, v, pad, causal, scale)
    do = mk()
    dq, dk, dv = _ext().attn_bwd(q, k, v, o, do, lse, pad, causal, scale)

    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    ref = _attn_ref(qf, kf, vf, pad, causal)
    ref.backward(do.float())
    assert_close(dq, qf.grad, 0.06, "attn_dq")
    assert_close(dk, kf.grad, 0.06, "attn_dk")
    assert_close(dv, vf.grad, 0.06, "attn_dv")


def test_attn_fwd_no_pad_mask():
    torch.manual_seed(2)
    B, S, H, dh = 1, 256, 2, 64
    q = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    o, _ = _ext().attn_fwd(q, k, v, torch.Tensor(), True, 1.0 / math.sqrt(dh))
    ref = _attn_ref(q, k, v, None, True)
    assert_close(o, ref, 0.04, "attn_nopad")


# ---------------------------------------------------------------------------
# Packed-QKV attention (strided kernel I/O, packed dQKV backward)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("causal", [False, True])
def test_attn_packed_self_matches_unpacked(causal):
    from transformer_amd import ops
    torch.manual_seed(11)
    B, S, H, dh = 3, 80, 4, 64
    qkv = torch.randn(B, S, 3, H, dh, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    pad = torch.zeros(B, S, dtype=torch.bool, device="cuda")
    pad[:, -7:] = True
    out = ops.self_attention(qkv, kv_pad=pad, causal=causal)
    g = torch.randn_like(out)
    out.backward(g)
    dqkv_packed = qkv.grad.clone()

    qkv2 = qkv.detach().clone().requires_grad_(True)
    q, k, v = (t.contiguous() for t in qkv2.unbind(dim=2))
    out2 = ops.fused_attention(q.detach().clone().requires_grad_(True),
                               k.detach().clone().requires_grad_(True),
                               v.detach().clone().requires_grad_(True),
                               kv_pad=pad, causal=causal)
    assert_close(out, out2, 0.02, "packed fwd vs unpacked")

    # backward parity vs fp32 reference through the eager path
    qkv3 = qkv.detach().float().clone().requires_grad_(True)
    out3 = ops.self_attention(qkv3.cpu(), kv_pad=pad.cpu(), causal=causal)
    out3.backward(g.float().cpu())
    assert_close(dqkv_packed, qkv3.grad, 0.06, "packed dqkv vs fp32 ref")


def test_attn_packed_cross_matches_reference():
    from transformer_amd import ops
    torch.manual_seed(12)
    B, Sq, Sk, H, dh = 2, 40, 96, 4, 64
    q = torch.randn(B, Sq, H, dh, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    kv = torch.randn(B, Sk, 2, H, dh, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    pad = torch.zeros(B, Sk, dtype=torch.bool, device="cuda")
    pad[:, -5:] = True
    out = ops.cross_attention(q, kv, kv_pad=pad)
    g = torch.randn_like(out)
    out.backward(g)

    q3 = q.detach().float().cpu().clone().requires_grad_(True)
    kv3 = kv.detach().float().cpu().clone().requires_grad_(True)
    out3 = ops.cross_attention(q3, kv3, kv_pad=pad.cpu())
    out3.backward(g.float().cpu())
    assert_close(out, out3, 0.02, "cross fwd")
    assert_close(q.grad, q3.grad, 0.06, "cross dq")
    assert_close(kv.grad, kv3.grad, 0.06, "cross dkv")


def test_dropout_residual_layernorm_fused():
    """Fused LN(dropout(x)+res): extract the mask from the fwd outputs and
    check fwd/bwd against a composed fp32 reference using that mask."""
    torch.manual_seed(9)
    R, D, p = 64, 128, 0.3
    E = _ext()
    x = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
    res = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
    gamma = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    beta = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    y, s, mean, rstd, mask = E.ln_fwd(x, res, gamma, beta, 1e-6, p, 1234)
    mk = mask.view(R, D).float()
    keep = mk.mean().item()
    assert 0.55 < keep < 0.85, keep  # ~1-p kept
    sd = x.float() * mk / (1 - p) + res.float()
    ref_mu = sd.mean(-1, keepdim=True)
    ref_rs = 1.0 / (sd.var(-1, unbiased=False, keepdim=True) + 1e-6).sqrt()
    ref_y = (sd - ref_mu) * ref_rs * gamma.float() + beta.float()
    assert_close(y, ref_y, 0.06, "fused drop-LN fwd")

    dy = torch.randn_like(y)
    dres, dgamma, dbeta, dxm = E.ln_bwd(dy, s, gamma, mean, rstd, None, None,
                                        mask, p)
    # reference backward via autograd on the composed fp32 graph
    xr = x.float().requires_grad_()
    rr = res.float().requires_grad_()
    gr = gamma.float().requires_grad_()
    br = beta.float().requires_grad_()
    sd2 = xr * mk / (1 - p) + rr
    mu = sd2.mean(-1, keepdim=True)
    rs2 = 1.0 / (sd2.var(-1, unbiased=False, keepdim=True) + 1e-6).sqrt()
    (((sd2 - mu) * rs2 * gr + br) * dy.float()).sum().backward()
    assert_close(dxm, xr.grad, 0.08, "fused drop-LN dx")
    assert_close(dres, rr.grad, 0.08, "fused drop-LN dres")
    assert_close(dgamma, gr.grad, 0.08, "fused drop-LN dgamma")
    assert_close(dbeta, br.grad, 0.08, "fused drop-LN dbeta")


def test_fwd_gemm_backends_agree():
    """The two forward-GEMM backends (_fwd_gemm dispatch: hand-written
    gemm_nt for small-M/forced, hipBLASLt fused epilogues for training
    shapes) must agree numerically at both epilogues."""
    from transformer_amd import ops
    E = ops.ext()
    torch.manual_seed(5)
    for M in (256, 4096):
        x = torch.randn(M, 512, device="cuda").bfloat16().contiguous()
        w = torch.randn(1536, 512, device="cuda").bfloat16().contiguous()
        b = torch.randn(1536, device="cuda").bfloat16().contiguous()
        y_hip = E.gemm_nt(x, w, b, 0).float()
        y_lib = torch.nn.functional.linear(x, w, b).float()
        err = (y_hip - y_lib).abs().max() / y_lib.abs().max().clamp(min=1)
        assert err < 1e-2, f"bias epilogue disagreement {err} at M={M}"
        y_hip = E.gemm_nt(x, w, b, 1).float()
        y_lib = torch._addmm_activation(b, x, w.t()).float()
        err = (y_hip - y_lib).abs().max() / y_lib.abs().max().clamp(min=1)
        assert err < 1e-2, f"relu epilogue disagreement {err} at M={M}"


# ---------------------------------------------------------------------------
# gemm_uni: deep-pipelined counted-vmcnt kernels (round 2)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("m,n,k,epi", [
    (16384, 512, 512, 0),     # out-proj fwd shape (BN=128 grid)
    (16384, 1536, 512, 0),    # packed QKV fwd
    (16384, 2048, 512, 1),    # FFN1 + ReLU epilogue
    (16384, 512, 2048, 0),    # FFN2
    (16320, 1000, 512, 0),    # ragged M + ragged N (NT clamps both)
    (2048, 32770, 512, 0),    # logits-head slice (big-N edge)
])
def test_gemm_uni_nt(m, n, k, epi):
    torch.manual_seed(3)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_uni_nt(a, w, b, epi)
    ref = a.float() @ w.float().T + b.float()
    if epi == 1:
        ref = torch.relu(ref)
    assert_close(c, ref, 0.03, f"gemm_uni_nt {m}x{n}x{k}")


@pytest.mark.parametrize("m,n,k", [
    (16384, 512, 1536),   # dX of packed QKV
    (16384, 2048, 512),   # dX of FFN2's weight... (dY @ W with W (512,2048))
    (16384, 512, 2048),   # dX of FFN1
    (16320, 512, 512),    # ragged-M dX of out-proj
])
def test_gemm_uni_nn(m, n, k):
    """dX: C[M,N] = A[M,K] @ B[K,N], B (the weight) read red-major via
    tr16 — asymmetric operands catch transposed fragment maps."""
    torch.manual_seed(4)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(k, n, device="cuda", dtype=torch.bfloat16) * 0.1
    c = _ext().gemm_uni_nn(a, w)
    ref = a.float() @ w.float()
    assert_close(c, ref, 0.03, f"gemm_uni_nn {m}x{n}x{k}")


@pytest.mark.parametrize("mt,n,k,splitr", [
    (16320, 32768, 512, 1),   # wide (logits-like) dW, aligned vocab
    (16320, 2048, 512, 1),    # mid dW
    (16384, 512, 512, 8),     # narrow deep dW, split contraction
    (16384, 1536, 512, 4),    # QKV dW, split contraction
])
def test_gemm_uni_tn(mt, n, k, splitr):
    """dW: C[N,K] = dY[Mt,N]^T @ X[Mt,K], both operands tr16-read."""
    torch.manual_seed(5)
    dy = torch.randn(mt, n, device="cuda", dtype=torch.bfloat16) * 0.1
    x = torch.randn(mt, k, device="cuda", dtype=torch.bfloat16) * 0.1
    c = _ext().gemm_uni_tn(dy, x, None, splitr)
    ref = dy.float().T @ x.float()
    assert_close(c, ref, 0.03, f"gemm_uni_tn {mt}x{n}x{k} s{splitr}")


def test_gemm_uni_tn_padded_vocab():
    """The ragged-vocab dW path: dY is a (M, V) view of a 256-padded
    buffer with zero pad columns (what ce_bwd emits)."""
    torch.manual_seed(6)
    mt, v, k = 4096, 32770, 512
    vp = (v + 255) // 256 * 256
    full = torch.zeros(mt, vp, device="cuda", dtype=torch.bfloat16)
    full[:, :v].normal_().mul_(0.1)
    dy = full[:, :v]
    x = torch.randn(mt, k, device="cuda", dtype=torch.bfloat16) * 0.1
    c = _ext().gemm_uni_tn(dy, x, None, 1)
    ref = dy.float().T @ x.float()
    assert_close(c, ref, 0.03, "gemm_uni_tn padded vocab")


def test_ce_bwd_pad_columns_zero():
    """ce_bwd returns (R, Vp) with Vp = roundup(V, 256); pad columns must
    be exactly zero (downstream GEMMs contract over Vp), and the [:, :V]
    grad must match the fp32 reference."""
    torch.manual_seed(7)
    R, V = 64, 1000
    vp = (V + 255) // 256 * 256
    logits = torch.randn(R, V, device="cuda", dtype=torch.bfloat16)
    tgt = torch.randint(1, V, (R,), device="cuda")
    tgt[::7] = 0  # some padded positions
    loss, lse = _ext().ce_fwd(logits, tgt, 4.0, 0.1)
    dfull = _ext().ce_bwd(logits, tgt, lse,
                          torch.ones(1, device="cuda"), 4.0, 0.1)
    assert dfull.shape == (R, vp)
    assert (dfull[:, V:].float() == 0).all(), "pad columns not zero"
    # fp32 reference grad
    lf = logits.detach().float().requires_grad_(True)
    from transformer_amd.ops import reference as Rf
    ref_loss = Rf.masked_cross_entropy(lf.view(1, R, V), tgt.view(1, R),
                                       4, 0.1)
    ref_loss.backward()
    assert_close(dfull[:, :V], lf.grad, 0.05, "ce pad grad")


def test_dx_gemm_padded_logits_path():
    """_dx_gemm on a padded-strided dY must match the dense reference
    (exercises _wt_padded + the NT padded contraction)."""
    from transformer_amd.ops import functional as F
    torch.manual_seed(8)
    M, V, d = 4096, 32770, 512
    vp = (V + 255) // 256 * 256
    full = torch.zeros(M, vp, device="cuda", dtype=torch.bfloat16)
    full[:, :V].normal_().mul_(0.05)
    dy = full[:, :V]
    w = torch.randn(V, d, device="cuda", dtype=torch.bfloat16) * 0.05
    F.bump_weight_version()
    dx = F._dx_gemm(_ext(), dy, w)
    ref = dy.float() @ w.float()
    assert_close(dx, ref, 0.05, "padded dx")
    # cache refresh: after a weight update + version bump, result follows
    with torch.no_grad():
        w.mul_(2.0)
    F.bump_weight_version()
    dx2 = F._dx_gemm(_ext(), dy, w)
    assert_close(dx2, ref * 2.0, 0.05, "padded dx after update")


def test_colsum_fused_finalize_repeated():
    """colsum's single-launch last-arriver finalize: repeated L1-warm
    calls on the same cached workspace must stay exact (stale-L1 reads by
    the finalizing block are the failure mode — guide G16 pitfall 3)."""
    torch.manual_seed(9)
    for N in (512, 1536, 2048):
        for it in range(20):
            M = 1024 + 128 * (it % 5)
            a = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
            got = _ext().colsum(a)
            ref = a.float().sum(dim=0)
            assert_close(got, ref, 0.02, f"colsum N={N} it={it}")


def test_ln_gb_fused_finalize_repeated():
    """ln_bwd's dgamma/dbeta single-launch finalize under repeated calls."""
    torch.manual_seed(10)
    D = 512
    gamma = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    beta = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    for it in range(15):
        R = 2048 + 256 * (it % 3)
        x = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
        res = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
        y, s, mean, rstd = _ext().ln_fwd(x, res, gamma, beta, 1e-6, 0.0, 0,
                                         None)
        dy = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
        dx, dgamma, dbeta = _ext().ln_bwd(dy, s, gamma, mean, rstd, None,
                                          None, None, 0.0)
        sf = s.float()
        xh = (sf - sf.mean(-1, keepdim=True)) \
            * torch.rsqrt(sf.var(-1, unbiased=False, keepdim=True) + 1e-6)
        ref_g = (dy.float() * xh).sum(0)
        ref_b = dy.float().sum(0)
        assert_close(dgamma, ref_g, 0.03, f"dgamma it={it}")
        assert_close(dbeta, ref_b, 0.03, f"dbeta it={it}")


def test_transpose_batch_descriptor_table():
    """Batched weight transpose: several weights in one launch, padded
    destinations, values match per-weight transpose."""
    torch.manual_seed(11)
    shapes = [(512, 512), (1536, 512), (1000, 128), (2048, 512)]
    rows = []
    pairs = []
    for (m, n) in shapes:
        w = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
        npad = (m + 255) // 256 * 256
        buf = torch.zeros(n, npad, device="cuda", dtype=torch.bfloat16)
        pairs.append((w, buf))
        for bm in range(0, m, 64):
            for bn in range(0, n, 64):
                rows.append([w.data_ptr(), buf.data_ptr(), m, n,
                             buf.stride(0), bm, bn])
    desc = torch.tensor(rows, dtype=torch.int64).cuda()
    _ext().transpose_batch(desc)
    torch.cuda.synchronize()
    for w, buf in pairs:
        m = w.shape[0]
        assert torch.equal(buf[:, :m], w.t().contiguous()), w.shape
        assert (buf[:, m:].float() == 0).all()


def test_attn_defer_max_spike_forces_rescale():
    """T13/rule-26: the defer-max branch is data-dependent — spike one K
    row in a LATE tile so the running max jumps past THR=8 after several
    deferred tiles; output must still match the fp32 reference."""
    from transformer_amd.ops import reference as R
    torch.manual_seed(12)
    B, S, H, dh = 2, 512, 4, 64
    q = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16) * 0.5
    k = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16) * 0.5
    v = torch.randn(B, S, H, dh, device="cuda", dtype=torch.bfloat16)
    # spike key row 350 (tile 5 of 8) against everything: raw q.k ~ dh*4
    k[:, 350] = 4.0
    q[:, :, :, :8] = 2.0  # give q mass so the spike dominates post-scale
    o, lse = _ext().attn_fwd(q, k, v, torch.Tensor(), False,
                             1.0 / math.sqrt(dh), 1)
    qt, kt, vt = (t.permute(0, 2, 1, 3).float() for t in (q, k, v))
    ref, _ = R.scaled_dot_product_attention(qt, kt, vt, None,
                                            return_weights=True)
    ref = ref.permute(0, 2, 1, 3)
    assert_close(o, ref, 0.04, "defer-max spike")
    # and the no-spike path still matches (deferred tiles only)
    k2 = torch.randn_like(k) * 0.3
    o2, _ = _ext().attn_fwd(q, k2, v, torch.Tensor(), False,
                            1.0 / math.sqrt(dh), 1)
    kt2 = k2.permute(0, 2, 1, 3).float()
    ref2, _ = R.scaled_dot_product_attention(qt, kt2, vt, None,
                                             return_weights=True)
    assert_close(o2, ref2.permute(0, 2, 1, 3), 0.04, "defer-max smooth")


def test_relu_bwd_db_fused():
    """dz = dy*(y>0) with fused bias grad vs separate reference, under
    repeated L1-warm calls on the cached workspace."""
    torch.manual_seed(13)
    for it in range(10):
        M, N = 2048 + 256 * (it % 3), 2048
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        y = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        dz, db = _ext().relu_bwd_db(dy, y)
        ref_dz = dy.float() * (y.float() > 0)
        assert_close(dz, ref_dz, 1e-3, f"relu_bwd_db dz it={it}")
        assert_close(db, ref_dz.sum(0), 0.02, f"relu_bwd_db db it={it}")


def test_dx_gemm_wt_path_nonpadded():
    """The default dX backend (cached transposed weight through the NT
    kernel) vs the fp32 reference at a 64-aligned training shape, across
    a weight update (cache refresh)."""
    from transformer_amd.ops import functional as F
    torch.manual_seed(14)
    M, N, K = 4096, 1536, 512
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.05
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.05
    F.bump_weight_version()
    dx = F._dx_gemm(_ext(), dy, w)
    assert_close(dx, dy.float() @ w.float(), 0.03, "wt dx")
    with torch.no_grad():
        w.mul_(0.5)
    F.bump_weight_version()
    dx2 = F._dx_gemm(_ext(), dy, w)
    assert_close(dx2, dy.float() @ w.float(), 0.03, "wt dx refreshed")


def test_ce_fused_matches_reference_and_seed_scaling():
    """ce_fused (loss + grad in one kernel): loss and grad vs the fp32
    reference; ce_scale no-ops at seed 1.0 and scales exactly otherwise."""
    from transformer_amd.ops import reference as Rf
    torch.manual_seed(16)
    R, V = 96, 32770
    vp = (V + 255) // 256 * 256
    logits = (torch.randn(R, V, device="cuda") * 2).bfloat16()
    tgt = torch.randint(1, V, (R,), device="cuda")
    tgt[::5] = 0
    loss, dfull = _ext().ce_fused(logits, tgt, 8.0, 0.1)
    assert dfull.shape == (R, vp)
    assert (dfull[:, V:].float() == 0).all()
    lf = logits.detach().float().requires_grad_(True)
    ref_loss = Rf.masked_cross_entropy(lf.view(1, R, V), tgt.view(1, R),
                                       8, 0.1)
    ref_loss.backward()
    assert abs(loss.item() - ref_loss.item()) / max(abs(ref_loss.item()),
                                                    1) < 0.01
    assert_close(dfull[:, :V], lf.grad, 0.05, "ce_fused grad")
    # seed 1.0: no-op
    before = dfull.clone()
    _ext().ce_scale(dfull, torch.ones(1, device="cuda"))
    assert torch.equal(dfull, before)
    # seed 2.5: exact scale
    _ext().ce_scale(dfull, torch.full((1,), 2.5, device="cuda"))
    assert_close(dfull[:, :V], lf.grad * 2.5, 0.06, "ce_scale 2.5")


def test_ce_autograd_path_nonunit_seed():
    """masked_cross_entropy through autograd with a NON-unit upstream
    gradient (loss * 3).backward() must scale the logits grad by 3."""
    from transformer_amd import ops
    torch.manual_seed(17)
    B, T, V = 2, 24, 1000
    logits = torch.randn(B, T, V, device="cuda",
                         dtype=torch.bfloat16).requires_grad_(True)
    tgt = torch.randint(1, V, (B, T), device="cuda")
    loss = ops.masked_cross_entropy(logits, tgt, B, 0.1)
    (loss * 3.0).backward()
    g3 = logits.grad.clone()
    logits.grad = None
    loss2 = ops.masked_cross_entropy(logits, tgt, B, 0.1)
    loss2.backward()
    assert_close(g3, logits.grad.float() * 3.0, 0.03, "seed-3 scaling")
