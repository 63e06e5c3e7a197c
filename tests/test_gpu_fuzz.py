"""Seeded shape-fuzz across the kernel library (GPU): random-but-
deterministic shapes through every kernel family against the fp32
references — catches edge/guard bugs the fixed-shape tests miss."""

import math
import random

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from transformer_amd.ops import ext
    return ext()


def assert_close(got, ref, tol, name):
    got = got.float().cpu()
    ref = ref.float().cpu()
    scale = ref.abs().max().clamp(min=1.0)
    err = (got - ref).abs().max() / scale
    assert err < tol, f"{name}: rel-max err {err:.4f} (tol {tol})"


@pytest.mark.parametrize("case", range(12))
def test_fuzz_gemm_nt(case):
    rng = random.Random(100 + case)
    m = rng.randrange(1, 900)
    n = rng.randrange(1, 900)
    k = rng.randrange(8, 700)
    torch.manual_seed(case)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    epi = case % 2
    c = _ext().gemm_nt(a, w, b, epi)
    ref = a.float() @ w.float().T + b.float()
    if epi:
        ref = torch.relu(ref)
    assert_close(c, ref, 0.04, f"gemm_nt {m}x{n}x{k}")


@pytest.mark.parametrize("case", range(8))
def test_fuzz_gemm256(case):
    rng = random.Random(200 + case)
    m = rng.randrange(1, 1200)
    n = rng.randrange(1, 1200)
    k = 64 * rng.randrange(2, 16)
    torch.manual_seed(case)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm256_nt(a, w, torch.Tensor(), 0)
    assert_close(c, a.float() @ w.float().T, 0.04, f"gemm256 {m}x{n}x{k}")


@pytest.mark.parametrize("case", range(8))
def test_fuzz_gemm_dw(case):
    rng = random.Random(300 + case)
    mt = rng.randrange(1, 5000)
    n = rng.randrange(1, 700)
    k = rng.randrange(1, 700)
    torch.manual_seed(case)
    dy = torch.randn(mt, n, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(mt, k, device="cuda", dtype=torch.bfloat16)
    c = _ext().gemm_dw(dy, x)
    assert_close(c, dy.float().t() @ x.float(), 0.04, f"dw {mt}x{n}x{k}")


@pytest.mark.parametrize("case", range(10))
def test_fuzz_attention_fwd_bwd(case):
    rng = random.Random(400 + case)
    B = rng.randrange(1, 5)
    H = rng.choice([1, 2, 3, 5])
    Sq = rng.randrange(1, 300)
    dh = rng.choice([32, 64, 128])
    causal = rng.random() < 0.5
    Sk = Sq if causal else rng.randrange(1, 300)
    torch.manual_seed(case)
    q = torch.randn(B, Sq, H, dh, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
    pad = (torch.rand(B, Sk, device="cuda") < 0.2).to(torch.uint8)
    pad[:, 0] = 0  # never a fully-padded kv row 0
    sc = 1.0 / math.sqrt(dh)
    o, lse = _ext().attn_fwd(q, k, v, pad, causal, sc)

    # fp32 reference
    qt = q.float().permute(0, 2, 1, 3)
    kt = k.float().permute(0, 2, 1, 3)
    vt = v.float().permute(0, 2, 1, 3)
    s = qt @ kt.transpose(-1, -2) * sc
    s = s + pad.float().view(B, 1, 1, Sk) * -1e9
    if causal:
        mask = torch.triu(torch.ones(Sq, Sk, device="cuda"), 1)
        s = s + mask * -1e9
    p = torch.softmax(s, -1)
    ref = (p @ vt).permute(0, 2, 1, 3)
    assert_close(o, ref, 0.05, f"attn fwd case{case}")

    do = torch.randn_like(o)
    dq, dk, dv = _ext().attn_bwd(q, k, v, o, do, lse, pad, causal, sc, 0)
    qr = q.float().requires_grad_()
    kr = k.float().requires_grad_()
    vr = v.float().requires_grad_()
    s2 = (qr.permute(0, 2, 1, 3) @ kr.permute(0, 2, 1, 3).transpose(-1, -2)
          * sc) + pad.float().view(B, 1, 1, Sk) * -1e9
    if causal:
        s2 = s2 + mask * -1e9
    out2 = (torch.softmax(s2, -1) @ vr.permute(0, 2, 1, 3)).permute(0, 2, 1, 3)
    (out2 * do.float()).sum().backward()
    assert_close(dq, qr.grad, 0.06, f"attn dq case{case}")
    assert_close(dk, kr.grad, 0.06, f"attn dk case{case}")
    assert_close(dv, vr.grad, 0.06, f"attn dv case{case}")


@pytest.mark.parametrize("case", range(8))
def test_fuzz_residual_ln(case):
    rng = random.Random(500 + case)
    R = rng.randrange(1, 800)
    D = rng.choice([8, 16, 64, 96, 128, 512, 1024, 520])
    torch.manual_seed(case)
    x = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
    res = torch.randn(R, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(D, device="cuda", dtype=torch.bfloat16)
    y, s, mean, rstd = _ext().ln_fwd(x, res, g, b, 1e-6)
    sd = x.float() + res.float()
    mu = sd.mean(-1, keepdim=True)
    rs = 1.0 / (sd.var(-1, unbiased=False, keepdim=True) + 1e-6).sqrt()
    ref = (sd - mu) * rs * g.float() + b.float()
    assert_close(y, ref, 0.06, f"ln fwd R{R} D{D}")
