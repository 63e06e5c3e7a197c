"""Cross-check the fp32 reference oracle (ops/reference.py) against
torch's own built-ins.  Every HIP kernel's numerics test compares against
this oracle, so a silent oracle bug would mis-validate the whole kernel
fleet — these tests pin the oracle to an independent implementation."""

import math

import pytest
import torch
import torch.nn.functional as F

from transformer_amd.ops import reference as R


def test_attention_oracle_vs_torch_sdpa():
    torch.manual_seed(0)
    B, H, Sq, Sk, dh = 2, 3, 7, 9, 16
    q = torch.randn(B, H, Sq, dh)
    k = torch.randn(B, H, Sk, dh)
    v = torch.randn(B, H, Sk, dh)
    pad = torch.zeros(B, Sk, dtype=torch.bool)
    pad[0, 6:] = True
    mask = pad.float()[:, None, None, :]           # 1 = masked (reference)
    out, _ = R.scaled_dot_product_attention(q, k, v, mask,
                                            return_weights=True)
    ref = F.scaled_dot_product_attention(
        q, k, v, attn_mask=~pad[:, None, None, :].expand(B, 1, Sq, Sk))
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_attention_oracle_causal_vs_torch():
    torch.manual_seed(1)
    B, H, S, dh = 2, 2, 8, 8
    q, k, v = (torch.randn(B, H, S, dh) for _ in range(3))
    la = R.create_look_ahead_mask(S)
    out, _ = R.scaled_dot_product_attention(q, k, v, la[None, None],
                                            return_weights=True)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_layernorm_oracle_vs_torch():
    torch.manual_seed(2)
    x = torch.randn(5, 32)
    res = torch.randn(5, 32)
    g = torch.randn(32)
    b = torch.randn(32)
    out = R.residual_layernorm(x, res, g, b, eps=1e-6)
    ref = F.layer_norm(x + res, (32,), g, b, eps=1e-6)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


@pytest.mark.parametrize("eps_ls", [0.0, 0.1])
def test_cross_entropy_oracle_vs_torch(eps_ls):
    torch.manual_seed(3)
    B, T, V = 3, 6, 40
    logits = torch.randn(B, T, V)
    targets = torch.randint(1, V, (B, T))
    targets[0, 4:] = 0  # padding
    ours = R.masked_cross_entropy(logits, targets, batch_size=B,
                                  label_smoothing=eps_ls)
    per = F.cross_entropy(logits.reshape(-1, V), targets.reshape(-1),
                          reduction="none", label_smoothing=eps_ls)
    mask = (targets.reshape(-1) != 0).float()
    ref = (per * mask).sum() / B
    assert torch.allclose(ours, ref, atol=1e-5), (ours, ref)


def test_positional_encoding_formula():
    pe = R.positional_encoding(16, 8).squeeze(0)  # concat layout (Q2)
    d = 8
    for pos in (0, 3, 15):
        for i in range(d // 2):
            angle = pos / (10000 ** (2 * i / d))
            assert abs(pe[pos, i].item() - math.sin(angle)) < 1e-5
            assert abs(pe[pos, d // 2 + i].item() - math.cos(angle)) < 1e-5


def test_noam_schedule_formula():
    for step in (1, 100, 60000, 200000):
        lr = R.noam_lr(step, 512, 60000)
        ref = (512 ** -0.5) * min(step ** -0.5, step * 60000 ** -1.5)
        assert abs(lr - ref) < 1e-12
