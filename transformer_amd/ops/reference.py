"""Pure-PyTorch reference implementations of every device op.

These are the semantic contracts of the HIP kernels:
each hand-written CDNA4 kernel in csrc/ is numerics-tested against the
function of the same name here, and they double as the CPU execution path
(config 1: toy-corpus training on CPU).  fp32 math throughout, matching the
reference TF2 model (reference Attention.py:21, positionalencoding.py:23).
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# K4 / C4 — sinusoidal positional encoding, CONCAT layout:
# sines of all even-index angles first, then cosines of all odd-index angles,
# exactly as reference positionalencoding.py:8-23 (not the paper's interleave).
# ---------------------------------------------------------------------------

def positional_encoding(position: int, d_model: int, dtype=torch.float32,
                        device=None) -> torch.Tensor:
    pos = torch.arange(position, dtype=torch.float64, device=device).unsqueeze(1)
    i = torch.arange(d_model, dtype=torch.float64, device=device).unsqueeze(0)
    angle_rates = 1.0 / torch.pow(10000.0, (2 * (i // 2)) / float(d_model))
    angle_rads = pos * angle_rates  # (position, d_model)
    sines = torch.sin(angle_rads[:, 0::2])
    cosines = torch.cos(angle_rads[:, 1::2])
    pe = torch.cat([sines, cosines], dim=-1)  # (position, d_model)
    return pe.unsqueeze(0).to(dtype)  # (1, position, d_model)


# ---------------------------------------------------------------------------
# K14 / C5 — mask construction (reference positionalencoding.py:25-52).
# Mask convention: 1.0 = masked (added as mask * -1e9 to logits).
# ---------------------------------------------------------------------------

def create_padding_mask(seq: torch.Tensor) -> torch.Tensor:
    """(B, S) int tokens -> (B, 1, 1, S) float mask, 1.0 where pad (==0)."""
    return (seq == 0).to(torch.float32)[:, None, None, :]


def create_look_ahead_mask(size: int, device=None) -> torch.Tensor:
    """(T, T) float, 1.0 strictly above the diagonal (future positions)."""
    return 1.0 - torch.tril(torch.ones(size, size, device=device))


def create_masks(inp: torch.Tensor, tar: torch.Tensor):
    """Returns (enc_padding_mask, combined_mask, dec_padding_mask) exactly as
    reference positionalencoding.py:37-52."""
    enc_padding_mask = create_padding_mask(inp)
    dec_padding_mask = create_padding_mask(inp)
    look_ahead = create_look_ahead_mask(tar.shape[1], device=tar.device)
    dec_target_padding_mask = create_padding_mask(tar)
    combined_mask = torch.maximum(dec_target_padding_mask, look_ahead)
    return enc_padding_mask, combined_mask, dec_padding_mask


# ---------------------------------------------------------------------------
# K2-K5 / C1 — scaled dot-product attention (reference Attention.py:3-34).
# q,k,v: (B, H, Sq|Sk, dh); mask broadcastable to (B, H, Sq, Sk), 1.0=masked.
# ---------------------------------------------------------------------------

def scaled_dot_product_attention(q, k, v, mask=None, return_weights=False):
    dk = q.shape[-1]
    logits = torch.matmul(q, k.transpose(-2, -1)) / math.sqrt(dk)
    if mask is not None:
        logits = logits + mask * -1e9
    weights = torch.softmax(logits, dim=-1)
    out = torch.matmul(weights, v)
    if return_weights:
        return out, weights
    return out


# ---------------------------------------------------------------------------
# K9 — fused residual-add + LayerNorm (post-LN residual, eps=1e-6;
# reference Encoder.py:13-14,23,27).
# ---------------------------------------------------------------------------

def residual_layernorm(x, residual, gamma, beta, eps: float = 1e-6):
    return F.layer_norm(x + residual, (x.shape[-1],), gamma, beta, eps)


# ---------------------------------------------------------------------------
# K10 — embedding lookup * sqrt(d) + PE slice (reference Encoder.py:51-53).
# ---------------------------------------------------------------------------

def embedding_scale_pe(tokens, weight, pe):
    """tokens (B,S) int64; weight (V,d); pe (1,P,d) with P >= S."""
    d = weight.shape[1]
    x = F.embedding(tokens, weight) * math.sqrt(d)
    return x + pe[:, : tokens.shape[1], :].to(x.dtype)


# ---------------------------------------------------------------------------
# K13 / C13 — padding-masked cross entropy, sum/batch_size scaling
# (reference train.py:83-88) with optional label smoothing (SURVEY.md §8 Q9:
# eps=0 reproduces reference numerics).
# ---------------------------------------------------------------------------

def masked_cross_entropy(logits, targets, batch_size: int,
                         label_smoothing: float = 0.0):
    """logits (B,T,V) float; targets (B,T) int64; returns scalar
    sum(per-token CE * pad_mask) / batch_size  (batch_size = GLOBAL batch,
    SURVEY.md §8 Q4)."""
    V = logits.shape[-1]
    if logits.dtype in (torch.bfloat16, torch.float16):
        logits = logits.float()
    logp = F.log_softmax(logits, dim=-1)
    nll = -logp.gather(-1, targets.unsqueeze(-1)).squeeze(-1)  # (B,T)
    if label_smoothing > 0.0:
        smooth = -logp.mean(dim=-1)  # uniform-over-V component
        nll = (1.0 - label_smoothing) * nll + label_smoothing * smooth
    mask = (targets != 0).to(nll.dtype)
    return (nll * mask).sum() / float(batch_size)


# ---------------------------------------------------------------------------
# C14 — masked token accuracy (we mask pad positions, the
# reference does not — intended behaviour implemented, divergence noted).
# ---------------------------------------------------------------------------

def masked_accuracy(logits, targets):
    pred = logits.argmax(dim=-1)
    mask = targets != 0
    correct = ((pred == targets) & mask).sum()
    total = mask.sum().clamp(min=1)
    return correct.float() / total.float()


# ---------------------------------------------------------------------------
# K11 — dropout (reference rate 0.1).  Reference path uses torch's RNG.
# ---------------------------------------------------------------------------

def dropout(x, p: float, training: bool):
    return F.dropout(x, p=p, training=training)


# ---------------------------------------------------------------------------
# K15 / C11+C12 — Noam LR schedule + Adam(β1=.9, β2=.98, eps=1e-9)
# (reference train.py:21-34, 65-66).
# ---------------------------------------------------------------------------

def noam_lr(step: int, d_model: int, warmup_steps: int = 60000) -> float:
    step = max(step, 1)
    return (d_model ** -0.5) * min(step ** -0.5, step * warmup_steps ** -1.5)


def adam_step_reference(param, grad, m, v, step: int, lr: float,
                        beta1=0.9, beta2=0.98, eps=1e-9):
    """Single-tensor fp32 Adam update (the contract of the fused HIP kernel
    K15).  Mutates m, v, param in place; `param` is the fp32 master weight."""
    m.mul_(beta1).add_(grad, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    mhat = m / (1 - beta1 ** step)
    vhat = v / (1 - beta2 ** step)
    param.sub_(lr * mhat / (vhat.sqrt() + eps))
    return param
