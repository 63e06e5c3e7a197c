import math

import torch
import pytest

from transformer_amd.ops import reference as R


def test_positional_encoding_concat_layout():
    # concat layout — first d/2 dims are sines of the
    # even-index angle rates, last d/2 are cosines of the odd-index rates.
    d, P = 8, 16
    pe = R.positional_encoding(P, d).squeeze(0)
    pos = torch.arange(P, dtype=torch.float64).unsqueeze(1)
    i = torch.arange(d, dtype=torch.float64).unsqueeze(0)
    ang = pos / torch.pow(torch.tensor(10000.0, dtype=torch.float64),
                          (2 * (i // 2)) / d)
    expect = torch.cat([torch.sin(ang[:, 0::2]), torch.cos(ang[:, 1::2])], -1)
    assert torch.allclose(pe, expect.float(), atol=1e-6)
    assert pe[0, d // 2:].allclose(torch.ones(d // 2))  # cos(0) = 1


def test_masks():
    inp = torch.tensor([[5, 6, 0, 0], [1, 2, 3, 4]])
    tar = torch.tensor([[7, 8, 0], [7, 8, 9]])
    enc, comb, dec = R.create_masks(inp, tar)
    assert enc.shape == (2, 1, 1, 4)
    assert enc[0, 0, 0].tolist() == [0, 0, 1, 1]
    assert dec.equal(enc)
    assert comb.shape == (2, 1, 3, 3)
    # causal upper triangle masked
    assert comb[1, 0].tolist() == [[0, 1, 1], [0, 0, 1], [0, 0, 0]]
    # pad position of tar also masked as a key
    assert comb[0, 0, 2].tolist() == [0, 0, 1]


def test_sdpa_matches_manual():
    torch.manual_seed(0)
    q = torch.randn(2, 2, 4, 8)
    k = torch.randn(2, 2, 6, 8)
    v = torch.randn(2, 2, 6, 8)
    mask = torch.zeros(2, 1, 1, 6)
    mask[0, 0, 0, -2:] = 1.0
    out, w = R.scaled_dot_product_attention(q, k, v, mask, return_weights=True)
    assert out.shape == (2, 2, 4, 8)
    assert torch.allclose(w.sum(-1), torch.ones(2, 2, 4), atol=1e-5)
    assert w[0, :, :, -2:].abs().max() < 1e-6  # masked keys got ~0 weight


def test_noam_schedule_values():
    # C11: lr = d^-0.5 * min(step^-0.5, step*warmup^-1.5), warmup 60000 (Q3)
    from transformer_amd.runtime import NoamSchedule
    s = NoamSchedule(512, warmup_steps=60000)
    assert s(1) == pytest.approx(512 ** -0.5 * 1 * 60000 ** -1.5)
    assert s(60000) == pytest.approx(512 ** -0.5 * 60000 ** -0.5)
    assert s(240000) == pytest.approx(512 ** -0.5 * 240000 ** -0.5)
    # peak at warmup boundary
    assert s(60000) >= s(59999) and s(60000) >= s(60001)


def test_masked_ce_scaling():
    # Q4: sum over non-pad tokens / GLOBAL batch size
    torch.manual_seed(0)
    logits = torch.randn(2, 3, 11)
    targets = torch.tensor([[1, 2, 0], [3, 4, 5]])
    loss = R.masked_cross_entropy(logits, targets, batch_size=4)
    lp = torch.log_softmax(logits, -1)
    manual = 0.0
    for b in range(2):
        for t in range(3):
            if targets[b, t] != 0:
                manual += -lp[b, t, targets[b, t]].item()
    assert loss.item() == pytest.approx(manual / 4.0, rel=1e-5)


def test_masked_ce_label_smoothing_zero_matches_plain():
    torch.manual_seed(0)
    logits = torch.randn(2, 3, 11)
    targets = torch.randint(1, 11, (2, 3))
    a = R.masked_cross_entropy(logits, targets, 2, label_smoothing=0.0)
    b = R.masked_cross_entropy(logits, targets, 2, label_smoothing=0.1)
    assert not torch.isclose(a, b)  # smoothing changes the value
    # smoothed loss = (1-eps)*nll + eps*uniform-CE
    lp = torch.log_softmax(logits, -1)
    nll = -lp.gather(-1, targets[..., None]).squeeze(-1)
    smooth = -lp.mean(-1)
    expect = (0.9 * nll + 0.1 * smooth).sum() / 2
    assert b.item() == pytest.approx(expect.item(), rel=1e-5)


def test_masked_accuracy_ignores_pad():
    logits = torch.zeros(1, 3, 5)
    logits[0, 0, 2] = 9.0  # predict 2 (correct)
    logits[0, 1, 1] = 9.0  # predict 1 (wrong, target 3)
    logits[0, 2, 4] = 9.0  # target pad -> excluded
    targets = torch.tensor([[2, 3, 0]])
    acc = R.masked_accuracy(logits, targets)
    assert acc.item() == pytest.approx(0.5)


def test_adam_reference_matches_torch():
    torch.manual_seed(0)
    p0 = torch.randn(64)
    g = torch.randn(64)
    # torch Adam single step
    pt = p0.clone().requires_grad_(True)
    opt = torch.optim.Adam([pt], lr=1e-3, betas=(0.9, 0.98), eps=1e-9)
    pt.grad = g.clone()
    opt.step()
    # ours
    p = p0.clone()
    m = torch.zeros(64)
    v = torch.zeros(64)
    R.adam_step_reference(p, g.clone(), m, v, step=1, lr=1e-3)
    assert torch.allclose(p, pt.detach(), atol=1e-6)
