import torch

from transformer_amd.models import Transformer
from transformer_amd.runtime import CheckpointManager, NoamAdam
from transformer_amd.ops import reference as R


def _model(seed=0):
    torch.manual_seed(seed)
    return Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                       input_vocab_size=50, target_vocab_size=50, rate=0.0,
                       max_position=32)


def _step(model, opt, seed):
    torch.manual_seed(seed)
    inp = torch.randint(1, 50, (2, 5))
    tar = torch.randint(1, 50, (2, 5))
    logits, _ = model((inp, tar), training=True)
    loss = R.masked_cross_entropy(logits, tar, 2)
    opt.zero_grad()
    loss.backward()
    opt.step()
    return loss.item()


def test_save_restore_resume_parity(tmp_path):
    """save -> restart -> resume gives a bit-identical
    loss trajectory vs uninterrupted training."""
    mA, mB = _model(), _model()
    oA = NoamAdam(mA, 16, warmup_steps=10)
    oB = NoamAdam(mB, 16, warmup_steps=10)

    for s in range(3):
        _step(mA, oA, s)
        _step(mB, oB, s)

    cm = CheckpointManager(mA, oA, str(tmp_path / "ck"), max_to_keep=3)
    cm.save(step=3, epoch=0)

    # fresh model+optimizer restored from checkpoint
    mC = _model(seed=123)
    oC = NoamAdam(mC, 16, warmup_steps=10)
    cmC = CheckpointManager(mC, oC, str(tmp_path / "ck"), max_to_keep=3)
    meta = cmC.restore()
    assert meta["step"] == 3
    assert oC.step_count == 3

    for s in range(3, 6):
        la = _step(mA, oA, s)
        lc = _step(mC, oC, s)
        assert la == lc, (s, la, lc)


def test_rolling_window(tmp_path):
    m = _model()
    o = NoamAdam(m, 16)
    cm = CheckpointManager(m, o, str(tmp_path / "ck"), max_to_keep=2)
    for s in range(1, 5):
        cm.save(step=s)
    import os
    files = [f for f in os.listdir(tmp_path / "ck") if f.endswith(".pt")]
    assert sorted(files) == ["ckpt-3.pt", "ckpt-4.pt"]
    assert cm.latest_checkpoint.endswith("ckpt-4.pt")


def test_restore_none_when_empty(tmp_path):
    m = _model()
    cm = CheckpointManager(m, None, str(tmp_path / "empty"))
    assert cm.restore() is None


def test_export_and_reload_forward_parity(tmp_path):
    """C24: export (SavedModel-equivalent dir) -> load -> same logits."""
    import torch
    from transformer_amd.models import Transformer
    from transformer_amd.runtime import export_model
    from transformer_amd.runtime.export import load_exported

    torch.manual_seed(3)
    m = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                    input_vocab_size=40, target_vocab_size=44, rate=0.0,
                    max_position=32)
    cfg = dict(num_layers=1, d_model=16, num_heads=2, dff=32,
               input_vocab_size=40, target_vocab_size=44, dropout_rate=0.0,
               max_position=32)
    d = str(tmp_path / "model")
    export_model(m, d, cfg)
    m2, cfg2 = load_exported(d)
    assert cfg2["d_model"] == 16
    src = torch.randint(1, 39, (2, 7))
    tar = torch.randint(1, 43, (2, 5))
    with torch.no_grad():
        a, _ = m((src, tar), training=False)
        b, _ = m2((src, tar), training=False)
    assert torch.equal(a, b)


def test_serve_endpoints(tmp_path, toy_corpus):
    """serve.py: export a tiny trained-ish model + vocabs, then exercise
    /health and /translate through FastAPI's in-process test client."""
    import torch
    from fastapi.testclient import TestClient

    from transformer_amd.data.dataset import load_dataset
    from transformer_amd.models import Transformer
    from transformer_amd.runtime import export_model
    import serve

    _, _, src_tok, tgt_tok = load_dataset(
        toy_corpus, str(tmp_path / "sv"), str(tmp_path / "tv"),
        sequence_length=50, batch_size=4, seed=1)
    torch.manual_seed(0)
    m = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                    input_vocab_size=src_tok.vocab_size + 2,
                    target_vocab_size=tgt_tok.vocab_size + 2,
                    rate=0.0, max_position=64)
    cfg = dict(num_layers=1, d_model=16, num_heads=2, dff=32,
               input_vocab_size=src_tok.vocab_size + 2,
               target_vocab_size=tgt_tok.vocab_size + 2,
               dropout_rate=0.0, max_position=64)
    export_model(m, str(tmp_path / "model"), cfg)

    app = serve.build_app(str(tmp_path / "model"), str(tmp_path / "sv"),
                          str(tmp_path / "tv"))
    client = TestClient(app)
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    r = client.post("/translate", json={"text": "one two", "max_len": 4})
    assert r.status_code == 200
    body = r.json()
    assert isinstance(body["tokens"], list) and len(body["tokens"]) >= 1
    assert "text" in body


def test_restore_tolerates_shape_mismatch(tmp_path, capsys):
    """expect_partial semantics: a checkpoint from a differently-sized
    model must not crash restore — mismatched entries are skipped with a
    warning (reference train.py:163)."""
    small = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                        input_vocab_size=50, target_vocab_size=50, rate=0.0,
                        max_position=32)
    CheckpointManager(small, None, str(tmp_path), 2).save(1)
    big = Transformer(num_layers=1, d_model=32, num_heads=2, dff=64,
                      input_vocab_size=50, target_vocab_size=50, rate=0.0,
                      max_position=32)
    mgr = CheckpointManager(big, None, str(tmp_path), 2)
    before = {k: v.clone() for k, v in big.state_dict().items()}
    meta = mgr.restore()
    assert meta is not None and meta["step"] == 1
    out = capsys.readouterr().out
    assert "shape_mismatch" in out
    # mismatched params untouched
    for k, v in big.state_dict().items():
        if v.shape != small.state_dict().get(k, v).shape:
            assert torch.equal(v, before[k])
