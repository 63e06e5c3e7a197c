import os

import torch
import pytest

from transformer_amd.data import load_dataset, BatchedDataset
from transformer_amd.models import Transformer
from transformer_amd.runtime import Train


def _mk_train(tmp_path, toy_corpus, epochs=1, **model_kw):
    train_ds, test_ds, src_tok, tgt_tok = load_dataset(
        toy_corpus, str(tmp_path / "sv"), str(tmp_path / "tv"),
        sequence_length=50, batch_size=8, seed=3)
    kw = dict(num_layers=1, d_model=32, num_heads=2, dff=64, rate=0.0,
              max_position=64)
    kw.update(model_kw)
    torch.manual_seed(0)
    model = Transformer(input_vocab_size=src_tok.vocab_size + 2,
                        target_vocab_size=tgt_tok.vocab_size + 2, **kw)
    tr = Train(epochs=epochs, enable_function=False, transformer=model,
               src_tokenizer=src_tok, tgt_tokenizer=tgt_tok, batch_size=8,
               train_log_dir=str(tmp_path / "logs/train"),
               test_log_dir=str(tmp_path / "logs/test"),
               max_ckpt_keep=2, ckpt_path=str(tmp_path / "ckpt"),
               d_model=kw["d_model"], warmup_steps=100, log_interval=10 ** 9,
               eval_steps=2)
    return tr, train_ds, test_ds


def test_overfit_small_corpus(tmp_path, toy_corpus):
    """Convergence: loss decreases markedly when overfitting a small slice
    (SURVEY.md §4 item 2)."""
    tr, train_ds, test_ds = _mk_train(tmp_path, toy_corpus)
    small = BatchedDataset(train_ds.pairs[:16], 8, shuffle=False)
    first_loss, last_loss = None, None
    for it in range(30):
        for batch in small:
            loss = tr.train_step(batch)
            if first_loss is None:
                first_loss = loss.item()
            last_loss = loss.item()
    assert last_loss < first_loss * 0.5, (first_loss, last_loss)


def test_training_loop_and_summaries(tmp_path, toy_corpus):
    tr, train_ds, test_ds = _mk_train(tmp_path, toy_corpus, epochs=1)
    small_train = BatchedDataset(train_ds.pairs[:24], 8, shuffle=True)
    tr.training_loop(small_train, test_ds)
    # C15: summary event files exist in both log dirs
    for d in ("logs/train", "logs/test"):
        files = os.listdir(tmp_path / d)
        assert any(f.startswith("events.out.tfevents") for f in files)
    # C16: checkpoint written at final epoch (Q7 intent)
    assert tr.ckpt_manager.latest_checkpoint is not None


def test_predict_greedy(tmp_path, toy_corpus):
    tr, *_ = _mk_train(tmp_path, toy_corpus)
    out = tr.predict("one two three")
    assert out.dim() == 1
    assert out[0].item() == tr.tgt_tokenizer.vocab_size  # tgt start token (Q6)
    assert out.shape[0] <= 1 + tr.max_decode_len + 1


def test_sigterm_checkpoints_and_stops(tmp_path, toy_corpus):
    """SURVEY §5 failure handling: a stop request mid-epoch checkpoints and
    exits the loop cleanly; a fresh Train resumes from that step."""
    import signal

    tr, train_ds, test_ds = _mk_train(tmp_path, toy_corpus, epochs=50)
    tr.install_signal_handler()
    try:
        os.kill(os.getpid(), signal.SIGTERM)  # handler sets the stop flag
        tr.training_loop(train_ds, test_ds)
        assert tr.ckpt_manager.latest_checkpoint is not None
        tr2, *_ = _mk_train(tmp_path, toy_corpus, epochs=50)
        meta = tr2.load_ckpt()
        assert meta is not None and meta.get("step", 0) >= 1
    finally:
        signal.signal(signal.SIGTERM, signal.SIG_DFL)
        signal.signal(signal.SIGINT, signal.default_int_handler)


def test_trace_steps_writes_chrome_trace(tmp_path, toy_corpus):
    """SURVEY §5 tracing: torch.profiler trace of a few steps."""
    tr, train_ds, _ = _mk_train(tmp_path, toy_corpus)
    path = tr.trace_steps(train_ds, 2, str(tmp_path / "traces"))
    assert os.path.exists(path) and os.path.getsize(path) > 100


def test_predict_batch(tmp_path, toy_corpus):
    """predict() with a list runs one padded batch through the KV-cached
    decoder; per-row results must match single-sentence decodes."""
    import torch

    tr, *_ = _mk_train(tmp_path, toy_corpus)
    sents = ["one two three", "four", "two three four one two"]
    batch_out = tr.predict(sents)
    assert batch_out.dim() == 2 and batch_out.shape[0] == 3
    for i, s in enumerate(sents):
        single = tr.predict(s)
        n = min(len(single), batch_out.shape[1])
        assert torch.equal(batch_out[i, :n], single[:n]), (i, s)


def test_train_entry_point_subprocess(tmp_path, toy_corpus):
    """C25: `python train.py` end-to-end — flags, dataset+tokenizer build,
    training loop, checkpoint write, predict smoke, export — as the user
    would run it."""
    import subprocess
    import sys
    import os

    out = subprocess.run(
        [sys.executable, "train.py",
         "--dataset_path", str(toy_corpus),
         "--src_vocab_file", str(tmp_path / "sv"),
         "--tgt_vocab_file", str(tmp_path / "tv"),
         "--ckpt_path", str(tmp_path / "ckpt"),
         "--epochs", "1", "--num_layers", "1", "--d_model", "32",
         "--dff", "64", "--num_heads", "2", "--batch_size", "8",
         "--sequence_length", "40", "--noenable_function",
         "--steps_per_epoch", "3"],
        capture_output=True, text=True, timeout=600,
        env={**os.environ, "TFMX_EXPORT_DIR": str(tmp_path / "model")})
    assert out.returncode == 0, out.stderr[-3000:]
    assert os.path.isdir(tmp_path / "ckpt"), "no checkpoint directory"
    assert "Epoch 1" in out.stdout or "epoch" in out.stdout.lower(), \
        out.stdout[-1500:]


def test_checkpoint_cadence_q7_intent(tmp_path):
    """the reference's precedence bug saved every epoch
    EXCEPT multiples of 5; the implemented intent is save at epochs
    5, 10, ... and at the last epoch."""
    from transformer_amd.models import Transformer
    from transformer_amd.runtime import Train

    class Tok:
        vocab_size = 30
        def encode(self, s):
            return [1]
        def decode(self, ids):
            return "x"

    saved = []
    torch.manual_seed(0)
    m = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                    input_vocab_size=32, target_vocab_size=32, rate=0.0,
                    max_position=16)
    tr = Train(epochs=7, enable_function=False, transformer=m,
               src_tokenizer=Tok(), tgt_tokenizer=Tok(), batch_size=2,
               train_log_dir=None, test_log_dir=None, max_ckpt_keep=3,
               ckpt_path=str(tmp_path), d_model=16)
    tr.ckpt_manager.save = lambda step, epoch=None: saved.append(epoch)
    for epoch in range(7):
        tr._save_if_due(epoch, step=epoch + 1)
    assert saved == [4, 6], saved  # epochs 5 and 7 (0-indexed 4, 6)


def test_eval_runs_fixed_batch_count_q8(tmp_path):
    """the reference's distributed eval effectively ran ONE
    test step (never-reset modulo counter); here eval consumes exactly
    min(eval_steps, len(dataset)) batches every time it runs."""
    from transformer_amd.models import Transformer
    from transformer_amd.runtime import Train

    class Tok:
        vocab_size = 30
        def encode(self, s):
            return [1]
        def decode(self, ids):
            return "x"

    torch.manual_seed(0)
    m = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                    input_vocab_size=32, target_vocab_size=32, rate=0.0,
                    max_position=16)
    tr = Train(epochs=1, enable_function=False, transformer=m,
               src_tokenizer=Tok(), tgt_tokenizer=Tok(), batch_size=2,
               train_log_dir=None, test_log_dir=None, max_ckpt_keep=1,
               ckpt_path=str(tmp_path), d_model=16, eval_steps=3)
    batch = (torch.randint(1, 30, (2, 6)), torch.randint(1, 30, (2, 6)))
    consumed = []

    def gen(n):
        for i in range(n):
            consumed.append(i)
            yield batch

    tr._run_eval(gen(10))
    assert len(consumed) == 3, consumed        # capped at eval_steps
    consumed.clear()
    tr._run_eval(gen(2))
    assert len(consumed) == 2, consumed        # short dataset: all of it
    assert tr.test_loss.count == 2             # metrics reset per eval
