"""Single-device training entry point — flag-compatible with the reference's
train.py (reference train.py:216-251): load dataset, build Transformer, run
the training loop, restore latest checkpoint, greedy-predict smoke test,
export the model.

Usage examples:
    python train.py --num_layers 2 --d_model 128 --epochs 1        # toy CPU
    python train.py --synthetic_data --d_model 512 --num_layers 6  # bench-ish
"""

from __future__ import annotations

import datetime
import os
import sys

import torch

from transformer_amd.config import parse_flags, flags_dict
from transformer_amd.data import load_dataset, SyntheticSeq2SeqDataset
from transformer_amd.models import Transformer
from transformer_amd.runtime import Train, export_model


class _SyntheticTok:
    """Tokenizer stand-in for --synthetic_data (vocab only; encode maps words
    to arbitrary in-vocab ids so predict still runs)."""

    def __init__(self, vocab_size):
        self.vocab_size = vocab_size

    def encode(self, text):
        # stable across processes/runs (Python hash() is PYTHONHASHSEED-
        # dependent, which would desync DP ranks on predict)
        import zlib
        return [2 + (zlib.crc32(w.encode()) % (self.vocab_size - 2))
                for w in text.split()]

    def decode(self, ids):
        return " ".join(str(int(i)) for i in ids)


def pick_device_dtype(device_flag, dtype_flag):
    if device_flag:
        device = torch.device(device_flag)
    else:
        device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    if dtype_flag:
        dtype = {"bf16": torch.bfloat16, "fp32": torch.float32}[dtype_flag]
    else:
        dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    return device, dtype


def main(epochs, enable_function, buffer_size, batch_size, sequence_length,
         dataset_path, src_vocab_file, tgt_vocab_file, num_layers, d_model,
         dff, num_heads, max_ckpt_keep, ckpt_path, dropout_rate,
         warmup_steps=60000, label_smoothing=0.0, device=None, dtype=None,
         seed=1234, log_interval=100, eval_steps=50, synthetic_data=False,
         synthetic_vocab=32768, steps_per_epoch=100, max_decode_len=10,
         trace_dir=None, debug_sync=False, **_ignored):
    if debug_sync:
        # race/fault localization: every kernel launch is
        # serialized and synchronous so a fault is attributed to the
        # launching line, and reductions run in a deterministic order.
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"
    torch.manual_seed(seed)
    device, dtype = pick_device_dtype(device, dtype)

    ts = datetime.datetime.now().strftime("%Y%m%d-%H%M%S")
    train_log_dir = os.path.join("logs", "gradient_tape", ts, "train")
    test_log_dir = os.path.join("logs", "gradient_tape", ts, "test")

    if synthetic_data:
        src_tok = tgt_tok = _SyntheticTok(synthetic_vocab)
        train_ds = SyntheticSeq2SeqDataset(synthetic_vocab, batch_size,
                                           sequence_length, steps_per_epoch,
                                           seed=seed)
        test_ds = SyntheticSeq2SeqDataset(synthetic_vocab, batch_size,
                                          sequence_length, max(1, eval_steps),
                                          seed=seed + 1)
    else:
        train_ds, test_ds, src_tok, tgt_tok = load_dataset(
            dataset_path, src_vocab_file, tgt_vocab_file,
            sequence_length, batch_size, buffer_size, seed)

    input_vocab_size = src_tok.vocab_size + 2
    target_vocab_size = tgt_tok.vocab_size + 2
    max_position = max(4096, sequence_length)

    transformer = Transformer(num_layers, d_model, num_heads, dff,
                              input_vocab_size, target_vocab_size,
                              rate=dropout_rate, max_position=max_position)
    transformer = transformer.to(device=device, dtype=dtype)

    train = Train(epochs, enable_function, transformer, src_tok, tgt_tok,
                  batch_size, train_log_dir, test_log_dir, max_ckpt_keep,
                  ckpt_path, d_model, warmup_steps=warmup_steps,
                  label_smoothing=label_smoothing, device=device,
                  log_interval=log_interval, eval_steps=eval_steps,
                  max_decode_len=max_decode_len)
    train.load_ckpt()
    train.install_signal_handler()  # SIGTERM/SIGINT -> checkpoint + exit
    if trace_dir:
        train.trace_steps(train_ds, 5, trace_dir)
    train.training_loop(train_ds, test_ds)
    print(train.predict("he go to school"))
    export_model(transformer, "model", {
        "num_layers": num_layers, "d_model": d_model, "num_heads": num_heads,
        "dff": dff, "input_vocab_size": input_vocab_size,
        "target_vocab_size": target_vocab_size, "dropout_rate": dropout_rate,
        "max_position": max_position,
    })


def run_main(argv=None):
    args = parse_flags(argv)
    main(**flags_dict(args))


if __name__ == "__main__":
    run_main(sys.argv[1:])
