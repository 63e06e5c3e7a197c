"""Post-convergence translation sampler (VERDICT.md item 4 evidence):
restore the latest checkpoint and greedy-decode a few TRAINING sentences
from the bundled En->De corpus, printing src / reference / hypothesis.

    python tools/translate_sample.py [--ckpt_path model_dist] [--n 5]
"""
import argparse
import sys

import torch

sys.path.insert(0, ".")

from transformer_amd.data import load_dataset  # noqa: E402
from transformer_amd.models import Transformer  # noqa: E402
from transformer_amd.runtime import Train  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ckpt_path", default="model_dist")
    p.add_argument("--dataset_path", default="data/")
    p.add_argument("--num_layers", type=int, default=4)
    p.add_argument("--d_model", type=int, default=512)
    p.add_argument("--dff", type=int, default=1024)
    p.add_argument("--num_heads", type=int, default=4)
    p.add_argument("--sequence_length", type=int, default=50)
    p.add_argument("--n", type=int, default=5)
    a = p.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dt = torch.bfloat16 if dev.type == "cuda" else torch.float32
    _, _, src_tok, tgt_tok = load_dataset(
        a.dataset_path, "src_vocab.txt", "tgt_vocab.txt",
        sequence_length=a.sequence_length, batch_size=64,
        buffer_size=1000)
    model = Transformer(
        num_layers=a.num_layers, d_model=a.d_model, num_heads=a.num_heads,
        dff=a.dff, input_vocab_size=src_tok.vocab_size + 2,
        target_vocab_size=tgt_tok.vocab_size + 2, rate=0.0,
        max_position=max(128, a.sequence_length + 8)).to(dev, dt)
    tr = Train(epochs=1, enable_function=False, transformer=model,
               src_tokenizer=src_tok, tgt_tokenizer=tgt_tok, batch_size=64,
               train_log_dir=None, test_log_dir=None, max_ckpt_keep=5,
               ckpt_path=a.ckpt_path, d_model=a.d_model,
               max_decode_len=40, device=dev)
    meta = tr.load_ckpt()
    print("restored:", meta)
    src_lines = open(a.dataset_path + "src-train.txt").read().splitlines()
    tgt_lines = open(a.dataset_path + "tgt-train.txt").read().splitlines()
    idxs = [3, 57, 1001, 4242, 9000][: a.n]
    for i in idxs:
        src = src_lines[i]
        if len(src.split()) > 30:
            src = " ".join(src.split()[:30])
        out = tr.predict(src)
        ids = [t for t in out.tolist()
               if 0 < t < tgt_tok.vocab_size]
        hyp = tgt_tok.decode(ids)
        print(f"--- [{i}]")
        print("SRC:", src)
        print("REF:", tgt_lines[i][:200])
        print("HYP:", hyp[:200])


if __name__ == "__main__":
    main()
