"""Generate the bundled toy parallel corpus (digit-words English -> German).

The reference ships a 10k-line En->De corpus under data/ (reference
README.md, data/src-train.txt).  We generate our own learnable toy
translation corpus instead of copying theirs: random sequences of number
words translated word-for-word, plus a small phrase table — enough signal
for the overfit/convergence tests and the CPU plumbing config.

Usage: python tools/make_toy_corpus.py [outdir] [n_lines]
"""

from __future__ import annotations

import os
import random
import sys

_EN_DE = [
    ("zero", "null"), ("one", "eins"), ("two", "zwei"), ("three", "drei"),
    ("four", "vier"), ("five", "fünf"), ("six", "sechs"), ("seven", "sieben"),
    ("eight", "acht"), ("nine", "neun"), ("ten", "zehn"),
    ("and", "und"), ("plus", "plus"), ("minus", "minus"),
    ("he", "er"), ("she", "sie"), ("goes", "geht"), ("to", "zur"),
    ("school", "schule"), ("the", "die"), ("cat", "katze"), ("dog", "hund"),
    ("sees", "sieht"), ("a", "ein"), ("house", "haus"), ("is", "ist"),
    ("big", "gross"), ("small", "klein"), ("red", "rot"), ("blue", "blau"),
]


def generate(outdir: str, n_lines: int = 2000, seed: int = 0):
    rng = random.Random(seed)
    os.makedirs(outdir, exist_ok=True)
    with open(os.path.join(outdir, "src-train.txt"), "w", encoding="utf-8") as fs, \
            open(os.path.join(outdir, "tgt-train.txt"), "w", encoding="utf-8") as ft:
        for _ in range(n_lines):
            k = rng.randint(2, 8)
            pairs = [rng.choice(_EN_DE) for _ in range(k)]
            fs.write(" ".join(p[0] for p in pairs) + "\n")
            ft.write(" ".join(p[1] for p in pairs) + "\n")


if __name__ == "__main__":
    out = sys.argv[1] if len(sys.argv) > 1 else "data"
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 2000
    generate(out, n)
    print(f"wrote {n} parallel lines to {out}/")
