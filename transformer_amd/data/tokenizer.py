"""Subword tokenizer — capability equivalent of tfds SubwordTextEncoder.

The reference builds two SubwordTextEncoders from the parallel corpus at a
target vocab of 2**15 and persists them to `<name>.subwords` files (reference
utils.py:83-111).  This is a from-scratch greedy longest-match subword
tokenizer with the same capability surface:

    build_from_corpus(lines, target_vocab_size)
    encode(text) -> list[int]      (ids >= 1; 0 is reserved for padding)
    decode(ids) -> text
    vocab_size                      (dataset start/end tokens are vocab_size
                                     and vocab_size+1, reference utils.py:99)
    save_to_file(prefix) / load_from_file(prefix)   (.subwords files)

ID layout: 0 = pad, 1..S = subwords, S+1..S+256 = byte fallback tokens.
Round-trips any input text (byte fallback covers OOV characters).
"""

from __future__ import annotations

import collections
import re
import os

_WORD_RE = re.compile(r"\S+")
_MAX_SUBWORD_LEN = 20


def _words_with_markers(line: str):
    """Split on whitespace; append '_' end-of-word marker (decode joins on
    it).  Escapes: backslash -> '\\\\' first, then '_' -> '\\u', so a
    literal backslash-u in the input cannot collide with the underscore
    escape (caught by the hypothesis round-trip fuzz)."""
    for w in _WORD_RE.findall(line):
        yield w.replace("\\", "\\\\").replace("_", "\\u") + "_"


def _unescape(text: str) -> str:
    out = []
    i, n = 0, len(text)
    while i < n:
        c = text[i]
        if c == "\\" and i + 1 < n:
            nxt = text[i + 1]
            if nxt == "\\":
                out.append("\\")
                i += 2
                continue
            if nxt == "u":
                out.append("_")
                i += 2
                continue
        out.append(c)
        i += 1
    return "".join(out)


class SubwordTokenizer:
    def __init__(self, subwords: list[str]):
        self._subwords = list(subwords)
        self._index = {s: i + 1 for i, s in enumerate(self._subwords)}
        self._max_len = max((len(s) for s in self._subwords), default=1)

    # -- properties ---------------------------------------------------------
    @property
    def vocab_size(self) -> int:
        return 1 + len(self._subwords) + 256  # pad + subwords + byte fallback

    @property
    def subwords(self):
        return list(self._subwords)

    # -- encode / decode ----------------------------------------------------
    def _encode_word(self, word: str, out: list[int]):
        i, n = 0, len(word)
        base_byte = 1 + len(self._subwords)
        while i < n:
            j = min(n, i + self._max_len)
            while j > i:
                tok = self._index.get(word[i:j])
                if tok is not None:
                    out.append(tok)
                    i = j
                    break
                j -= 1
            else:
                for b in word[i].encode("utf-8"):
                    out.append(base_byte + b)
                i += 1

    def encode(self, text: str) -> list[int]:
        out: list[int] = []
        for w in _words_with_markers(text):
            self._encode_word(w, out)
        return out

    def decode(self, ids) -> str:
        parts: list[str] = []
        pend_bytes: list[int] = []
        base_byte = 1 + len(self._subwords)

        def flush():
            if pend_bytes:
                parts.append(bytes(pend_bytes).decode("utf-8", errors="replace"))
                pend_bytes.clear()

        for t in ids:
            t = int(t)
            if t <= 0 or t >= self.vocab_size:
                continue  # pad / out-of-range (start/end tokens)
            if t >= base_byte:
                pend_bytes.append(t - base_byte)
            else:
                flush()
                parts.append(self._subwords[t - 1])
        flush()
        return _unescape("".join(parts).replace("_", " ")).rstrip()

    # -- persistence (the reference's `.subwords` files, utils.py:92-97) ----
    def save_to_file(self, prefix: str):
        # atomic (tmp + rename, pid-unique): concurrent builders (e.g. DP
        # ranks racing before the rank-0 gate existed) can never interleave
        # writes into a corrupt vocab file.
        import os
        tmp = f"{prefix}.subwords.{os.getpid()}.tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            for s in self._subwords:
                f.write("'" + s.replace("\\", "\\\\").replace("'", "\\'") + "'\n")
        os.replace(tmp, prefix + ".subwords")

    @classmethod
    def load_from_file(cls, prefix: str) -> "SubwordTokenizer":
        subwords = []
        with open(prefix + ".subwords", encoding="utf-8") as f:
            for line in f:
                line = line.rstrip("\n")
                if len(line) >= 2 and line[0] == "'" and line[-1] == "'":
                    line = line[1:-1]
                subwords.append(line.replace("\\'", "'").replace("\\\\", "\\"))
        return cls(subwords)

    # -- corpus build -------------------------------------------------------
    @classmethod
    def build_from_corpus(cls, lines, target_vocab_size: int = 2 ** 15,
                          iterations: int = 3) -> "SubwordTokenizer":
        """Iterative greedy subword induction (same spirit as tfds): start
        from characters, repeatedly (a) segment the corpus greedily with the
        current vocab, (b) count all candidate substrings beginning at
        segmentation points, (c) keep the highest-count candidates up to the
        target size (longer subwords preferred on ties)."""
        word_counts: collections.Counter[str] = collections.Counter()
        for line in lines:
            for w in _words_with_markers(line):
                word_counts[w] += 1

        # seed: all single characters
        alphabet = {c for w in word_counts for c in w}
        tok = cls(sorted(alphabet))
        for _ in range(iterations):
            cand: collections.Counter[str] = collections.Counter()
            for w, c in word_counts.items():
                ids: list[int] = []
                # positions where greedy segmentation starts a new subword
                i, n = 0, len(w)
                starts = []
                while i < n:
                    starts.append(i)
                    j = min(n, i + tok._max_len)
                    adv = 1
                    while j > i:
                        if w[i:j] in tok._index:
                            adv = j - i
                            break
                        j -= 1
                    i += adv
                for s in starts:
                    for e in range(s + 1, min(n, s + _MAX_SUBWORD_LEN) + 1):
                        cand[w[s:e]] += c
            # keep alphabet always; fill remaining slots by count
            budget = max(target_vocab_size - 257, len(alphabet))
            chosen = set(alphabet)
            for sub, c in sorted(cand.items(), key=lambda kv: (-kv[1], -len(kv[0]), kv[0])):
                if len(chosen) >= budget:
                    break
                if c < 2 and len(sub) > 1:
                    continue
                chosen.add(sub)
            # longer subwords first on equal count so greedy matching uses them
            tok = cls(sorted(chosen, key=lambda s: (-len(s), s)))
        return tok
