"""Dataset pipeline (capability parity with reference utils.py:65-161).

read_data          — text-pair loading via glob (reference utils.py:65-80)
load_or_create_tokenizer — build-from-corpus at 2**15 or load persisted
                     `.subwords` files (reference utils.py:83-111)
load_dataset       — encode with start/end tokens (= vocab_size,
                     vocab_size+1), length-filter, shuffle, padded-batch with
                     pad id 0 (reference utils.py:114-161)

Fixes over the reference: when the test files are missing,
a held-out split is carved from the train pairs instead of silently
evaluating on an empty dataset.  Encoded corpora are cached to a binary file
so steady-state startup skips re-tokenization.

SyntheticSeq2SeqDataset provides the synthetic-token benchmark path
(BASELINE.json configs 2-5): random tokens of the benchmark shape with
random-init weights, no network needed.
"""

from __future__ import annotations

import glob
import hashlib
import os
import random

import torch

from .tokenizer import SubwordTokenizer


def read_data(src_file: str, tgt_file: str):
    """Read zipped parallel text files (reference utils.py:65-80); glob
    semantics preserved — a missing pattern yields an empty list."""
    def _read(pattern):
        lines = []
        for path in sorted(glob.glob(pattern)):
            with open(path, encoding="utf-8") as f:
                lines.extend(line.rstrip("\n") for line in f)
        return lines

    src, tgt = _read(src_file), _read(tgt_file)
    n = min(len(src), len(tgt))
    return list(zip(src[:n], tgt[:n]))


def load_or_create_tokenizer(pairs, src_vocab_file: str, tgt_vocab_file: str,
                             target_vocab_size: int = 2 ** 15):
    """Build two subword tokenizers from the corpus or load the persisted
    `.subwords` files (reference utils.py:83-111)."""
    toks = []
    for vocab_file, idx in ((src_vocab_file, 0), (tgt_vocab_file, 1)):
        if os.path.exists(vocab_file + ".subwords"):
            toks.append(SubwordTokenizer.load_from_file(vocab_file))
        else:
            tok = SubwordTokenizer.build_from_corpus(
                (p[idx] for p in pairs), target_vocab_size)
            tok.save_to_file(vocab_file)
            toks.append(tok)
    return toks[0], toks[1]


class BatchedDataset:
    """Shuffle + padded-batch over encoded pairs (reference utils.py:154-159:
    shuffle(buffer).padded_batch(batch, pad id 0)).  Supports DP sharding:
    rank r of world W sees batches of batch_size//W drawn from its shard
    (the reference's MirroredStrategy input split)."""

    def __init__(self, encoded_pairs, batch_size: int, shuffle: bool = True,
                 buffer_size: int = 100000, seed: int = 0,
                 rank: int = 0, world_size: int = 1, drop_last: bool = False):
        if world_size > 1 and batch_size % world_size != 0:
            raise ValueError(  # reference distributed_train.py:154-158
                f"Batch size {batch_size} not divisible by world size {world_size}")
        self.pairs = encoded_pairs
        self.global_batch = batch_size
        self.batch_size = batch_size // world_size
        self.shuffle = shuffle
        self.buffer_size = buffer_size
        self.seed = seed
        self.rank, self.world_size = rank, world_size
        self.drop_last = drop_last
        self.epoch = 0

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        n = len(self.pairs) // self.global_batch
        if not self.drop_last and len(self.pairs) % self.global_batch:
            n += 1
        return n

    def __iter__(self):
        order = list(range(len(self.pairs)))
        if self.shuffle:
            random.Random(self.seed + self.epoch).shuffle(order)
        gb = self.global_batch
        for i in range(0, len(order), gb):
            idx = order[i:i + gb]
            if len(idx) < gb and self.drop_last:
                break
            # A ragged tail smaller than the world would leave some ranks
            # with an empty shard — those ranks would skip the batch while
            # the others step, desynchronizing the collectives (deadlock).
            # The decision must be UNIFORM across ranks: every rank drops
            # such a tail.  For len(idx) >= world the floor-split below
            # gives every rank at least one item.
            if self.world_size > 1 and len(idx) < self.world_size:
                continue
            # per-replica shard of the global batch
            shard = idx[self.rank * len(idx) // self.world_size:
                        (self.rank + 1) * len(idx) // self.world_size]
            src = [self.pairs[j][0] for j in shard]
            tgt = [self.pairs[j][1] for j in shard]
            yield _pad_batch(src), _pad_batch(tgt)


def _pad_batch(seqs):
    mx = max(len(s) for s in seqs)
    out = torch.zeros(len(seqs), mx, dtype=torch.int64)
    for i, s in enumerate(seqs):
        out[i, : len(s)] = torch.as_tensor(s, dtype=torch.int64)
    return out


def _encode_corpus(pairs, src_tok, tgt_tok, cache_path=None):
    """Encode with start/end tokens = vocab_size / vocab_size+1 (reference
    utils.py:99-103); binary cache keyed by corpus+vocab hash."""
    if cache_path:
        h = hashlib.sha1()
        h.update(str(len(pairs)).encode())
        for s, t in pairs[:64]:
            h.update(s.encode()); h.update(t.encode())
        h.update(str(src_tok.vocab_size).encode())
        h.update(str(tgt_tok.vocab_size).encode())
        cache_file = f"{cache_path}.{h.hexdigest()[:12]}.pt"
        if os.path.exists(cache_file):
            return torch.load(cache_file, weights_only=True)
    ss, se = src_tok.vocab_size, src_tok.vocab_size + 1
    ts, te = tgt_tok.vocab_size, tgt_tok.vocab_size + 1
    enc = [([ss] + src_tok.encode(s) + [se], [ts] + tgt_tok.encode(t) + [te])
           for s, t in pairs]
    if cache_path:
        tmp = f"{cache_file}.{os.getpid()}.tmp"  # pid-unique: no rank races
        torch.save(enc, tmp)
        os.replace(tmp, cache_file)
    return enc


def load_dataset(dataset_path: str, src_vocab_file: str, tgt_vocab_file: str,
                 sequence_length: int = 50, batch_size: int = 64,
                 buffer_size: int = 100000, seed: int = 0,
                 rank: int = 0, world_size: int = 1):
    """Returns (train_ds, test_ds, src_tok, tgt_tok) — the reference's
    load_dataset contract (utils.py:114-161)."""
    train_pairs = read_data(os.path.join(dataset_path, "src-train.txt"),
                            os.path.join(dataset_path, "tgt-train.txt"))
    test_pairs = read_data(os.path.join(dataset_path, "src-test.txt"),
                           os.path.join(dataset_path, "tgt-test.txt"))
    if not train_pairs:
        raise FileNotFoundError(f"no training pairs under {dataset_path}")
    if not test_pairs:
        # Q10 fix: hold out the last 2% (>=1 batch) instead of empty eval.
        k = max(batch_size, len(train_pairs) // 50)
        train_pairs, test_pairs = train_pairs[:-k], train_pairs[-k:]

    # In DP, rank 0 builds the vocab + encoded caches alone and the other
    # ranks wait at a barrier, then hit the persisted files — concurrent
    # builders raced on the shared cache/vocab writes (os.replace of a
    # just-replaced tmp) and wasted (world-1) duplicate tokenizations.
    import torch.distributed as dist
    is_dist = (dist.is_available() and dist.is_initialized()
               and dist.get_world_size() > 1)
    if is_dist and dist.get_rank() != 0:
        dist.barrier()  # rank 0 building
    src_tok, tgt_tok = load_or_create_tokenizer(train_pairs, src_vocab_file,
                                                tgt_vocab_file)
    cache = os.path.join(dataset_path, "encoded_cache")
    enc_train = _encode_corpus(train_pairs, src_tok, tgt_tok, cache + "_train")
    enc_test = _encode_corpus(test_pairs, src_tok, tgt_tok, cache + "_test")
    if is_dist and dist.get_rank() == 0:
        dist.barrier()  # release the waiting ranks

    # length filter (train only, reference utils.py:145-153)
    enc_train = [(s, t) for s, t in enc_train
                 if len(s) <= sequence_length and len(t) <= sequence_length]

    train_ds = BatchedDataset(enc_train, batch_size, shuffle=True,
                              buffer_size=buffer_size, seed=seed,
                              rank=rank, world_size=world_size)
    test_ds = BatchedDataset(enc_test, batch_size, shuffle=False,
                             rank=rank, world_size=world_size)
    return train_ds, test_ds, src_tok, tgt_tok


class SyntheticSeq2SeqDataset:
    """Random-token src/tgt batches of a fixed shape (BASELINE.json: synthetic
    data, random-init weights).  Tokens uniform in [2, vocab); position 0 is
    a start token (vocab), last is end (vocab+1); no padding (full-length
    sequences) so every step does identical work."""

    def __init__(self, vocab_size: int, batch_size: int, seq_len: int,
                 steps_per_epoch: int = 100, seed: int = 0,
                 rank: int = 0, world_size: int = 1):
        if world_size > 1 and batch_size % world_size != 0:
            raise ValueError(
                f"Batch size {batch_size} not divisible by world size {world_size}")
        self.vocab = vocab_size
        self.global_batch = batch_size
        self.batch_size = batch_size // world_size
        self.seq_len = seq_len
        self.steps = steps_per_epoch
        self.seed = seed + rank * 10007
        self.epoch = 0

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        return self.steps

    def __iter__(self):
        g = torch.Generator().manual_seed(self.seed + self.epoch)
        for _ in range(self.steps):
            def mk():
                x = torch.randint(2, self.vocab, (self.batch_size, self.seq_len),
                                  generator=g, dtype=torch.int64)
                x[:, 0] = self.vocab
                x[:, -1] = self.vocab + 1
                return x
            yield mk(), mk()
