"""Distributed (synchronous data-parallel) training entry point — the
capability of the reference's distributed_train.py (reference
distributed_train.py:124-179, MirroredStrategy over N GPUs), rebuilt
MI355X-native: one process per GPU via torchrun, RCCL (torch.distributed
"nccl" backend) bucketed all-reduce over xGMI overlapped with backward
(transformer_amd/parallel/ddp.py).

Launch (reference: `python distributed_train.py --num_gpu N`):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 distributed_train.py --num_gpu N ...
Single-process invocation with --num_gpu N re-execs itself under torchrun
for flag-compatibility with the reference CLI.
"""

from __future__ import annotations

import datetime
import os
import subprocess
import sys

import torch

from transformer_amd.config import parse_flags, flags_dict
from transformer_amd.data import load_dataset, SyntheticSeq2SeqDataset
from transformer_amd.models import Transformer
from transformer_amd.parallel import init_distributed
from transformer_amd.runtime import DistributedTrain, export_model
from train import _SyntheticTok, pick_device_dtype


def main(epochs, enable_function, buffer_size, batch_size, sequence_length,
         dataset_path, src_vocab_file, tgt_vocab_file, num_layers, d_model,
         dff, num_heads, max_ckpt_keep, ckpt_path, dropout_rate, num_gpu=1,
         warmup_steps=60000, label_smoothing=0.0, device=None, dtype=None,
         seed=1234, log_interval=100, eval_steps=50, synthetic_data=False,
         synthetic_vocab=32768, steps_per_epoch=100, max_decode_len=10,
         debug_sync=False, **_ignored):
    if debug_sync:  # race/fault localization mode
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"
    rank, local_rank, world_size = init_distributed()
    torch.manual_seed(seed)  # same init on every rank (then X1 broadcast)
    if device is None and torch.cuda.is_available():
        # clamp to visible devices (world-2-on-one-GPU gloo smoke; on a
        # real node each rank keeps its own GPU)
        device = f"cuda:{min(local_rank, torch.cuda.device_count() - 1)}"
    device, dtype = pick_device_dtype(device, dtype)

    # global-batch divisibility guard (reference distributed_train.py:154-158)
    if batch_size % max(world_size, 1) != 0:
        raise ValueError(
            f"Batch size {batch_size} not divisible by number of replicas "
            f"{world_size}")

    ts = datetime.datetime.now().strftime("%Y%m%d-%H%M%S")
    train_log_dir = os.path.join("logs", "gradient_tape", ts, "train")
    test_log_dir = os.path.join("logs", "gradient_tape", ts, "test")

    if synthetic_data:
        src_tok = tgt_tok = _SyntheticTok(synthetic_vocab)
        train_ds = SyntheticSeq2SeqDataset(synthetic_vocab, batch_size,
                                           sequence_length, steps_per_epoch,
                                           seed=seed, rank=rank,
                                           world_size=world_size)
        test_ds = SyntheticSeq2SeqDataset(synthetic_vocab, batch_size,
                                          sequence_length, max(1, eval_steps),
                                          seed=seed + 1, rank=rank,
                                          world_size=world_size)
    else:
        train_ds, test_ds, src_tok, tgt_tok = load_dataset(
            dataset_path, src_vocab_file, tgt_vocab_file, sequence_length,
            batch_size, buffer_size, seed, rank=rank, world_size=world_size)

    input_vocab_size = src_tok.vocab_size + 2
    target_vocab_size = tgt_tok.vocab_size + 2
    max_position = max(4096, sequence_length)

    transformer = Transformer(num_layers, d_model, num_heads, dff,
                              input_vocab_size, target_vocab_size,
                              rate=dropout_rate, max_position=max_position)
    transformer = transformer.to(device=device, dtype=dtype)

    train = DistributedTrain(
        epochs, enable_function, transformer, src_tok, tgt_tok, batch_size,
        train_log_dir, test_log_dir, max_ckpt_keep, ckpt_path, d_model,
        warmup_steps=warmup_steps, label_smoothing=label_smoothing,
        device=device, log_interval=log_interval, eval_steps=eval_steps,
        max_decode_len=max_decode_len, is_rank0=(rank == 0))
    train.load_ckpt()
    train.training_loop(train_ds, test_ds)
    if rank == 0:
        print(train.predict(["he goes to school"]))
        export_model(transformer, "model", {
            "num_layers": num_layers, "d_model": d_model,
            "num_heads": num_heads, "dff": dff,
            "input_vocab_size": input_vocab_size,
            "target_vocab_size": target_vocab_size,
            "dropout_rate": dropout_rate, "max_position": max_position,
        })


def run_main(argv=None):
    args = parse_flags(
        argv, extra=lambda p: p.add_argument(
            "--num_gpu", type=int, default=4, help="Number of GPUs"))
    kwargs = flags_dict(args)
    kwargs["num_gpu"] = args.num_gpu

    # Reference CLI compatibility: plain `python distributed_train.py
    # --num_gpu N` re-execs under torchrun, one process per GPU.
    if "RANK" not in os.environ and args.num_gpu > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.num_gpu}", "--master-addr=127.0.0.1",
               "--master-port=29517", os.path.abspath(__file__)] + \
              (argv if argv is not None else sys.argv[1:])
        raise SystemExit(subprocess.call(cmd))
    main(**kwargs)


if __name__ == "__main__":
    run_main(sys.argv[1:])
