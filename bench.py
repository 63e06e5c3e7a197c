"""Benchmark driver contract: flagship training step, tokens/sec whole-job.

    python bench.py --gpus N --steps K --warmup W
    (N>1 is launched by the driver as torchrun --nproc-per-node N ... )

Measures the BASELINE.json metric — tokens/sec (whole node) for
Transformer-base NMT training — on synthetic src/tgt token data (no
network) with random-init weights, bf16 compute, one rank per GPU over
RCCL, weak scaling (fixed per-GPU batch).  Tokens counted per step =
global_batch * (src_len + tgt_len): every counted token is embedded and
processed by the full encoder or decoder stack each step.

Rank 0 prints exactly ONE JSON line with the result.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


MODELS = {
    "base": dict(num_layers=6, d_model=512, num_heads=8, dff=2048),
    "big": dict(num_layers=6, d_model=1024, num_heads=16, dff=4096),
    "tiny": dict(num_layers=2, d_model=128, num_heads=4, dff=256),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--model", type=str, default="base", choices=list(MODELS))
    p.add_argument("--seq_len", type=int, default=256)
    p.add_argument("--batch", type=int, default=64,
                   help="per-GPU batch (weak scaling)")
    p.add_argument("--vocab", type=int, default=32768)
    p.add_argument("--label_smoothing", type=float, default=0.1)
    p.add_argument("--dropout", type=float, default=0.1)
    p.add_argument("--graph", action="store_true",
                   help="capture the step into a hipGraph and replay "
                        "(single-GPU only)")
    args = p.parse_args()

    from transformer_amd.models import Transformer
    from transformer_amd.parallel import init_distributed, BucketedDataParallel
    from transformer_amd.runtime import NoamAdam
    from transformer_amd import ops

    rank, local_rank, world = init_distributed()
    assert world == args.gpus or world == 1, \
        f"launched world {world} != --gpus {args.gpus}"
    use_cuda = torch.cuda.is_available()
    # clamp to the visible device count so the world-2-on-one-GPU gloo
    # smoke (tools/dp_parity.py setup) runs; on a real node each rank
    # keeps its own GPU
    dev_idx = min(local_rank, max(torch.cuda.device_count() - 1, 0)) \
        if use_cuda else 0
    device = torch.device(f"cuda:{dev_idx}") if use_cuda else torch.device("cpu")
    dtype = torch.bfloat16 if use_cuda else torch.float32
    if use_cuda:
        ops.ext()  # fail loudly if the HIP extension is missing

    torch.manual_seed(1234)
    cfg = MODELS[args.model]
    vocab = args.vocab + 2
    model = Transformer(input_vocab_size=vocab, target_vocab_size=vocab,
                        rate=args.dropout, max_position=max(4096, args.seq_len),
                        **cfg).to(device, dtype)
    opt = NoamAdam(model, cfg["d_model"], warmup_steps=60000, use_flat=True)
    ddp = None
    if world > 1:
        ddp = BucketedDataParallel(opt.flat)
        ddp.broadcast_parameters()

    S = args.seq_len
    B = args.batch
    global_batch = B * world
    g = torch.Generator(device="cpu").manual_seed(4321 + rank)
    n_distinct = 8

    def make_batch():
        x = torch.randint(2, args.vocab, (B, S), generator=g, dtype=torch.int64)
        x[:, 0] = args.vocab
        x[:, -1] = args.vocab + 1
        return x.to(device, non_blocking=True)

    batches = [(make_batch(), make_batch()) for _ in range(n_distinct)]

    captured = None
    if args.graph and world == 1 and use_cuda:
        from transformer_amd.runtime.graph import CapturedTrainStep

        def loss_fn(real, pred):
            return ops.masked_cross_entropy(pred, real, global_batch,
                                            args.label_smoothing)

        captured = CapturedTrainStep(model, opt, loss_fn, (B, S), (B, S),
                                     device)

    def step(i):
        src, tar = batches[i % n_distinct]
        if captured is not None:
            return captured(src, tar)
        tar_inp = tar[:, :-1].contiguous()
        tar_real = tar[:, 1:].contiguous()
        logits, _ = model((src, tar_inp), training=True)
        loss = ops.masked_cross_entropy(logits, tar_real, global_batch,
                                        args.label_smoothing)
        opt.zero_grad()
        loss.backward()
        if ddp is not None:
            ddp.finalize()
        opt.step()
        return loss

    for i in range(args.warmup):
        step(i)

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    last = None
    for i in range(args.steps):
        last = step(args.warmup + i)
    if use_cuda:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    if world > 1:
        dist.barrier()

    elapsed = t1 - t0
    # MAX over ranks (tensor must live on the backend's device: RCCL
    # rejects CPU tensors)
    if world > 1:
        red_dev = device if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=red_dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000.0
    tokens_per_step = global_batch * (S + S)
    value = tokens_per_step / (elapsed / args.steps)
    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec (whole node), Transformer-%s NMT training"
                      % args.model,
            "value": value,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic (random src/tgt tokens, random-init weights; "
                    "tokens = global_batch*(src_len+tgt_len))",
            "loss": float(last.detach()),
            "config": {
                "model": f"transformer-{args.model}",
                "num_layers": cfg["num_layers"], "d_model": cfg["d_model"],
                "num_heads": cfg["num_heads"], "dff": cfg["dff"],
                "global_batch": global_batch, "seq_len": S,
                "vocab": vocab,
                "label_smoothing": args.label_smoothing,
                "dropout": args.dropout,
                "parallelism": f"dp{world}",
            },
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
