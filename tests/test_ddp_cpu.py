"""Distributed tests without a cluster: gloo backend,
world_size=2, single node — DP=2 must produce the same parameters as DP=1
on the equal global batch (the reference's MirroredStrategy semantics,
SURVEY.md §2.4 X1/X2 + §8 Q4 loss scaling)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from transformer_amd.models import Transformer
from transformer_amd.runtime import NoamAdam
from transformer_amd.parallel import BucketedDataParallel
from transformer_amd.ops import reference as R


def _model():
    torch.manual_seed(0)
    return Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                       input_vocab_size=50, target_vocab_size=50, rate=0.0,
                       max_position=32)


def _batches(n_steps, global_batch=4, seq=6):
    torch.manual_seed(99)
    out = []
    for _ in range(n_steps):
        out.append((torch.randint(1, 50, (global_batch, seq)),
                    torch.randint(1, 50, (global_batch, seq))))
    return out


def _train_steps(model, opt, batches, ddp=None, rank=0, world=1):
    losses = []
    for inp, tar in batches:
        b = inp.shape[0]
        lo, hi = rank * b // world, (rank + 1) * b // world
        logits, _ = model((inp[lo:hi], tar[lo:hi]), training=True)
        loss = R.masked_cross_entropy(logits, tar[lo:hi], batch_size=b)
        opt.zero_grad()
        loss.backward()
        if ddp is not None:
            ddp.finalize()
        opt.step()
        losses.append(loss.item())
    return losses


def _worker(rank, world, tmpdir, q):
    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/pg",
                            rank=rank, world_size=world)
    try:
        model = _model()
        opt = NoamAdam(model, 16, warmup_steps=10, use_flat=True)
        ddp = BucketedDataParallel(opt.flat, bucket_mb=0.01)  # many buckets
        ddp.broadcast_parameters()
        # one backward + all-reduce, capture the synchronized gradient
        (inp, tar) = _batches(1)[0]
        b = inp.shape[0]
        lo, hi = rank * b // world, (rank + 1) * b // world
        logits, _ = model((inp[lo:hi], tar[lo:hi]), training=True)
        loss = R.masked_cross_entropy(logits, tar[lo:hi], batch_size=b)
        opt.zero_grad()
        loss.backward()
        ddp.finalize()
        if rank == 0:
            q.put((opt.flat.flat_g.detach().clone(), loss.item()))
        # then 2 full optimizer steps must run without deadlock
        _train_steps(model, opt, _batches(2), ddp=ddp, rank=rank, world=world)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_dp2_gradients_match_dp1(tmp_path):
    """After bucketed SUM all-reduce, the DP=2 gradient equals the full
    global-batch gradient (X2 semantics; Q4 loss pre-scaling makes SUM the
    correct mean — no extra division)."""
    model = _model()
    opt = NoamAdam(model, 16, warmup_steps=10, use_flat=True)
    (inp, tar) = _batches(1)[0]
    logits, _ = model((inp, tar), training=True)
    loss = R.masked_cross_entropy(logits, tar, batch_size=inp.shape[0])
    opt.zero_grad()
    loss.backward()
    base_g = opt.flat.flat_g.detach().clone()

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, 2, str(tmp_path), q))
          for r in range(2)]
    for p in ps:
        p.start()
    dp_g, dp_loss = q.get()
    for p in ps:
        p.join(120)
        assert p.exitcode == 0
    # sum-of-halves == full-batch gradient up to fp32 summation roundoff
    scale = base_g.abs().max().item()
    assert torch.allclose(base_g, dp_g, atol=1e-5 * max(scale, 1.0)), \
        (base_g - dp_g).abs().max()


def test_bucket_partition_covers_all_params():
    model = _model()
    opt = NoamAdam(model, 16, use_flat=True)
    ddp = BucketedDataParallel(opt.flat, bucket_mb=0.01)
    covered = set()
    for b in ddp.buckets:
        for p in b["params"]:
            covered.add(id(p))
        assert b["hi"] > b["lo"]
    assert covered == {id(p) for p in opt.flat.params}
    # buckets assemble in reverse registration order
    first_bucket_params = ddp.buckets[0]["params"]
    assert opt.flat.params[-1] in first_bucket_params


def test_flat_grad_aliasing_survives_backward():
    model = _model()
    opt = NoamAdam(model, 16, use_flat=True)
    inp = torch.randint(1, 50, (2, 5))
    tar = torch.randint(1, 50, (2, 5))
    for _ in range(2):  # twice: first sets, second accumulates
        logits, _ = model((inp, tar), training=True)
        loss = R.masked_cross_entropy(logits, tar, 2)
        loss.backward()
    opt.flat.check()
    assert opt.flat.flat_g.abs().sum() > 0


def _cb_worker(rank, world, tmpdir, q):
    """Drives the OPS-LAYER readiness callback path (the one the fused GPU
    kernels fire via _grad_ready) instead of autograd hooks: fill flat_g
    directly, fire callbacks in reverse order for MOST params, leave a
    straggler bucket for finalize()."""
    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/pg2",
                            rank=rank, world_size=world)
    try:
        from transformer_amd.ops import functional as F

        model = _model()
        opt = NoamAdam(model, 16, warmup_steps=10, use_flat=True)
        ddp = BucketedDataParallel(opt.flat, bucket_mb=0.01)
        ddp.broadcast_parameters()
        # fill only the PARAM regions: alignment padding between params
        # is deliberately outside every bucket (its gradient is always
        # zero in real training, so it needs no all-reduce)
        torch.manual_seed(1000 + rank)
        full = torch.randn_like(opt.flat.flat_g)
        opt.flat.flat_g.zero_()
        for pr, off in zip(opt.flat.params, opt.flat.offsets):
            opt.flat.flat_g[off:off + pr.numel()] = full[off:off + pr.numel()]
        mine = opt.flat.flat_g.detach().clone()
        # fire callbacks for all params EXCEPT the ones in the last bucket
        skip = {id(p) for p in ddp.buckets[-1]["params"]}
        for p in reversed(opt.flat.params):
            if id(p) not in skip:
                F._grad_ready(p)
        ddp.finalize()  # must launch the straggler bucket and wait all
        if rank == 0:
            q.put((mine, opt.flat.flat_g.detach().clone()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_callback_readiness_path_and_stragglers(tmp_path):
    """The kernel-enqueue readiness callbacks (used by the flat GPU path)
    launch buckets as they complete, and finalize() handles buckets whose
    params never fired — the all-reduced flat gradient must equal the sum
    of both ranks' local gradients in every bucket."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_cb_worker, args=(r, 2, str(tmp_path), q))
          for r in range(2)]
    for p in ps:
        p.start()
    mine0, reduced = q.get()
    for p in ps:
        p.join(120)
        assert p.exitcode == 0
    torch.manual_seed(1001)
    other_full = torch.randn_like(mine0)
    other = torch.zeros_like(mine0)
    # rank 1 masked the same param regions
    from transformer_amd.runtime.optimizer import FlatParams
    m = _model()
    fp = FlatParams(m)
    for pr, off in zip(fp.params, fp.offsets):
        other[off:off + pr.numel()] = other_full[off:off + pr.numel()]
    assert torch.allclose(reduced, mine0 + other, atol=1e-5), \
        (reduced - mine0 - other).abs().max()


# ---------------------------------------------------------------------------
# End-to-end DistributedTrain loop under gloo world_size=2 (C20/C25): the
# exact class distributed_train.py drives, on synthetic batches — parameter
# broadcast, bucketed all-reduce finalize, rank-0 checkpointing, and
# rank-identical weights at the end.
# ---------------------------------------------------------------------------

def _dt_worker(rank, world, tmpdir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/pg2",
                            rank=rank, world_size=world)
    try:
        from transformer_amd.runtime.train_loop import DistributedTrain

        class Tok:
            vocab_size = 48
            def encode(self, s):
                return [1, 2]
            def decode(self, ids):
                return "x"

        torch.manual_seed(0)
        model = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                            input_vocab_size=50, target_vocab_size=50,
                            rate=0.0, max_position=32)
        tr = DistributedTrain(
            epochs=1, enable_function=False, transformer=model,
            src_tokenizer=Tok(), tgt_tokenizer=Tok(), batch_size=4,
            train_log_dir=None, test_log_dir=None, max_ckpt_keep=2,
            ckpt_path=f"{tmpdir}/ckpt", d_model=16, warmup_steps=10,
            is_rank0=(rank == 0))
        torch.manual_seed(7)  # same batches on both ranks, shard by rank
        steps = []
        for _ in range(3):
            src = torch.randint(1, 50, (4, 6))
            tar = torch.randint(1, 50, (4, 6))
            lo, hi = rank * 2, rank * 2 + 2
            steps.append((src[lo:hi], tar[lo:hi]))
        tr.training_loop(steps, [])
        flat = torch.cat([p.detach().reshape(-1)
                          for p in model.parameters()])
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        if rank == 0:
            q.put((torch.equal(gathered[0], gathered[1]),
                   os.path.isdir(f"{tmpdir}/ckpt")))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_train_loop_gloo(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dt_worker, args=(r, 2, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    same, ckpt_ok = q.get()
    for p in procs:
        p.join(150)
        assert p.exitcode == 0
    assert same, "ranks diverged after DistributedTrain loop"
    assert ckpt_ok, "rank-0 checkpoint directory missing"


@pytest.mark.timeout(600)
def test_distributed_train_entry_point_subprocess(tmp_path, toy_corpus):
    """C25: `python distributed_train.py --num_gpu 2` end-to-end on CPU —
    the self-exec under torchrun, gloo DP=2, shared dataset build, rank-0
    checkpointing."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "distributed_train.py", "--num_gpu", "2",
         "--dataset_path", str(toy_corpus),
         "--src_vocab_file", str(tmp_path / "sv"),
         "--tgt_vocab_file", str(tmp_path / "tv"),
         "--ckpt_path", str(tmp_path / "ckpt"),
         "--epochs", "1", "--num_layers", "1", "--d_model", "32",
         "--dff", "64", "--num_heads", "2", "--batch_size", "8",
         "--sequence_length", "40", "--noenable_function"],
        capture_output=True, text=True, timeout=540)
    assert out.returncode == 0, (out.stderr[-3000:], out.stdout[-1000:])
    assert os.path.isdir(tmp_path / "ckpt"), "no rank-0 checkpoint directory"


def test_ddp_force_mode_self_collectives(monkeypatch):
    """TFMX_DDP_FORCE=1 runs the bucket collectives even at world=1 (the
    single-GPU overlap-evidence mode): buckets must launch from the
    readiness callbacks and finalize must complete."""
    import os
    import torch
    import torch.distributed as dist
    from transformer_amd.models import Transformer
    from transformer_amd.runtime.optimizer import FlatParams
    from transformer_amd.parallel import BucketedDataParallel

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group("gloo", rank=0, world_size=1)
    monkeypatch.setenv("TFMX_DDP_FORCE", "1")
    m = Transformer(num_layers=1, d_model=16, num_heads=2, dff=32,
                    input_vocab_size=30, target_vocab_size=30, rate=0.0,
                    max_position=16)
    flat = FlatParams(m)
    ddp = BucketedDataParallel(flat, bucket_mb=0.01)
    assert ddp.active
    inp = torch.randint(1, 30, (2, 6))
    tar = torch.randint(1, 30, (2, 6))
    logits, _ = m((inp, tar), training=True)
    logits.sum().backward()
    ddp.finalize()
    assert ddp.stats["callback"] + ddp.stats["finalize"] == len(ddp.buckets)
    assert ddp.stats["callback"] > 0
    ddp.detach()
