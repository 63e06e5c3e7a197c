"""Data-parallel runtime: bucketed gradient all-reduce over RCCL/xGMI.

Replaces the reference's tf.distribute.MirroredStrategy (reference
distributed_train.py:137-158; implicit collectives SURVEY.md §2.4 X1-X4)
with the MI355X-native design: one process per GPU, torch.distributed with
the "nccl" backend (= RCCL on ROCm), gradients in ONE flat bf16 buffer
(runtime/optimizer.py FlatParams) all-reduced in bucket slices as soon as
each bucket's grads are ready during backward — communication overlaps the
remaining backward compute.  xGMI is 7 point-to-point links ≈153 GB/s per
GPU, so ring all-reduce is per-link bound: default bucket size 25 MiB keeps
several buckets in flight to saturate the links (sweepable via
TFMX_BUCKET_MB).

Gradient averaging semantics follow the reference: the
loss is pre-scaled by 1/global_batch on every replica, so a SUM all-reduce
yields the exact global-batch gradient — no extra division.

Not stock DistributedDataParallel: the bucket manager is ours, built on
post-accumulate-grad hooks + async process-group collectives.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

from ..runtime.optimizer import FlatParams


def init_distributed(backend: str | None = None) -> tuple[int, int, int]:
    """Initialise torch.distributed from torchrun env vars; returns
    (rank, local_rank, world_size).  No-op (0,0,1) outside torchrun."""
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 0, 1
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    # TFMX_BACKEND=gloo lets 2 ranks share ONE GPU (RCCL rejects
    # duplicate devices) — used by tools/dp_parity.py to exercise the
    # real bucket manager + GPU kernels on a single-GPU box.
    backend = os.environ.get("TFMX_BACKEND", backend)
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=300))
    if backend == "nccl":
        torch.cuda.set_device(local)
    return rank, local, world


class BucketedDataParallel:
    """Bucketed, backward-overlapped SUM all-reduce of a FlatParams grad
    buffer.

    Buckets partition the flat buffer at parameter boundaries, assembled in
    REVERSE registration order (the order backward produces grads); each
    bucket launches an async all-reduce the moment its last gradient has
    been accumulated.  `finalize()` launches any stragglers and waits for
    all in-flight work before the optimizer step."""

    def __init__(self, flat: FlatParams, bucket_mb: float | None = None,
                 process_group=None):
        self.flat = flat
        self.pg = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        # TFMX_DDP_FORCE=1: launch the bucket collectives even at world=1
        # (self all-reduce) so a single-GPU rocprof run shows the REAL
        # RCCL launches from the enqueue callbacks overlapping backward.
        self.active = self.world > 1 or os.environ.get("TFMX_DDP_FORCE") == "1"
        if bucket_mb is None:
            bucket_mb = float(os.environ.get("TFMX_BUCKET_MB", "25"))
        cap = int(bucket_mb * 2 ** 20 / flat.flat_g.element_size())

        # bucket assembly: reverse order, contiguous flat slices
        self.buckets = []  # list of dicts
        params = list(zip(flat.params, flat.offsets))
        cur, cur_n = [], 0
        for p, off in reversed(params):
            cur.append((p, off))
            cur_n += p.numel()
            if cur_n >= cap:
                self._push_bucket(cur)
                cur, cur_n = [], 0
        if cur:
            self._push_bucket(cur)

        self._param_bucket = {}
        for bi, b in enumerate(self.buckets):
            for p in b["params"]:
                self._param_bucket[id(p)] = bi
        # readiness: GPU flat path fires the ops-layer callback (weights are
        # hidden from autograd there); CPU/plain path fires post-accumulate
        # hooks.  A param triggers exactly once per step on exactly one path.
        self._hooks = [
            p.register_post_accumulate_grad_hook(self._make_hook(p))
            for p in flat.params
        ]
        from ..ops import functional as _F
        _F.set_grad_ready_callback(flat.params, self._on_param_ready)
        # launch-origin accounting (overlap evidence: callback launches
        # happen DURING backward, finalize launches are stragglers)
        self.stats = {"callback": 0, "finalize": 0}
        self._reset_step()

    def detach(self):
        """Unregister hooks/callbacks (a discarded instance must not keep
        firing into dead bucket state)."""
        from ..ops import functional as _F
        _F.set_grad_ready_callback(self.flat.params, None)
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def _on_param_ready(self, p):
        if not self.active:
            return
        bi = self._param_bucket.get(id(p))
        if bi is None:
            return
        b = self.buckets[bi]
        b["pending"] -= 1
        if b["pending"] == 0:
            self.stats["callback"] += 1
            self._launch(b)

    def _push_bucket(self, plist):
        los = [off for _, off in plist]
        his = [off + p.numel() for p, off in plist]
        self.buckets.append({
            "params": [p for p, _ in plist],
            "lo": min(los),
            "hi": max(his),
            "pending": len(plist),
            "work": None,
        })

    def _reset_step(self):
        for b in self.buckets:
            b["pending"] = len(b["params"])
            b["work"] = None

    def _make_hook(self, p):
        bi = self._param_bucket[id(p)]

        def hook(_param):
            if not self.active:
                return
            b = self.buckets[bi]
            b["pending"] -= 1
            if b["pending"] == 0:
                self.stats["callback"] += 1
                self._launch(b)
        return hook

    def _launch(self, b):
        sl = self.flat.flat_g[b["lo"]:b["hi"]]
        b["work"] = dist.all_reduce(sl, op=dist.ReduceOp.SUM, group=self.pg,
                                    async_op=True)

    def broadcast_parameters(self):
        """X1: initial parameter (+nothing else) broadcast from rank 0."""
        if self.world > 1:
            dist.broadcast(self.flat.flat_w, src=0, group=self.pg)

    def finalize(self):
        """Call after loss.backward(), before optimizer.step()."""
        if not self.active:
            return
        for b in self.buckets:
            if b["work"] is None and b["pending"] > 0:
                # params that produced no grad this step (robustness)
                self.stats["finalize"] += 1
                self._launch(b)
        for b in self.buckets:
            if b["work"] is not None:
                b["work"].wait()
        self._reset_step()
