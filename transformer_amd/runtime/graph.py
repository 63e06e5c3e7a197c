"""HIP-graph captured training step (the reference's
`--enable_function` tf.function toggle maps to capturing the whole
fwd+bwd+Adam step into one hipGraph and replaying it).

The step is captured once with static input/output buffers; each call
copies the batch into the static buffers and replays the graph — ~1100
kernel launches collapse into one hipGraphLaunch, removing host launch
overhead and CPU jitter (which matters most for multi-rank lockstep).

Replay-variant state lives on device:
  - the Noam schedule / Adam bias corrections read a device step tensor
    advanced in-graph (ops adam_coefs kernel, optimizer.step_captured);
  - dropout seeds combine that step tensor with a per-call-site salt
    (ops.functional.set_graph_rng), so masks differ every replay.

DP>1 is not captured (RCCL collectives inside graph capture are not
supported on this stack) — callers fall back to the eager step.
"""

from __future__ import annotations

import torch


class CapturedTrainStep:
    """Captures fwd+loss+bwd+opt for fixed-shape (src, tar) batches."""

    def __init__(self, model, optimizer, loss_fn, src_shape, tar_shape,
                 device, warmup_iters: int = 2):
        from .. import ops
        self.model = model
        self.opt = optimizer
        self.loss_fn = loss_fn
        self.src = torch.zeros(src_shape, dtype=torch.int64, device=device)
        self.tar = torch.zeros(tar_shape, dtype=torch.int64, device=device)

        step_t, _ = optimizer.graph_state()

        def run():
            tar_inp = self.tar[:, :-1].contiguous()
            tar_real = self.tar[:, 1:].contiguous()
            logits, _ = model((self.src, tar_inp), training=True)
            loss = loss_fn(tar_real, logits)
            optimizer.zero_grad()
            loss.backward()
            optimizer.step_captured()
            return loss, logits, tar_real

        # warmup on a side stream (cuDNN-style), then capture.  The warmup
        # iterations execute REAL fwd+bwd+Adam steps on the zero-filled
        # static buffers — harmless at first capture (m=v=0) but a
        # mid-training recapture (batch exceeding the captured shape) would
        # apply zero-gradient Adam updates to live state.  Snapshot the fp32
        # master + m/v around the warmup so recapture is state-neutral.
        ops.functional.set_graph_rng(step_t)
        snap = (optimizer.master.clone(), optimizer.m.clone(),
                optimizer.v.clone())
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(warmup_iters):
                    run()
            torch.cuda.current_stream().wait_stream(s)
            optimizer.master.copy_(snap[0])
            optimizer.m.copy_(snap[1])
            optimizer.v.copy_(snap[2])
            optimizer.flat.flat_w.copy_(
                optimizer.master.to(optimizer.flat.flat_w.dtype))
            del snap

            # make the weight-derived caches (transposed-weight buffers)
            # stale so the capture RECORDS their batched refresh — replays
            # then refresh them after each in-graph Adam step.  The
            # descriptor table must exist BEFORE capture (H2D copy).
            ops.functional.prepare_weight_caches()
            ops.functional.bump_weight_version()
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.loss, self.logits, self.tar_real = run()
        finally:
            ops.functional.set_graph_rng(None)
        # warmup advanced the device step; resync to the host counter
        step_t.fill_(optimizer.step_count)

    def fits(self, src, tar) -> bool:
        return (src.shape[0] <= self.src.shape[0]
                and src.shape[1] <= self.src.shape[1]
                and tar.shape[0] <= self.tar.shape[0]
                and tar.shape[1] <= self.tar.shape[1])

    def __call__(self, src: torch.Tensor, tar: torch.Tensor) -> torch.Tensor:
        """Replay on a batch; smaller batches are padded with pad id 0
        (zero rows are fully masked by kv_pad/the CE pad mask)."""
        if src.shape == self.src.shape and tar.shape == self.tar.shape:
            self.src.copy_(src, non_blocking=True)
            self.tar.copy_(tar, non_blocking=True)
        else:
            self.src.zero_()
            self.tar.zero_()
            self.src[:src.shape[0], :src.shape[1]].copy_(src,
                                                         non_blocking=True)
            self.tar[:tar.shape[0], :tar.shape[1]].copy_(tar,
                                                         non_blocking=True)
        self.graph.replay()
        self.opt.step_count += 1
        from ..ops import functional as _F
        _F.bump_weight_version()  # replay updated weights in-graph
        return self.loss
