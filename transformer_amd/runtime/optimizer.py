"""Noam-scheduled Adam (C11+C12, K15) with flat-buffer fused GPU path.

MI355X design: all model parameters live as views into ONE contiguous bf16
buffer, gradients accumulate into one contiguous bf16 buffer, and the Adam
update (β1=0.9, β2=0.98, ε=1e-9 — reference train.py:65-66) runs as a single
fused multi-element HIP kernel over fp32 master weights + m/v state
(SURVEY.md §2.4 design (b): bf16 wire gradients, fp32 Adam master).  The
flat gradient buffer is also what the DP runtime all-reduces in bucket
slices (parallel/ddp.py) — fewer, larger collectives for xGMI.

CPU path: per-parameter fp32 Adam via ops/reference.py (the kernel's
numerics oracle).
"""

from __future__ import annotations

import torch

from ..ops import ext, has_ext, reference as R
from .schedule import NoamSchedule

_ALIGN = 64  # elements; keeps every param slice 128-byte aligned for kernels


class FlatParams:
    """Rebind every parameter of `model` as a view into one flat buffer and
    hand every backward kernel a gradient DESTINATION view into a flat
    gradient buffer (`p._flat_grad`).

    Must be called AFTER model.to(device, dtype).  The custom autograd
    Functions (ops/functional.py) write weight gradients straight into these
    views and return them, so autograd binds p.grad to the flat slice with
    ZERO per-parameter accumulate kernels per step (~200 launches saved on
    transformer-base).  Each parameter gets exactly one gradient
    contribution per backward (true for this model family); `check()`
    asserts post-backward that .grad aliases the flat buffer."""

    def __init__(self, model: torch.nn.Module):
        params = [p for p in model.parameters() if p.requires_grad]
        if not params:
            raise ValueError("model has no trainable parameters")
        dev, dt = params[0].device, params[0].dtype
        offs, total = [], 0
        for p in params:
            offs.append(total)
            total += (p.numel() + _ALIGN - 1) // _ALIGN * _ALIGN
        self.flat_w = torch.zeros(total, device=dev, dtype=dt)
        self.flat_g = torch.zeros(total, device=dev, dtype=dt)
        self.params = params
        self.offsets = offs
        self.numel = total
        # GPU: backward kernels write into p._flat_grad views directly
        # (zero accumulate kernels).  CPU (plain torch autograd): pre-assign
        # .grad views so AccumulateGrad lands gradients in the flat buffer.
        self.direct_write = dev.type == "cuda"
        for p, off in zip(params, offs):
            n = p.numel()
            self.flat_w[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_w[off:off + n].view(p.shape)
            if self.direct_write:
                p._flat_grad = self.flat_g[off:off + n].view(p.shape)
            else:
                p.grad = self.flat_g[off:off + n].view(p.shape)

    def check(self):
        if self.direct_write:
            # GPU path: weights are hidden from autograd; kernels write into
            # p._flat_grad views directly, so there is no .grad to verify.
            for p in self.params:
                fg = getattr(p, "_flat_grad", None)
                if fg is None:
                    raise RuntimeError("parameter lost its _flat_grad view")
            return
        base = self.flat_g.data_ptr()
        end = base + self.flat_g.numel() * self.flat_g.element_size()
        for p in self.params:
            if p.grad is None or not (base <= p.grad.data_ptr() < end):
                raise RuntimeError(
                    "flat-gradient aliasing broken: autograd rebound .grad "
                    "(check torch accumulate_grad in-place semantics)")

    def zero_grad(self):
        self.flat_g.zero_()


class NoamAdam:
    """Adam with the Noam schedule computed host-side per step (K15).

    GPU: one fused HIP kernel over the whole flat parameter set.
    CPU: reference per-tensor fp32 Adam."""

    def __init__(self, model: torch.nn.Module, d_model: int,
                 warmup_steps: int = 60000, betas=(0.9, 0.98), eps: float = 1e-9,
                 use_flat: bool | None = None):
        self.schedule = NoamSchedule(d_model, warmup_steps)
        self.betas, self.eps = betas, eps
        self.step_count = 0
        p0 = next(model.parameters())
        self.is_cuda = p0.is_cuda
        if use_flat is None:
            use_flat = self.is_cuda
        self.flat: FlatParams | None = None
        if use_flat:
            self.flat = FlatParams(model)
            self.master = self.flat.flat_w.float()
            self.m = torch.zeros_like(self.master)
            self.v = torch.zeros_like(self.master)
        else:
            self.params = [p for p in model.parameters() if p.requires_grad]
            self.state = [
                {"m": torch.zeros_like(p, dtype=torch.float32),
                 "v": torch.zeros_like(p, dtype=torch.float32),
                 "master": p.detach().float().clone()}
                for p in self.params
            ]

    @property
    def lr(self) -> float:
        return self.schedule(max(self.step_count, 1))

    # ---- HIP-graph capture support (Q12) ---------------------------------
    def graph_state(self):
        """Device-side (step, coefs) tensors for captured steps.  The step
        tensor doubles as the dropout seed epoch (ops.set_graph_rng)."""
        assert self.flat is not None and self.is_cuda
        if not hasattr(self, "_step_t"):
            dev = self.flat.flat_w.device
            self._step_t = torch.tensor([self.step_count], dtype=torch.int64,
                                        device=dev)
            self._coefs = torch.zeros(3, dtype=torch.float32, device=dev)
        return self._step_t, self._coefs

    @torch.no_grad()
    def step_captured(self):
        """Graph-capturable optimizer step: the Noam lr and Adam bias
        corrections are derived ON DEVICE from a step tensor advanced
        in-graph, so a replayed graph keeps the schedule moving.  The host
        `step_count` must be advanced by the replay wrapper."""
        from ..ops import ext as _ext_fn
        st, cf = self.graph_state()
        b1, b2 = self.betas
        self.flat.check()
        _ext_fn().adam_fused_dev(self.master, self.m, self.v,
                                  self.flat.flat_g, self.flat.flat_w, st, cf,
                                  self.schedule.d_model,
                                  self.schedule.warmup_steps, b1, b2,
                                  self.eps)

    def zero_grad(self, set_to_none: bool = False):
        if self.flat is not None:
            self.flat.zero_grad()
        else:
            for p in self.params:
                if p.grad is not None:
                    p.grad.zero_()

    @torch.no_grad()
    def step(self):
        from ..ops import functional as _F
        _F.bump_weight_version()  # invalidate weight-derived caches
        self.step_count += 1
        lr = self.schedule(self.step_count)
        b1, b2 = self.betas
        if self.flat is not None:
            self.flat.check()
            if self.is_cuda:
                ext().adam_fused(self.master, self.m, self.v,
                                 self.flat.flat_g, self.flat.flat_w,
                                 lr, b1, b2, self.eps, self.step_count)
            else:
                g = self.flat.flat_g.float()
                R.adam_step_reference(self.master, g, self.m, self.v,
                                      self.step_count, lr, b1, b2, self.eps)
                self.flat.flat_w.copy_(self.master.to(self.flat.flat_w.dtype))
        else:
            for p, st in zip(self.params, self.state):
                if p.grad is None:
                    continue
                R.adam_step_reference(st["master"], p.grad.float(), st["m"],
                                      st["v"], self.step_count, lr, b1, b2,
                                      self.eps)
                p.data.copy_(st["master"].to(p.dtype))

    # -- checkpointing (C16) ------------------------------------------------
    def state_dict(self):
        d = {"step": self.step_count,
             "d_model": self.schedule.d_model,
             "warmup_steps": self.schedule.warmup_steps}
        if self.flat is not None:
            d.update(master=self.master, m=self.m, v=self.v)
        else:
            d.update(state=[{k: v for k, v in st.items()} for st in self.state])
        return d

    def load_state_dict(self, d):
        self.step_count = d["step"]
        if hasattr(self, "_step_t"):  # keep captured-graph schedule in sync
            self._step_t.fill_(self.step_count)
        if self.flat is not None:
            self.master.copy_(d["master"].to(self.master.device))
            self.m.copy_(d["m"].to(self.m.device))
            self.v.copy_(d["v"].to(self.v.device))
            self.flat.flat_w.copy_(self.master.to(self.flat.flat_w.dtype))
        else:
            for st, sd in zip(self.state, d["state"]):
                for k in ("m", "v", "master"):
                    st[k].copy_(sd[k].to(st[k].device))
            for p, st in zip(self.params, self.state):
                p.data.copy_(st["master"].to(p.dtype))
