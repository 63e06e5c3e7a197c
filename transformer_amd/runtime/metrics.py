"""Streaming metrics (C14 — reference train.py:70-73: Mean loss + accuracy,
reset per epoch).  In DP, `sync()` all-reduces the counters so `.result()`
matches the reference's mirrored-metric reads."""

from __future__ import annotations

import torch
import torch.distributed as dist


class Mean:
    def __init__(self, name: str = "mean"):
        self.name = name
        self.total = 0.0
        self.count = 0.0

    def update(self, value, weight: float = 1.0):
        self.total += float(value) * weight
        self.count += weight

    def result(self) -> float:
        return self.total / self.count if self.count else 0.0

    def reset(self):
        self.total = 0.0
        self.count = 0.0

    def sync(self):
        if dist.is_available() and dist.is_initialized():
            # RCCL ("nccl") only reduces device tensors
            dev = (torch.device("cuda", torch.cuda.current_device())
                   if dist.get_backend() == "nccl" and torch.cuda.is_available()
                   else torch.device("cpu"))
            t = torch.tensor([self.total, self.count], dtype=torch.float64,
                             device=dev)
            dist.all_reduce(t)
            self.total, self.count = t[0].item(), t[1].item()
