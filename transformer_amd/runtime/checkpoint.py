"""Checkpoint / resume (C16 — reference train.py:77-80, 159-164, 208-211).

Same contract as tf.train.Checkpoint + CheckpointManager: model weights +
optimizer state (fp32 master, m, v, step) + epoch, rolling window of
`max_to_keep` (default 5, --max_ckpt_keep), latest-checkpoint discovery and
restore-if-present (`expect_partial` analog: missing keys tolerated with a
warning).  Rank 0 writes in DP.
"""

from __future__ import annotations

import json
import os

import torch


class CheckpointManager:
    def __init__(self, model, optimizer, directory: str, max_to_keep: int = 5):
        self.model = model
        self.optimizer = optimizer
        self.directory = directory
        self.max_to_keep = max_to_keep
        os.makedirs(directory, exist_ok=True)

    # -- discovery ----------------------------------------------------------
    @property
    def _index_path(self):
        return os.path.join(self.directory, "checkpoint.json")

    def _read_index(self) -> list[str]:
        try:
            with open(self._index_path) as f:
                return json.load(f)["checkpoints"]
        except (FileNotFoundError, json.JSONDecodeError, KeyError):
            return []

    @property
    def latest_checkpoint(self) -> str | None:
        ckpts = self._read_index()
        return os.path.join(self.directory, ckpts[-1]) if ckpts else None

    # -- save / restore -----------------------------------------------------
    def save(self, step: int, epoch: int | None = None) -> str:
        name = f"ckpt-{step}.pt"
        path = os.path.join(self.directory, name)
        tmp = path + ".tmp"
        torch.save({
            "model": self.model.state_dict(),
            "optimizer": self.optimizer.state_dict() if self.optimizer else None,
            "step": step,
            "epoch": epoch,
        }, tmp)
        os.replace(tmp, path)
        ckpts = [c for c in self._read_index() if c != name] + [name]
        while len(ckpts) > self.max_to_keep:
            old = ckpts.pop(0)
            try:
                os.remove(os.path.join(self.directory, old))
            except FileNotFoundError:
                pass
        # atomic index write: a crash mid-write must not orphan the window
        itmp = self._index_path + ".tmp"
        with open(itmp, "w") as f:
            json.dump({"checkpoints": ckpts}, f)
            f.flush()
            os.fsync(f.fileno())
        os.replace(itmp, self._index_path)
        return path

    def restore(self, path: str | None = None) -> dict | None:
        """Restore latest (or `path`); returns the metadata dict or None if
        nothing to restore (reference load_ckpt semantics, train.py:159-164)."""
        path = path or self.latest_checkpoint
        if path is None or not os.path.exists(path):
            return None
        blob = torch.load(path, map_location="cpu", weights_only=False)
        # expect_partial semantics (reference train.py:163): tolerate
        # missing/unexpected keys AND shape-mismatched entries (strict=False
        # alone still raises on size mismatch)
        state = blob["model"]
        current = self.model.state_dict()
        skipped = [k for k, v in state.items()
                   if k in current and current[k].shape != v.shape]
        for k in skipped:
            state.pop(k)
        missing, unexpected = self.model.load_state_dict(state, strict=False)
        if missing or unexpected or skipped:
            print(f"[checkpoint] partial restore: missing={missing} "
                  f"unexpected={unexpected} shape_mismatch={skipped}")
        if self.optimizer is not None and blob.get("optimizer") is not None:
            self.optimizer.load_state_dict(blob["optimizer"])
        return {"step": blob.get("step", 0), "epoch": blob.get("epoch")}
