"""Model export (C24 — reference train.py:246: tf.saved_model.save(...,
'model')).  SavedModel-equivalent: a self-describing directory with the
architecture config + weights, reloadable for inference without the
training script's flag state."""

from __future__ import annotations

import json
import os

import torch


def export_model(model, directory: str, config: dict):
    os.makedirs(directory, exist_ok=True)
    torch.save({k: v.cpu() for k, v in model.state_dict().items()},
               os.path.join(directory, "weights.pt"))
    with open(os.path.join(directory, "config.json"), "w") as f:
        json.dump(config, f, indent=2)


def load_exported(directory: str, device="cpu", dtype=None):
    from ..models import Transformer

    with open(os.path.join(directory, "config.json")) as f:
        config = json.load(f)
    model = Transformer(
        num_layers=config["num_layers"], d_model=config["d_model"],
        num_heads=config["num_heads"], dff=config["dff"],
        input_vocab_size=config["input_vocab_size"],
        target_vocab_size=config["target_vocab_size"],
        rate=config.get("dropout_rate", 0.1),
        max_position=config.get("max_position", 4096))
    sd = torch.load(os.path.join(directory, "weights.pt"), weights_only=True)
    model.load_state_dict(sd)
    model = model.to(device)
    if dtype is not None:
        model = model.to(dtype)
    return model, config
