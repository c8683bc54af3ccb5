"""Checkpoint/resume: flat LE fp32 weights (the parity format)
plus an optional JSON sidecar with training state for exact resume."""
from __future__ import annotations

import json
import os
from typing import Optional


def save_checkpoint(trainer, path: str) -> None:
    """Weights to `path`, training state to `path + '.meta.json'`."""
    trainer.model.save(path)
    meta = {
        "global_step": trainer.global_step,
        "model": trainer.cfg.model,
        "dt": trainer.cfg.dt,
        "grad_reduction": trainer.cfg.grad_reduction,
        "pool": getattr(trainer.cfg, "pool", "trainable"),
        "loss": getattr(trainer.cfg, "loss", "residual"),
        "n_params": int(trainer.model.params.numel()),
    }
    with open(path + ".meta.json", "w") as f:
        json.dump(meta, f, indent=1)


def load_checkpoint(trainer, path: str) -> Optional[dict]:
    """Loads weights; restores global_step from the sidecar if present.
    Returns the metadata dict (or None)."""
    trainer.model.load(path)
    meta_path = path + ".meta.json"
    if not os.path.exists(meta_path):
        return None
    with open(meta_path) as f:
        meta = json.load(f)
    if meta.get("n_params") not in (None, int(trainer.model.params.numel())):
        raise ValueError(
            f"checkpoint {path!r} is for a model with {meta['n_params']} "
            f"params, this model has {trainer.model.params.numel()}")
    # a matching parameter COUNT does not make a matching model: a
    # pool=max or loss=softmax_ce checkpoint would load silently into a
    # trainable/residual config and predict garbage
    for key, cur in (("model", trainer.cfg.model),
                     ("pool", getattr(trainer.cfg, "pool", "trainable")),
                     ("loss", getattr(trainer.cfg, "loss", "residual"))):
        want = meta.get(key)
        if want is not None and want != cur:
            raise ValueError(
                f"checkpoint {path!r} was trained with {key}={want!r}, "
                f"but the running config has {key}={cur!r}")
    trainer.global_step = int(meta.get("global_step", 0))
    return meta
