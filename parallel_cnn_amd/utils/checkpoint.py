"""Checkpoint/resume: flat LE fp32 weights (the parity format)
plus a JSON sidecar with training state for EXACT resume (global step,
epoch cursor, host/device RNG states — so train(2N epochs) ==
train(N) -> save -> load -> train(N))."""
from __future__ import annotations

import base64
import json
import os
from typing import Optional

import torch


def _rng_capture() -> dict:
    state = {"torch_cpu": base64.b64encode(
        torch.get_rng_state().numpy().tobytes()).decode()}
    if torch.cuda.is_available():
        state["torch_cuda"] = base64.b64encode(
            torch.cuda.get_rng_state().numpy().tobytes()).decode()
    return state


def _rng_restore(state: dict) -> None:
    import numpy as np
    cpu = state.get("torch_cpu")
    if cpu:
        torch.set_rng_state(torch.from_numpy(np.frombuffer(
            base64.b64decode(cpu), dtype=np.uint8).copy()))
    cuda = state.get("torch_cuda")
    if cuda and torch.cuda.is_available():
        torch.cuda.set_rng_state(torch.from_numpy(np.frombuffer(
            base64.b64decode(cuda), dtype=np.uint8).copy()))


def save_checkpoint(trainer, path: str) -> None:
    """Weights to `path`, training state to `path + '.meta.json'`."""
    trainer.model.save(path)
    meta = {
        "global_step": trainer.global_step,
        "epoch": getattr(trainer, "epoch", 0),
        "model": trainer.cfg.model,
        "dt": trainer.cfg.dt,
        "grad_reduction": trainer.cfg.grad_reduction,
        "pool": getattr(trainer.cfg, "pool", "trainable"),
        "loss": getattr(trainer.cfg, "loss", "residual"),
        "n_params": int(trainer.model.params.numel()),
        "rng": _rng_capture(),
    }
    with open(path + ".meta.json", "w") as f:
        json.dump(meta, f, indent=1)


def load_checkpoint(trainer, path: str) -> Optional[dict]:
    """Loads weights; restores global_step/epoch/RNG from the sidecar if
    present.  Returns the metadata dict (or None)."""
    trainer.model.load(path)
    if hasattr(trainer, "invalidate_weight_cache"):
        trainer.invalidate_weight_cache()
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
    if hasattr(trainer, "epoch"):
        trainer.epoch = int(meta.get("epoch", 0))
    if meta.get("rng"):
        _rng_restore(meta["rng"])
    return meta
