"""Config/flag system (the reference hardcodes every constant — SURVEY.md §5.6).

All reference hyperparameters are exposed with their reference defaults;
everything is overridable from YAML or CLI flags.
"""
from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass, field
from typing import Optional

from .ops.shapes import REF_DT, REF_THRESHOLD


@dataclass
class TrainConfig:
    # model / numerics
    model: str = "lenet5"
    deep_channels: str = "32,64,64"  # DeepCNN stage widths (x16 each)
    act_dtype: str = "bf16"          # activation storage on GPU: bf16 | fp16 | fp32
    pool: str = "trainable"          # trainable (reference) | max
    loss: str = "residual"           # residual (reference) | softmax_ce
    seed: int = 0

    # optimizer (reference semantics: p += dt * grad)
    dt: float = REF_DT
    grad_reduction: str = "mean"     # mean | sum over the global batch
    grad_accum: int = 1              # micro-batches per optimizer step
    threshold: float = REF_THRESHOLD  # early-stop when mean err-norm < this

    # schedule
    epochs: int = 1                  # reference trains exactly one epoch
    batch_size: int = 64             # per-rank batch

    # data
    data: str = "synthetic"          # synthetic | mnist
    data_dir: str = "data"
    train_count: int = 60000
    test_count: int = 10000

    # execution
    backend: str = "auto"            # auto | hip | cpu | torchref
    device: str = "auto"             # auto | cuda | cpu
    wgrad_chunk: int = 0             # conv1 wgrad slices/channel (0 = auto)
    overlap_comm: bool = False       # two-bucket DP: all-reduce the ready
                                     # bucket while the other wgrad runs
    fuse_wgrad: bool = False         # experimental: conv/pool wgrad inside
                                     # the fwdbwd kernel (measured slower)
    deep_implicit: bool = True       # DeepCNN: implicit-im2col GEMMs (no
                                     # materialized cols for Cin%8==0
                                     # stages, dgrad-as-conv w/ fused
                                     # sigmoid-bwd); False = round-1
                                     # materialized path

    # io / observability
    log_interval: int = 100          # steps between loss readouts
    ckpt_save: Optional[str] = None
    ckpt_load: Optional[str] = None

    def resolved_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch
        return "cuda" if torch.cuda.is_available() else "cpu"

    def resolved_backend(self) -> str:
        if self.backend != "auto":
            return self.backend
        return "hip" if self.resolved_device() == "cuda" else "cpu"

    @classmethod
    def from_yaml(cls, path: str) -> "TrainConfig":
        import yaml
        with open(path) as f:
            raw = yaml.safe_load(f) or {}
        known = {f.name for f in dataclasses.fields(cls)}
        unknown = set(raw) - known
        if unknown:
            raise ValueError(f"unknown config keys: {sorted(unknown)}")
        return cls(**raw)

    @classmethod
    def add_cli_args(cls, p: argparse.ArgumentParser) -> None:
        p.add_argument("--config", type=str, default=None,
                       help="YAML config file (CLI flags override it)")
        for f in dataclasses.fields(cls):
            name = "--" + f.name.replace("_", "-")
            if f.type == "bool":
                p.add_argument(name, type=lambda s: s.lower() in ("1", "true"),
                               default=None)
            else:
                p.add_argument(name, type=str, default=None)

    @classmethod
    def from_args(cls, args: argparse.Namespace) -> "TrainConfig":
        cfg = cls.from_yaml(args.config) if getattr(args, "config", None) \
            else cls()
        for f in dataclasses.fields(cls):
            v = getattr(args, f.name, None)
            if v is None:
                continue
            if isinstance(v, bool):
                # bool flags are already parsed by the CLI lambda; coercing
                # through str() would turn False into the truthy "False"
                setattr(cfg, f.name, v)
                continue
            typ = {int: int, float: float, str: str}.get(
                type(getattr(cfg, f.name)), str)
            setattr(cfg, f.name, typ(v))
        return cfg


@dataclass
class BenchConfig(TrainConfig):
    steps: int = 200
    warmup: int = 20
    data: str = "synthetic"
    field_note: str = field(default="", repr=False)
