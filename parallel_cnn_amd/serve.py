"""Inference serving — a minimal production endpoint over the framework's
INFER path (device argmax kernels on GPU, native CPU ops otherwise).

Run:   python -m parallel_cnn_amd.serve --ckpt weights.bin --port 8000
API:
  GET  /health            -> {"status": "ok", ...}
  GET  /info              -> model/config metadata
  POST /predict           -> {"labels": [...]}  (optionally {"probs": ...})
       body: {"images": [[... H*W*C floats in [0,1] ...], ...],
              "return_probs": false}
"""
from __future__ import annotations

import argparse
from typing import List, Optional

import torch
from fastapi import FastAPI, HTTPException
from pydantic import BaseModel

from .config import TrainConfig
from .engine.deep import DeepTrainer
from .engine.trainer import Trainer


class PredictRequest(BaseModel):
    images: List[List[float]]
    return_probs: bool = False


def create_app(cfg: Optional[TrainConfig] = None,
               ckpt: Optional[str] = None) -> FastAPI:
    cfg = cfg or TrainConfig()
    trainer = (DeepTrainer(cfg) if cfg.model == "deepcnn" else Trainer(cfg))
    if ckpt:
        trainer.model.load(ckpt)
        if hasattr(trainer, "invalidate_weight_cache"):
            trainer.invalidate_weight_cache()
    spec_pixels = (32 * 32 * 3 if cfg.model == "deepcnn" else 784)

    app = FastAPI(title="parallel_cnn_amd inference")

    @app.get("/health")
    def health():
        return {"status": "ok", "model": cfg.model,
                "backend": trainer.backend,
                "device": str(trainer.device)}

    @app.get("/info")
    def info():
        return {"model": cfg.model, "backend": trainer.backend,
                "input_pixels": spec_pixels,
                "n_params": int(trainer.model.params.numel()),
                "checkpoint": ckpt, "act_dtype": cfg.act_dtype}

    @app.post("/predict")
    def predict(req: PredictRequest):
        if not req.images:
            raise HTTPException(400, "no images")
        x = torch.tensor(req.images, dtype=torch.float32)
        if x.dim() != 2 or x.shape[1] != spec_pixels:
            raise HTTPException(
                400, f"each image must be a flat list of {spec_pixels} "
                     f"floats, got shape {tuple(x.shape)}")
        if cfg.model == "deepcnn":
            # chunked by the device workspace max batch — a request larger
            # than cfg.batch_size must never reach the raw forward
            y = trainer.forward_logits(x)
            labels = y.argmax(1)
        else:
            labels = trainer.classify(x)
            y = None
            if req.return_probs:
                from .ops import torch_ref
                _, _, y = torch_ref.forward(x, trainer.model.params.cpu(),
                                            cfg.pool, cfg.loss)
        out = {"labels": [int(v) for v in labels]}
        if req.return_probs and y is not None:
            out["probs"] = y.tolist()
        return out

    return app


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--model", default="lenet5",
                   choices=["lenet5", "deepcnn"])
    p.add_argument("--ckpt", default=None)
    p.add_argument("--device", default="auto")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args(argv)
    import uvicorn
    cfg = TrainConfig(model=args.model, device=args.device, log_interval=0)
    uvicorn.run(create_app(cfg, args.ckpt), host=args.host, port=args.port)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
