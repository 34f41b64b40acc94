"""Minimal ReID feature-extraction service.

The reference is a training/evaluation simulator with no serving story; this
module closes the deployment loop: load a trained client checkpoint, expose
the eval-mode embedding forward over HTTP.  On an MI355X the forward runs the
same fused eval path as validation (bf16 autocast, fused eval-BN, hipGraphs
are NOT used here — request batches vary).

    python -m flreid_amd.serve --method fedstil --model resnet50 \
        --ckpt ckpts/exp/client-0/fedstil_model.ckpt --port 8100

Endpoints:
    GET  /health            -> {"status": "ok", "device": ..., "model": ...}
    POST /embed             -> {"features": [[...], ...]}
        body: {"images": [[C][H][W] float lists, ...]}  (normalized crops)

Kept dependency-light: fastapi + uvicorn (both in the image); the app object
is importable for in-process testing (tests/test_serve.py uses Starlette's
TestClient).
"""

from __future__ import annotations

import argparse
from typing import Optional

import torch

from flreid_amd.runtime.precision import autocast


def build_model(method: str, model_opts: dict, ckpt: Optional[str],
                device: str):
    from flreid_amd.runtime.builder import parser_model

    model = parser_model(method, model_opts)
    if ckpt:
        state = torch.load(ckpt, map_location="cpu", weights_only=False)
        model.update_model(state)
    model.to(device)
    model.eval()
    return model


def create_app(model, device: str):
    from fastapi import Body, FastAPI, HTTPException

    app = FastAPI(title="flreid_amd ReID embedding service")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(device),
                "model": type(model).__name__}

    @app.post("/embed")
    def embed(payload: dict = Body(...)):
        images = payload.get("images")
        if images is None:
            raise HTTPException(400, "missing 'images'")
        try:
            x = torch.tensor(images, dtype=torch.float32)
        except (TypeError, ValueError) as e:
            raise HTTPException(400, f"bad image payload: {e}")
        if x.dim() != 4:
            raise HTTPException(400, f"expected [B, C, H, W], got {tuple(x.shape)}")
        x = x.to(device)
        with torch.no_grad(), autocast(device):
            feat = model(x)
        if isinstance(feat, tuple):          # train-mode style output guard
            feat = feat[1]
        feat = torch.nn.functional.normalize(feat.float(), dim=1)
        return {"features": feat.cpu().tolist()}

    return app


def main():
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--method", default="fedstil")
    p.add_argument("--model", default="resnet50")
    p.add_argument("--num-classes", type=int, default=8000)
    p.add_argument("--ckpt", default=None)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8100)
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    model_opts = {"name": args.model, "num_classes": args.num_classes,
                  "last_stride": 1, "neck": "bnneck",
                  "fine_tuning": ["base.layer4", "classifier"],
                  "atten_default": 0.9, "lambda_l1": 1e-4, "lambda_k": 2000}
    model = build_model(args.method, model_opts, args.ckpt, device)
    uvicorn.run(create_app(model, device), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
