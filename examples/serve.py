#!/usr/bin/env python
"""HTTP model server for the SE(3)-Transformer (FastAPI + uvicorn).

The reference library ships no serving path; this is the online-inference
counterpart to examples/infer.py (which measures offline throughput):
one resident model per GPU, requests run under `inference_mode` +
autocast-bf16 through the fused HIP kernel path. --checkpoint loads a
`{'model': state_dict, ...}` file (the format examples/denoise.py
writes) or a raw state_dict — the saved architecture must match the
--dim/--depth/... flags given here; otherwise the model serves
random-init weights (useful for latency testing only).

    python examples/serve.py [--port 8000] [--dim 64 ...] [--checkpoint ck.pt]

    POST /predict  {"feats": [[..dim floats..] x n], "coors": [[x,y,z] x n],
                    "return_type": 0}
      -> {"output": [...], "latency_ms": t}
    GET  /health   -> {"status": "ok", "device": "...", "n_params": N}

The app factory (`build_app`) is importable so tests drive it in-process
with fastapi.testclient — no socket needed.

(No `from __future__ import annotations` here: FastAPI must resolve the
endpoint's request-model annotation, and PredictRequest is local to
build_app — a stringified annotation would not resolve.)
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from se3_transformer_amd import SE3Transformer


def build_model(args, device):
    torch.manual_seed(0)
    with torch.device(device):
        model = SE3Transformer(
            dim=args.dim, heads=args.heads, dim_head=args.dim_head,
            depth=args.depth, num_degrees=args.num_degrees,
            num_neighbors=args.num_neighbors, valid_radius=args.valid_radius,
            attend_self=True).eval()
    if args.checkpoint:
        ck = torch.load(args.checkpoint, map_location=device,
                        weights_only=False)
        model.load_state_dict(ck['model'] if 'model' in ck else ck)
    return model


def build_app(args):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    use_cuda = torch.cuda.is_available()
    device = torch.device('cuda' if use_cuda else 'cpu')
    model = build_model(args, device)
    n_params = sum(p.numel() for p in model.parameters())

    class PredictRequest(BaseModel):
        feats: list  # (n, dim) scalar features
        coors: list  # (n, 3)
        return_type: int = 0

    app = FastAPI(title='se3-transformer-amd')

    @app.get('/health')
    def health():
        return {'status': 'ok', 'device': str(device), 'n_params': n_params}

    @app.post('/predict')
    def predict(req: PredictRequest):
        t0 = time.perf_counter()
        feats = torch.tensor(req.feats, dtype=torch.float32, device=device)
        coors = torch.tensor(req.coors, dtype=torch.float32, device=device)
        if feats.dim() != 2 or feats.shape[-1] != args.dim:
            raise HTTPException(422, f'feats must be (n, {args.dim})')
        if coors.dim() != 2 or coors.shape[-1] != 3 \
                or coors.shape[0] != feats.shape[0]:
            raise HTTPException(422, 'coors must be (n, 3) matching feats')
        if feats.shape[0] <= args.num_neighbors:
            raise HTTPException(422,
                                f'need more than {args.num_neighbors} points')
        mask = torch.ones(1, feats.shape[0], dtype=torch.bool, device=device)
        with torch.inference_mode(), \
                torch.autocast(device_type=device.type, dtype=torch.bfloat16,
                               enabled=use_cuda):
            out = model(feats.unsqueeze(0), coors.unsqueeze(0), mask,
                        return_type=req.return_type)
        if use_cuda:
            torch.cuda.synchronize()
        return {'output': out.float().squeeze(0).tolist(),
                'latency_ms': (time.perf_counter() - t0) * 1e3}

    return app


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--host', type=str, default='127.0.0.1')
    p.add_argument('--port', type=int, default=8000)
    p.add_argument('--dim', type=int, default=64)
    p.add_argument('--heads', type=int, default=4)
    p.add_argument('--dim-head', type=int, default=16)
    p.add_argument('--depth', type=int, default=2)
    p.add_argument('--num-degrees', type=int, default=2)
    p.add_argument('--num-neighbors', type=int, default=8)
    p.add_argument('--valid-radius', type=float, default=10.)
    p.add_argument('--checkpoint', type=str, default=None)
    return p.parse_args(argv)


def main():
    import uvicorn
    args = parse_args()
    uvicorn.run(build_app(args), host=args.host, port=args.port)


if __name__ == '__main__':
    main()
