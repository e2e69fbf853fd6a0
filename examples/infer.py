#!/usr/bin/env python
"""Batched inference / serving throughput for the SE(3)-Transformer.

The reference ships only a training script (denoise.py); this is the
serving-side counterpart for the MI355X framework: `inference_mode`,
autocast-bf16 through the fused kernel path, and (single-stream) hipGraph
capture of the whole forward so steady-state latency is replay-bound.

    python examples/infer.py [--points 1024] [--batch 4] [--iters 20]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from se3_transformer_amd import SE3Transformer


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--points', type=int, default=1024)
    p.add_argument('--batch', type=int, default=1)
    p.add_argument('--dim', type=int, default=512)
    p.add_argument('--heads', type=int, default=8)
    p.add_argument('--dim-head', type=int, default=64)
    p.add_argument('--depth', type=int, default=6)
    p.add_argument('--num-degrees', type=int, default=4)
    p.add_argument('--num-neighbors', type=int, default=8)
    p.add_argument('--iters', type=int, default=20)
    p.add_argument('--warmup', type=int, default=3)
    p.add_argument('--no-graph', action='store_true')
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device('cuda' if use_cuda else 'cpu')
    torch.manual_seed(0)
    with torch.device(device):
        model = SE3Transformer(
            dim=args.dim, heads=args.heads, dim_head=args.dim_head,
            depth=args.depth, num_degrees=args.num_degrees,
            num_neighbors=args.num_neighbors, valid_radius=10.,
            attend_self=True).eval()

    feats = torch.randn(args.batch, args.points, args.dim, device=device)
    coors = torch.randn(args.batch, args.points, 3, device=device) * 2.0
    mask = torch.ones(args.batch, args.points, dtype=torch.bool, device=device)

    def forward():
        with torch.autocast(device_type=device.type, dtype=torch.bfloat16,
                            enabled=use_cuda):
            return model(feats, coors, mask, return_type=0)

    run = forward
    with torch.inference_mode():
        out = forward()                     # allocator + cache warmup
        if use_cuda and not args.no_graph:
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                out = forward()
            run = g.replay
        for _ in range(args.warmup):
            run()
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run()
        if use_cuda:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters

    print(json.dumps({
        'what': 'inference throughput (forward only)',
        'samples_per_sec': args.batch / dt,
        'ms_per_batch': dt * 1000.,
        'batch': args.batch, 'points': args.points, 'dim': args.dim,
        'depth': args.depth, 'num_degrees': args.num_degrees,
        'num_neighbors': args.num_neighbors,
        'graph': use_cuda and not args.no_graph,
        'dtype': 'bf16' if use_cuda else 'fp32',
        'out_shape': list(out.shape),
    }))


if __name__ == '__main__':
    main()
