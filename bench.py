#!/usr/bin/env python
"""Flagship training benchmark: SE(3)-Transformer denoising-style step.

Metric (BASELINE.json): train samples/sec (fwd+bwd+optimizer), 1024-point
clouds, dim=512, heads=8, dim_head=64, depth=6, num_degrees=4, bf16 compute,
at 1/2/4/8 GPUs (weak scaling: per-GPU batch fixed). Synthetic point clouds,
random-init weights (no network access for datasets).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W            # single GPU/CPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

from se3_transformer_amd import SE3Transformer
from se3_transformer_amd.parallel import DistributedDataParallelSE3, setup_distributed


# BASELINE.json named configs beyond the headline (selected with --preset):
#   qm9        — 29-atom molecules, discrete edges + sparse-adjacency attention
#   af2        — 256-residue refinement, output_degrees=2, differentiable_coors
#   rev2048    — reversible trunk, 2048-node point cloud
PRESETS = {
    'qm9': dict(points=29, dim=128, heads=4, dim_head=32, depth=4,
                num_degrees=3, num_neighbors=8, batch=16,
                model_kwargs=dict(num_edge_tokens=4, edge_dim=16,
                                  attend_sparse_neighbors=True,
                                  num_adj_degrees=2, adj_dim=4,
                                  num_neighbors=0)),
    'af2': dict(points=256, dim=256, heads=8, dim_head=32, depth=4,
                num_degrees=2, num_neighbors=12, batch=1,
                model_kwargs=dict(output_degrees=2, differentiable_coors=True)),
    'rev2048': dict(points=2048, dim=256, heads=8, dim_head=32, depth=6,
                    num_degrees=3, num_neighbors=8, batch=1,
                    model_kwargs=dict(reversible=True)),
}


def build_model(args, device):
    kwargs = dict(
        dim=args.dim,
        heads=args.heads,
        dim_head=args.dim_head,
        depth=args.depth,
        num_degrees=args.num_degrees,
        valid_radius=args.valid_radius,
        num_neighbors=args.num_neighbors,
        attend_self=True,
    )
    kwargs.update(getattr(args, 'model_kwargs', {}))
    with torch.device(device):
        model = SE3Transformer(**kwargs)
    return model


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=5)
    p.add_argument('--warmup', type=int, default=2)
    p.add_argument('--batch', type=int, default=1, help='per-GPU batch size')
    p.add_argument('--points', type=int, default=1024)
    p.add_argument('--dim', type=int, default=512)
    p.add_argument('--heads', type=int, default=8)
    p.add_argument('--dim-head', type=int, default=64)
    p.add_argument('--depth', type=int, default=6)
    p.add_argument('--num-degrees', type=int, default=4)
    p.add_argument('--num-neighbors', type=int, default=8)
    p.add_argument('--valid-radius', type=float, default=10.)
    p.add_argument('--dtype', type=str, default='bf16', choices=['bf16', 'fp32'])
    p.add_argument('--graph', dest='graph', action='store_true', default=None,
                   help='capture the train step in a hipGraph and replay '
                        '(default: on for single-GPU CUDA runs)')
    p.add_argument('--no-graph', dest='graph', action='store_false')
    p.add_argument('--device', type=str, default=None)
    p.add_argument('--timers', action='store_true',
                   help='report per-phase (fwd/bwd/opt) hipEvent timings')
    p.add_argument('--grad-compression', default='none',
                   choices=['none', 'bf16'],
                   help='DP gradient all-reduce dtype (bf16 halves xGMI '
                        'traffic; see parallel/ddp.py for the error bound)')
    p.add_argument('--optimizer', default='sgd', choices=['sgd', 'adam'])
    p.add_argument('--zero1', action='store_true',
                   help='shard optimizer state across ranks (ZeRO-1, '
                        'parallel/zero.py). At the 18.2B-param headline '
                        'config Adam state is ~146 GB — it only fits '
                        'sharded across multiple GPUs.')
    p.add_argument('--preset', default=None, choices=sorted(PRESETS),
                   help='one of the BASELINE.json named configs')
    args = p.parse_args()
    args.model_kwargs = {}
    if args.preset:
        # preset values fill in wherever the user did not pass the flag
        # explicitly (an explicit --batch etc. wins over the preset)
        for k, v in PRESETS[args.preset].items():
            if k == 'model_kwargs' or getattr(args, k, None) == p.get_default(k):
                setattr(args, k, v)

    rank, world, local_rank = setup_distributed()
    use_cuda = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    else:
        device = torch.device(f'cuda:{local_rank}') if use_cuda else torch.device('cpu')
    if use_cuda:
        torch.cuda.set_device(device)

    torch.manual_seed(1234)  # identical init on every rank (bypasses broadcast cost)

    model = build_model(args, device)
    n_params = sum(p_.numel() for p_ in model.parameters())

    ddp = DistributedDataParallelSE3(
        model, sync_params=False,
        grad_compression=args.grad_compression) if world > 1 else None
    runner = ddp if ddp is not None else model

    optim_cls = torch.optim.AdamW if args.optimizer == 'adam' \
        else torch.optim.SGD
    opt_kwargs = dict(lr=1e-4)
    will_graph = (args.graph if args.graph is not None else world == 1) \
        and use_cuda
    if args.optimizer == 'adam' and will_graph:
        # lets AdamW.step() run under hipGraph capture (state tensors stay
        # on-device instead of python scalars)
        opt_kwargs['capturable'] = True
    if args.zero1:
        from se3_transformer_amd.parallel import Zero1Optimizer
        opt = Zero1Optimizer(model.parameters(), optim_cls, **opt_kwargs)
    else:
        opt = optim_cls(model.parameters(), **opt_kwargs)

    # synthetic protein-like point clouds, fixed per rank
    g = torch.Generator(device='cpu').manual_seed(1000 + rank)
    feats = torch.randn(args.batch, args.points, args.dim, generator=g).to(device)
    coors = (torch.randn(args.batch, args.points, 3, generator=g) * 2.0).to(device)
    mask = torch.ones(args.batch, args.points, dtype=torch.bool, device=device)
    target = torch.randn(args.batch, args.points, args.dim, generator=g).to(device)
    fwd_extra = {}
    if args.preset == 'qm9':
        i = torch.arange(args.points, device=device)
        fwd_extra['adj_mat'] = (i[:, None] - i[None, :]).abs() == 1  # molecular chain
        fwd_extra['edges'] = torch.randint(0, 4, (args.batch, args.points, args.points),
                                           generator=g, device='cpu').to(device)

    autocast_dtype = torch.bfloat16 if args.dtype == 'bf16' else torch.float32
    autocast_enabled = args.dtype == 'bf16'

    phase_events = []
    if args.timers and use_cuda:
        phase_events = [torch.cuda.Event(enable_timing=True) for _ in range(4)]

    def train_step():
        if ddp is not None:
            ddp.zero_grad_buffers()
        else:
            opt.zero_grad(set_to_none=False)
        if phase_events:
            phase_events[0].record()
        with torch.autocast(device_type=device.type, dtype=autocast_dtype,
                            enabled=autocast_enabled):
            out = runner(feats, coors, mask, return_type=0, **fwd_extra)
            loss = (out.float() - target).pow(2).mean()
        if phase_events:
            phase_events[1].record()
        loss.backward()
        if ddp is not None:
            ddp.finalize()
        if phase_events:
            phase_events[2].record()
        opt.step()
        if phase_events:
            phase_events[3].record()
        return loss

    def report_phases(tag):
        if not phase_events:
            return
        torch.cuda.synchronize()
        f = phase_events[0].elapsed_time(phase_events[1])
        bw = phase_events[1].elapsed_time(phase_events[2])
        op = phase_events[2].elapsed_time(phase_events[3])
        print(f'[bench] {tag}: fwd {f:.1f} ms | bwd {bw:.1f} ms | opt {op:.1f} ms',
              file=sys.stderr, flush=True)

    # hipGraph capture: the step is shape-static, so capture once and replay —
    # removes the Python/launch-gap overhead between the thousands of kernels
    # per step. Default on for single-GPU. For world > 1 pass --graph
    # explicitly: capture of an RCCL collective is VALIDATED at 1 rank
    # (tests/test_gpu_nccl.py::test_hipgraph_capture_of_rccl_allreduce), but
    # multi-rank capture has not run on real multi-GPU hardware yet and a
    # capture hang would kill a scaling run — the eager fallback costs ~15%
    # and is first-try safe.
    graph_mode = (args.graph if args.graph is not None else world == 1) \
        and use_cuda
    graph_captured = False
    if graph_mode:
        try:
            for _ in range(2):      # allocator warmup before capture
                train_step()
            torch.cuda.synchronize()
            if world > 1:
                torch.distributed.barrier()
            g_step = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g_step):
                train_step()
            train_step = lambda: g_step.replay()  # noqa: E731
            graph_captured = True
            if rank == 0:
                print('[bench] hipGraph capture ok', file=sys.stderr, flush=True)
        except Exception as e:      # pragma: no cover
            if rank == 0:
                print(f'[bench] graph capture failed ({e}); eager steps',
                      file=sys.stderr, flush=True)

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        t = time.perf_counter()
        train_step()
        if use_cuda:
            torch.cuda.synchronize()
        if rank == 0:
            print(f'[bench] warmup {i}: {time.perf_counter() - t:.2f}s',
                  file=sys.stderr, flush=True)
            report_phases(f'warmup {i}')

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        train_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks (slowest rank defines throughput)
    if world > 1:
        t = torch.tensor([elapsed], device=device if use_cuda else 'cpu')
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000.
    samples_per_sec = (args.batch * world * args.steps) / elapsed

    if rank == 0:
        metric = 'train samples/sec (fwd+bwd), 1024-pt dim=512 depth=6 num_degrees=4'
        if args.preset:
            metric = f'train samples/sec (fwd+bwd), preset={args.preset}'
        print(json.dumps({
            'metric': metric,
            'value': samples_per_sec,
            'unit': 'samples/sec',
            'n_gpus': world,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': ms_per_step,
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': args.dtype,
            'data': 'synthetic',
            'config': {
                'model': 'SE3Transformer',
                'dim': args.dim, 'heads': args.heads, 'dim_head': args.dim_head,
                'depth': args.depth, 'num_degrees': args.num_degrees,
                'points': args.points, 'num_neighbors': args.num_neighbors,
                'global_batch': args.batch * world,
                'parallelism': f'dp{world}',
                'n_params': n_params,
                'optimizer': args.optimizer + ('+zero1' if args.zero1 else ''),
                'preset': args.preset,
                'hipgraph': graph_captured,
            },
        }))

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == '__main__':
    main()
