"""Protein-backbone denoising example (MI355X port of reference denoise.py).

Mirrors /root/reference/denoise.py:22-93 — same model configuration (token
embeddings, chain adjacency via `attend_sparse_neighbors` + `num_adj_degrees`,
`differentiable_coors`, type-1 output refinement) and the same training loop
(Adam 1e-4, grad-accumulate 16, coordinate-MSE on masked residues).

This environment has no network access, so sidechainnet CASP12 is replaced by
synthetic protein-like backbone chains: a smooth random walk of N residues x 3
backbone atoms. Run on GPU: `python examples/denoise.py [--steps 100] [--bf16]`.

Multi-GPU data parallel (the full MI355X training stack: bucketed RCCL
all-reduce overlapped with backward, no_sync() across the 16 accumulation
micro-batches, optional ZeRO-1 Adam-state sharding):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/denoise.py --bf16 [--zero1]
"""
import argparse
import os
import sys

import torch
import torch.nn.functional as F
from torch.optim import Adam

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from se3_transformer_amd import SE3Transformer
from se3_transformer_amd.parallel import (DistributedDataParallelSE3,
                                          Zero1Optimizer, setup_distributed)

BATCH_SIZE = 1
GRADIENT_ACCUMULATE_EVERY = 16


def synthetic_backbone(batch, length, generator):
    """Smooth random-walk chain: (b, length*3, 3) coords + residue tokens."""
    steps = torch.randn(batch, length, 3, generator=generator)
    ca = torch.cumsum(0.4 * steps / steps.norm(dim=-1, keepdim=True).clamp(min=1e-6)
                      + 0.1 * torch.randn(batch, length, 3, generator=generator), dim=1)
    offs = torch.tensor([[-0.05, 0., 0.], [0., 0., 0.], [0.05, 0., 0.]])
    coords = (ca.unsqueeze(2) + offs).reshape(batch, length * 3, 3)
    seqs = torch.randint(0, 24, (batch, length), generator=generator)
    seq = seqs.repeat_interleave(3, dim=1)
    mask = torch.ones(batch, length * 3, dtype=torch.bool)
    return seq, coords, mask


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--steps', type=int, default=10000)
    p.add_argument('--length', type=int, default=128, help='residues per chain')
    p.add_argument('--bf16', action='store_true')
    p.add_argument('--zero1', action='store_true',
                   help='shard the Adam state across ranks (ZeRO-1)')
    p.add_argument('--checkpoint', type=str, default=None,
                   help='checkpoint path; resumes from it when it exists. '
                        'Under --zero1 each rank also writes its optimizer '
                        'shard to <path>.r<rank>')
    p.add_argument('--save-every', type=int, default=100)
    args = p.parse_args()

    rank, world, local_rank = setup_distributed()
    device = torch.device(f'cuda:{local_rank}'
                          if torch.cuda.is_available() else 'cpu')
    torch.manual_seed(7)   # identical init on every rank
    transformer = SE3Transformer(
        num_tokens=24,
        dim=8,
        dim_head=8,
        heads=2,
        depth=2,
        attend_self=True,
        input_degrees=1,
        output_degrees=2,
        reduce_dim_out=True,
        differentiable_coors=True,
        num_neighbors=0,
        attend_sparse_neighbors=True,
        num_adj_degrees=2,
        adj_dim=4,
        num_degrees=2,
    ).to(device)

    ddp = DistributedDataParallelSE3(transformer, sync_params=False) \
        if world > 1 else None
    runner = ddp if ddp is not None else transformer
    if args.zero1:
        optim = Zero1Optimizer(transformer.parameters(), Adam, lr=1e-4)
    else:
        optim = Adam(transformer.parameters(), lr=1e-4)
    g = torch.Generator().manual_seed(1000 + rank)   # data shard per rank

    # checkpoint/resume: model weights are replicated (rank 0 writes them);
    # optimizer state is rank-local under ZeRO-1, so each rank persists its
    # own shard file next to the main checkpoint
    start_step = 0
    shard_path = (f'{args.checkpoint}.r{rank}'
                  if args.checkpoint and args.zero1 and world > 1 else None)
    if args.checkpoint and os.path.exists(args.checkpoint):
        ck = torch.load(args.checkpoint, map_location=device,
                        weights_only=False)
        transformer.load_state_dict(ck['model'])
        if shard_path:
            if os.path.exists(shard_path):
                optim.load_state_dict(torch.load(
                    shard_path, map_location=device, weights_only=False))
        else:
            optim.load_state_dict(ck['opt'])
        start_step = ck['step']
        if rank == 0:
            print(f'resumed from {args.checkpoint} at step {start_step}',
                  flush=True)
        # advance the data stream so the resumed run sees fresh batches
        g.manual_seed(1000 + rank + start_step * 100003)

    def save_checkpoint(step):
        if not args.checkpoint:
            return
        if rank == 0:
            tmp = args.checkpoint + '.tmp'
            torch.save({'step': step, 'model': transformer.state_dict(),
                        'opt': {} if shard_path else optim.state_dict()}, tmp)
            os.replace(tmp, args.checkpoint)   # atomic: never a torn file
        if shard_path:
            tmp = shard_path + '.tmp'
            torch.save(optim.state_dict(), tmp)
            os.replace(tmp, shard_path)

    import contextlib

    def micro_batch():
        seq, coords, masks = synthetic_backbone(BATCH_SIZE, args.length, g)
        seq, coords, masks = seq.to(device), coords.to(device), masks.to(device)

        noised_coords = coords + torch.randn_like(coords)

        i = torch.arange(seq.shape[-1], device=device)
        adj_mat = (i[:, None] >= (i[None, :] - 1)) & (i[:, None] <= (i[None, :] + 1))

        with torch.autocast(device_type=device.type, dtype=torch.bfloat16,
                            enabled=args.bf16):
            out = runner(seq, noised_coords, mask=masks,
                         adj_mat=adj_mat, return_type=1)
        denoised_coords = noised_coords + out.float()
        loss = F.mse_loss(denoised_coords[masks], coords[masks])
        (loss / GRADIENT_ACCUMULATE_EVERY).backward()
        return loss

    for step in range(start_step, args.steps):
        if ddp is not None:
            ddp.zero_grad_buffers()
        # all-reduce only on the last accumulation micro-batch
        ctx = ddp.no_sync() if ddp is not None else contextlib.nullcontext()
        with ctx:
            for _ in range(GRADIENT_ACCUMULATE_EVERY - 1):
                micro_batch()
        loss = micro_batch()
        if ddp is not None:
            ddp.finalize()
        if rank == 0:
            print('loss:', loss.item(), flush=True)
        optim.step()
        optim.zero_grad()
        if (step + 1) % args.save_every == 0 or step + 1 == args.steps:
            save_checkpoint(step + 1)


if __name__ == '__main__':
    main()
