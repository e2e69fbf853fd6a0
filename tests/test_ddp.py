"""Multi-process data-parallel tests (gloo backend, CPU, world_size=2).

Checks gradient parity: DP-averaged grads over a split batch must equal
single-process grads over the full batch.
"""
import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

from se3_transformer_amd import SE3Transformer


def _build_model():
    torch.manual_seed(42)
    return SE3Transformer(dim=16, depth=1, num_degrees=2, num_neighbors=4,
                          heads=2, dim_head=8, output_degrees=2)


def _make_batch(b=4, n=12):
    g = torch.Generator().manual_seed(123)
    feats = torch.randn(b, n, 16, generator=g)
    coors = torch.randn(b, n, 3, generator=g)
    mask = torch.ones(b, n).bool()
    return feats, coors, mask


def _single_process_grads():
    model = _build_model()
    feats, coors, mask = _make_batch()
    out = model(feats, coors, mask, return_type=1)
    loss = out.pow(2).mean()
    loss.backward()
    return {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}


def _worker(rank, world, port, results):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from se3_transformer_amd.parallel import DistributedDataParallelSE3
        model = _build_model()
        ddp = DistributedDataParallelSE3(model, bucket_bytes=1 << 16)
        feats, coors, mask = _make_batch()
        sl = slice(rank * 2, rank * 2 + 2)  # each rank takes 2 of the 4 samples
        ddp.zero_grad_buffers()
        out = ddp(feats[sl], coors[sl], mask[sl], return_type=1)
        # match the single-process mean loss: per-sample mean / world handled
        # by averaging allreduce since each rank's loss is the mean over its
        # half; d/dw mean_full = avg of d/dw mean_half
        loss = out.pow(2).mean()
        loss.backward()
        ddp.finalize()
        if rank == 0:
            results['grads'] = {n: p.grad.clone() for n, p in model.named_parameters()
                                if p.grad is not None}
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_grad_parity_gloo():
    port = 29371
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker, args=(2, port, results), nprocs=2, join=True)
    ddp_grads = results['grads']
    ref_grads = _single_process_grads()
    # DDP pre-assigns grad buffers to every param (unused ones stay zero);
    # the single-process reference leaves unused params with grad=None.
    assert set(ref_grads.keys()) <= set(ddp_grads.keys())
    for name in ref_grads:
        a, b = ddp_grads[name], ref_grads[name]
        assert torch.allclose(a, b, atol=1e-5), (name, (a - b).abs().max())
    for name in set(ddp_grads) - set(ref_grads):
        assert ddp_grads[name].abs().max() == 0, name


def _worker_bf16(rank, world, tmpdir):
    import os
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT='29617',
                      RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
    import torch
    import torch.distributed as dist
    dist.init_process_group('gloo', rank=rank, world_size=world)
    from se3_transformer_amd.parallel import DistributedDataParallelSE3
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.Linear(16, 4))
    ddp = DistributedDataParallelSE3(m, grad_compression='bf16')
    torch.manual_seed(100 + rank)
    x = torch.randn(8, 16)
    ddp.zero_grad_buffers()
    ddp(x).pow(2).mean().backward()
    ddp.finalize()
    g = torch.cat([p.grad.flatten() for p in m.parameters()])
    torch.save(g, f'{tmpdir}/bf16_g{rank}.pt')
    dist.destroy_process_group()


def test_ddp_bf16_compression_gloo(tmp_path):
    import torch.multiprocessing as mp
    world = 2
    ctx = mp.get_context('spawn')
    ps = [ctx.Process(target=_worker_bf16, args=(r, world, str(tmp_path)))
          for r in range(world)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(240)
        assert p.exitcode == 0
    import torch
    g0 = torch.load(tmp_path / 'bf16_g0.pt')
    g1 = torch.load(tmp_path / 'bf16_g1.pt')
    assert torch.equal(g0, g1)  # both ranks hold the same reduced grads


def _accum_worker(rank, world, port, results):
    import os
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from se3_transformer_amd.parallel import DistributedDataParallelSE3
        model = _build_model()
        ddp = DistributedDataParallelSE3(model, bucket_bytes=1 << 16)
        feats, coors, mask = _make_batch()   # 4 samples total
        ddp.zero_grad_buffers()
        # each rank: 2 micro-batches of 1 sample under no_sync + final one
        my = [rank * 2, rank * 2 + 1]
        with ddp.no_sync():
            s = slice(my[0], my[0] + 1)
            out = ddp(feats[s], coors[s], mask[s], return_type=1)
            (out.pow(2).mean() / 2).backward()
        s = slice(my[1], my[1] + 1)
        out = ddp(feats[s], coors[s], mask[s], return_type=1)
        (out.pow(2).mean() / 2).backward()
        ddp.finalize()
        if rank == 0:
            results['grads'] = {n: p.grad.clone()
                                for n, p in model.named_parameters()
                                if p.grad is not None}
    finally:
        torch.distributed.destroy_process_group()


def test_ddp_no_sync_grad_accumulation():
    """2 ranks x (2 accumulated micro-batches of 1 sample) must reproduce
    the single-process full-batch gradients: no_sync defers the
    all-reduce, the final backward reduces the accumulated sum."""
    ref = _single_process_grads()
    ctx = mp.get_context('spawn')
    with ctx.Manager() as man:
        results = man.dict()
        mp.start_processes(_accum_worker, args=(2, 29873, results), nprocs=2,
                           join=True, start_method='spawn')
        grads = dict(results['grads'])
    for n, g in ref.items():
        err = (grads[n] - g).abs().max().item()
        assert err < 1e-5, f'{n}: {err}'
