"""Microbenchmark the fused pairconv kernels at headline shapes.

Usage (GPU box):  python scripts/bench_pairconv.py [--pair 3,3]
Prints per-kernel ms and effective TFLOP/s for fwd, bwd_dh, bwd_dw, bwd_du.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from se3_transformer_amd import _C


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--pair', default='3,3')
    p.add_argument('--E', type=int, default=9216)
    p.add_argument('--ch', type=int, default=512)
    p.add_argument('--only', default=None, choices=['fwd','dh','dw','du'])
    args = p.parse_args()
    di, do = map(int, args.pair.split(','))
    F = 2 * min(di, do) + 1
    O = 2 * do + 1
    mi = mo = args.ch
    miF = mi * F
    E, K = args.E, 128
    N = mo * miF
    dev = 'cuda'
    g = torch.Generator(device=dev).manual_seed(0)

    H = torch.randn(E, K, generator=g, device=dev).to(torch.bfloat16)
    W = (torch.randn(N, K, generator=g, device=dev) / K**0.5).to(torch.bfloat16)
    Ut = torch.randn(miF, O, E, generator=g, device=dev).to(torch.bfloat16)
    out = torch.zeros(E, mo, O, device=dev)
    gt = torch.randn(mo, O, E, generator=g, device=dev).to(torch.bfloat16)
    bias = torch.zeros(N, device=dev)

    gemm_fl = 2.0 * E * N * K
    epi_fl = 2.0 * E * N * O

    from se3_transformer_amd.ops.fused import _pack_w_dh, _pack_w_fwd
    P = _pack_w_fwd(W, mo, miF)

    if hasattr(_C, 'pack_w_both'):
        Pf2 = torch.empty(N * K, dtype=torch.bfloat16, device=dev)
        Pdh2 = torch.empty(N * K, dtype=torch.bfloat16, device=dev)
        _C.pack_w_both(W, Pf2, Pdh2, mo)
        okf = torch.equal(Pf2.view_as(P), P)
        okd = torch.equal(Pdh2.view(_pack_w_dh(W, mo, miF).shape),
                          _pack_w_dh(W, mo, miF))
        ms = timeit(lambda: _C.pack_w_both(W, Pf2, Pdh2, mo))
        gb = (N * K * 2 * 3) / 1e9   # read W bf16 + write both
        print(f'pack_w_both: {ms:8.3f} ms  {gb/ms:7.2f} TB/s  '
              f'parity fwd={okf} dh={okd}')

    if os.environ.get('SE3_SWEEP_UU'):
        for uu, mb2 in (('0', '1'), ('1', '1'), ('2', '1'),
                        ('0', '2'), ('1', '2'), ('2', '2')):
            os.environ['SE3_FWD_UU'] = uu
            os.environ['SE3_FWD_MB2'] = mb2
            ms = timeit(lambda: _C.pairconv_fwd(H, P, Ut, out, mo))
            print(f'fwd UU={uu} MB2={mb2} ({di},{do}): {ms:8.3f} ms  '
                  f'{gemm_fl/ms/1e9:7.1f} TF/s (gemm)')
        del os.environ['SE3_FWD_UU']
        for mb2 in ('1', '2'):
            os.environ['SE3_FWD_WP'] = '1'
            os.environ['SE3_FWD_MB2'] = mb2
            ms = timeit(lambda: _C.pairconv_fwd(H, P, Ut, out, mo))
            print(f'fwd WP=1 MB2={mb2} ({di},{do}): {ms:8.3f} ms  '
                  f'{gemm_fl/ms/1e9:7.1f} TF/s (gemm)')
        del os.environ['SE3_FWD_WP']
        del os.environ['SE3_FWD_MB2']
    if args.only:
        fn = {'fwd': lambda: _C.pairconv_fwd(H, P, Ut, out, mo),
              'dh': lambda: _C.pairconv_bwd_dh(gt, Ut, _pack_w_dh(W, mo, miF),
                                               torch.zeros(E, K, device=dev), mo),
              'dw': lambda: _C.pairconv_bwd_dw(gt, Ut, H.t().contiguous(),
                                               torch.empty(N, K, device=dev), mo),
              'du': lambda: _C.pairconv_bwd_du(H, P, bias, gt,
                                               torch.empty(miF, O, E, device=dev), mo)}[args.only]
        print(args.only, timeit(fn, iters=3, warmup=1), 'ms')
        return
    ms = timeit(lambda: _C.pairconv_fwd(H, P, Ut, out, mo))
    print(f'fwd    ({di},{do}): {ms:8.3f} ms  {gemm_fl/ms/1e9:7.1f} TF/s (gemm) '
          f'{(gemm_fl+epi_fl)/ms/1e9:7.1f} TF/s (total)')

    P1 = _pack_w_dh(W, mo, miF)
    dH = torch.zeros(E, K, device=dev)
    ms = timeit(lambda: _C.pairconv_bwd_dh(gt, Ut, P1, dH, mo))
    print(f'bwd_dh ({di},{do}): {ms:8.3f} ms  {gemm_fl/ms/1e9:7.1f} TF/s')

    Ht = H.t().contiguous()
    dW = torch.empty(N, K, device=dev)
    ms = timeit(lambda: _C.pairconv_bwd_dw(gt, Ut, Ht, dW, mo))
    print(f'bwd_dw ({di},{do}): {ms:8.3f} ms  {gemm_fl/ms/1e9:7.1f} TF/s')

    dU = torch.empty(miF, O, E, device=dev)
    ms = timeit(lambda: _C.pairconv_bwd_du(H, P, bias, gt, dU, mo))
    print(f'bwd_du ({di},{do}): {ms:8.3f} ms  {gemm_fl/ms/1e9:7.1f} TF/s')

    # library GEMM reference: the raw R = H @ W^T (what eager must do, without
    # even the contraction), on a slab 1/8 of N to fit memory
    Nr = N // 8
    Wr = W[:Nr]
    ms = timeit(lambda: H @ Wr.t(), iters=5)
    print(f'hipblaslt H@W[{Nr}] : {ms:8.3f} ms  {2.0*E*Nr*K/ms/1e9:7.1f} TF/s'
          f'  (writes R slab: {E*Nr*2/1e9:.2f} GB)')


if __name__ == '__main__':
    main()
