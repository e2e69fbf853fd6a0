#!/usr/bin/env python
"""Measure the ACTUAL reference library (lucidrains/se3-transformer-pytorch,
mounted read-only at /root/reference) against this framework on the same GPU.

The reference cannot run as shipped on this image:
  * its `data/J_dense.pt` Wigner-J blobs are absent (.MISSING_LARGE_BLOBS;
    SURVEY.md §2.1) — irr_repr.py:12-20 fails at import;
  * `filelock` is not installed here.
This script therefore (at runtime, nothing is copied into this repo):
  1. copies the reference package into a temp dir,
  2. GENERATES the J matrices from this framework's own Wigner-D solver:
     J_l = D_l(pi/2, pi/2, pi/2), the representation of the y<->z exchange
     rotation (R_J = diag-ish [[-1,0,0],[0,0,1],[0,1,0]], a proper rotation
     with R_J^2 = I), which satisfies the reference's factorization
     D(a,b,c) = x_a J x_b J x_c  (irr_repr.py:22-30),
  3. verifies the generated J with the reference's OWN equivariance
     identity Y(R x) = D(R) Y(x) (tests/test_irrep_repr.py property),
  4. times reference-eager and this framework on identical synthetic data
     at a config the reference's R-materializing path can actually fit.

Note the README headline config (n=1024, dim=512, num_degrees=4) is
infeasible for the reference implementation on ANY hardware: its per-edge
kernel matrix (se3_transformer_pytorch.py:326-343) for one (3,3) degree
pair alone is (7*512)x(7*512)x4B ~= 51 MB/edge -> tens of PB at n=1024.
The comparison config here is the largest common one; the new framework's
headline numbers live in bench.py / BENCH_rNN.json.
"""
from __future__ import annotations

import argparse
import json
import os
import shutil
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402


def stage_reference(tmp='/tmp/ref_baseline'):
    """Copy the reference package + shims into an importable temp tree.

    On GPU boxes /root/reference is not mounted; to run there, copy it
    into the git-ignored .ref_stage/ first (kept out of the repo history
    and DELETED after measuring — reference code never lives in the
    repo):  cp -r /root/reference/se3_transformer_pytorch .ref_stage/"""
    ref_src = '/root/reference/se3_transformer_pytorch'
    if not os.path.isdir(ref_src):
        ref_src = os.path.join(REPO, '.ref_stage', 'se3_transformer_pytorch')
    if not os.path.isdir(ref_src):
        raise SystemExit('reference not present on this box: ' + ref_src)
    pkg = os.path.join(tmp, 'se3_transformer_pytorch')
    if os.path.isdir(tmp):
        shutil.rmtree(tmp)
    shutil.copytree(ref_src, pkg)
    # minimal filelock shim (not installed in this image; single-process use)
    with open(os.path.join(tmp, 'filelock.py'), 'w') as f:
        f.write('class FileLock:\n'
                '    def __init__(self, *a, **k): pass\n'
                '    def acquire(self, *a, **k): pass\n'
                '    def release(self, *a, **k): pass\n'
                '    def __enter__(self): return self\n'
                '    def __exit__(self, *a): return False\n')
    return tmp, pkg


def generate_j(pkg, max_l=12):
    """J_l = D_l(pi/2, pi/2, pi/2) from this framework's Wigner solver."""
    from se3_transformer_amd.ops.wigner import wigner_d_matrix
    half = torch.tensor(torch.pi / 2, dtype=torch.float64)
    jd = []
    for l in range(max_l + 1):
        J = wigner_d_matrix(l, half, half, half).to(torch.float64)
        assert torch.allclose(J @ J, torch.eye(2 * l + 1, dtype=torch.float64),
                              atol=1e-8), f'J_{l} is not an involution'
        jd.append(J)
    os.makedirs(os.path.join(pkg, 'data'), exist_ok=True)
    torch.save(jd, os.path.join(pkg, 'data', 'J_dense.pt'))


def verify_reference(tol=1e-9):
    """The reference's own identity test (tests/test_irrep_repr.py:7-33):
    Y(Z(a)Y(b)Z(c) x) = D(a,b,c) Y(x) — run against the GENERATED J."""
    from se3_transformer_pytorch.irr_repr import (compose, irr_repr,
                                                  spherical_harmonics)
    from se3_transformer_pytorch.spherical_harmonics import \
        clear_spherical_harmonics_cache
    old = torch.get_default_dtype()
    torch.set_default_dtype(torch.float64)
    worst = 0.
    try:
        torch.manual_seed(0)
        for order in range(7):
            a, b = torch.rand(2)
            alpha, beta, gamma = torch.rand(3)
            ra, rb, _ = compose(alpha, beta, gamma, a, b, 0)
            y_rx = spherical_harmonics(order, ra, rb)
            clear_spherical_harmonics_cache()
            y = spherical_harmonics(order, a, b)
            clear_spherical_harmonics_cache()
            dry = irr_repr(order, alpha, beta, gamma) @ y
            worst = max(worst, ((y_rx - dry).abs().max()
                                / y.abs().max()).item())
    finally:
        torch.set_default_dtype(old)
    print(f'[ref_baseline] generated-J identity check (ref test, f64): '
          f'max rel err {worst:.2e}', file=sys.stderr, flush=True)
    assert worst < tol, 'generated J matrices fail the reference identity'
    return worst


def bench_model(model, device, steps, warmup, batch, n, dim, autocast_bf16):
    g = torch.Generator(device='cpu').manual_seed(5)
    feats = torch.randn(batch, n, dim, generator=g).to(device)
    coors = (torch.randn(batch, n, 3, generator=g) * 2.0).to(device)
    mask = torch.ones(batch, n, dtype=torch.bool, device=device)
    target = torch.randn(batch, n, dim, generator=g).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=1e-4)

    def step():
        opt.zero_grad(set_to_none=False)
        with torch.autocast(device_type='cuda', dtype=torch.bfloat16,
                            enabled=autocast_bf16):
            out = model(feats, coors, mask, return_type=0)
            loss = (out.float() - target).pow(2).mean()
        loss.backward()
        opt.step()
        return loss

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    return dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--points', type=int, default=128)
    p.add_argument('--dim', type=int, default=64)
    p.add_argument('--heads', type=int, default=4)
    p.add_argument('--dim-head', type=int, default=16)
    p.add_argument('--depth', type=int, default=2)
    p.add_argument('--num-degrees', type=int, default=3)
    p.add_argument('--num-neighbors', type=int, default=8)
    p.add_argument('--batch', type=int, default=1)
    p.add_argument('--steps', type=int, default=5)
    p.add_argument('--warmup', type=int, default=2)
    p.add_argument('--out', default=None, help='write JSON here as well')
    p.add_argument('--verify-only', action='store_true',
                   help='stage + generate J + run the equivariance check '
                        '(CPU-capable), skip the timing runs')
    args = p.parse_args()

    tmp, pkg = stage_reference()
    generate_j(pkg)
    sys.path.insert(0, tmp)
    j_err = verify_reference()
    if args.verify_only:
        print(json.dumps({'generated_J_max_rel_err': j_err, 'ok': True}))
        return
    assert torch.cuda.is_available(), 'run on the GPU box'
    device = torch.device('cuda')

    kwargs = dict(dim=args.dim, heads=args.heads, dim_head=args.dim_head,
                  depth=args.depth, num_degrees=args.num_degrees,
                  num_neighbors=args.num_neighbors, valid_radius=10.,
                  attend_self=True)

    from se3_transformer_pytorch import SE3Transformer as RefSE3
    torch.manual_seed(11)
    ref_model = RefSE3(**kwargs).to(device)
    ref_s = bench_model(ref_model, device, args.steps, args.warmup,
                        args.batch, args.points, args.dim, False)
    del ref_model
    torch.cuda.empty_cache()

    from se3_transformer_amd import SE3Transformer as AmdSE3
    torch.manual_seed(11)
    amd_model = AmdSE3(**kwargs).to(device)
    amd_fp32_s = bench_model(amd_model, device, args.steps, args.warmup,
                             args.batch, args.points, args.dim, False)
    amd_bf16_s = bench_model(amd_model, device, args.steps, args.warmup,
                             args.batch, args.points, args.dim, True)

    result = {
        'what': 'reference library (eager, fp32) vs se3_transformer_amd, '
                'same node / same synthetic data / random-init weights',
        'config': {k: getattr(args, k.replace('-', '_')) for k in
                   ('points', 'dim', 'heads', 'dim_head', 'depth',
                    'num_degrees', 'num_neighbors', 'batch')},
        'generated_J_max_rel_err': j_err,
        'reference_s_per_step_fp32': ref_s,
        'amd_s_per_step_fp32': amd_fp32_s,
        'amd_s_per_step_bf16': amd_bf16_s,
        'speedup_fp32': ref_s / amd_fp32_s,
        'speedup_bf16_vs_ref_fp32': ref_s / amd_bf16_s,
        'steps': args.steps, 'warmup': args.warmup,
        'device': torch.cuda.get_device_name(0),
    }
    line = json.dumps(result)
    print(line)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, 'w') as f:
            f.write(line + '\n')


if __name__ == '__main__':
    main()
