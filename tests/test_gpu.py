"""GPU-only tests: run on a real MI355X via `pytest -m gpu`.

Covers: forward/backward smoke on cuda, fp32 equivariance on device
(mirrors reference tests/test_equivariance.py:142-162 bound of 1e-4),
and autocast-bf16 stability of the flagship config at small size.
"""
import pytest
import torch

from se3_transformer_amd import SE3Transformer
from se3_transformer_amd.ops.wigner import rot

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason='no GPU')


@needs_gpu
def test_forward_backward_cuda():
    device = torch.device('cuda')
    model = SE3Transformer(dim=64, heads=4, dim_head=16, depth=2,
                           num_degrees=3, num_neighbors=8).to(device)
    feats = torch.randn(2, 64, 64, device=device)
    coors = torch.randn(2, 64, 3, device=device)
    mask = torch.ones(2, 64, dtype=torch.bool, device=device)
    out = model(feats, coors, mask, return_type=0)
    out.pow(2).mean().backward()
    torch.cuda.synchronize()
    assert out.shape == (2, 64, 64)
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


@needs_gpu
def test_equivariance_cuda_fp32():
    device = torch.device('cuda')
    torch.manual_seed(0)
    model = SE3Transformer(dim=32, heads=2, dim_head=16, depth=1,
                           num_degrees=2, num_neighbors=8,
                           output_degrees=2).to(device)
    feats = torch.randn(1, 32, 32, device=device)
    coors = torch.randn(1, 32, 3, device=device)
    mask = torch.ones(1, 32, dtype=torch.bool, device=device)
    R = rot(15., 97., 263.).to(device)
    out1 = model(feats, coors @ R, mask, return_type=1)
    out2 = model(feats, coors, mask, return_type=1) @ R
    err = (out1 - out2).abs().max().item()
    assert err < 1e-4, f'equivariance error {err}'


@needs_gpu
def test_bf16_autocast_finite():
    device = torch.device('cuda')
    model = SE3Transformer(dim=128, heads=4, dim_head=32, depth=2,
                           num_degrees=4, num_neighbors=8).to(device)
    feats = torch.randn(1, 128, 128, device=device)
    coors = torch.randn(1, 128, 3, device=device)
    mask = torch.ones(1, 128, dtype=torch.bool, device=device)
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        out = model(feats, coors, mask, return_type=0)
    out.float().pow(2).mean().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(out).all()


@needs_gpu
def test_reversible_bf16_fused():
    model = SE3Transformer(dim=32, heads=2, dim_head=16, depth=2,
                           num_degrees=2, num_neighbors=6,
                           reversible=True).to('cuda')
    feats = torch.randn(1, 48, 32, device='cuda')
    coors = torch.randn(1, 48, 3, device='cuda')
    mask = torch.ones(1, 48, dtype=torch.bool, device='cuda')
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        out = model(feats, coors, mask, return_type=0)
    out.float().pow(2).mean().backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


@needs_gpu
def test_egnn_trunk_cuda():
    model = SE3Transformer(dim=32, depth=2, num_degrees=2, num_neighbors=6,
                           use_egnn=True, egnn_hidden_dim=16,
                           num_edge_tokens=4, edge_dim=4).to('cuda')
    feats = torch.randn(1, 48, 32, device='cuda')
    coors = torch.randn(1, 48, 3, device='cuda')
    mask = torch.ones(1, 48, dtype=torch.bool, device='cuda')
    edges = torch.randint(0, 4, (1, 48, 48), device='cuda')
    out = model(feats, coors, mask, edges=edges, return_type=0)
    out.pow(2).mean().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(out).all()


@needs_gpu
def test_one_headed_kv_cuda_bf16():
    model = SE3Transformer(dim=32, heads=2, dim_head=16, depth=1,
                           num_degrees=2, num_neighbors=6,
                           one_headed_key_values=True).to('cuda')
    feats = torch.randn(1, 48, 32, device='cuda')
    coors = torch.randn(1, 48, 3, device='cuda')
    mask = torch.ones(1, 48, dtype=torch.bool, device='cuda')
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        out = model(feats, coors, mask, return_type=0)
    assert torch.isfinite(out).all()


@needs_gpu
def test_differentiable_coors_cuda():
    """AF2-style refinement config: grads must flow to coordinates."""
    model = SE3Transformer(dim=24, heads=2, dim_head=12, depth=1,
                           num_degrees=2, output_degrees=2, num_neighbors=6,
                           differentiable_coors=True).to('cuda')
    feats = torch.randn(1, 40, 24, device='cuda')
    coors = torch.randn(1, 40, 3, device='cuda', requires_grad=True)
    mask = torch.ones(1, 40, dtype=torch.bool, device='cuda')
    out = model(feats, coors, mask, return_type=1)
    out.pow(2).mean().backward()
    torch.cuda.synchronize()
    assert coors.grad is not None and torch.isfinite(coors.grad).all()


@needs_gpu
def test_equivariance_fused_bf16_autocast():
    """Rotation equivariance of the FULL model with every fused kernel
    active under autocast-bf16. The band is calibrated
    in-test against the measured bf16 rounding noise floor (fused-bf16 vs
    eager-fp32 on the SAME inputs): equivariance error must stay within a
    small multiple of that floor — i.e. rotation breaks nothing beyond the
    precision the dtype already costs."""
    from se3_transformer_amd.ops.wigner import rot
    torch.manual_seed(21)
    model = SE3Transformer(dim=64, heads=4, dim_head=16, depth=2,
                           num_degrees=3, num_neighbors=8, output_degrees=2,
                           attend_self=True).to('cuda')
    feats = torch.randn(1, 48, 64, device='cuda')
    coors = torch.randn(1, 48, 3, device='cuda') * 1.5
    mask = torch.ones(1, 48, dtype=torch.bool, device='cuda')
    R = rot(23., 117., 195.).float().to('cuda')
    # rotate OUTSIDE autocast: under autocast the matmul would round the
    # rotated coordinates to bf16, perturbing the k-NN selection itself
    # (discrete neighbor flips, not a kernel equivariance failure)
    coors_rot = coors @ R
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        out1 = model(feats, coors_rot, mask, return_type=1)
        out2 = model(feats, coors, mask, return_type=1)
    ref = model(feats, coors, mask, return_type=1)     # eager fp32 oracle
    denom = ref.abs().max().clamp(min=1e-6)
    noise = ((out2.float() - ref).abs().max() / denom).item()   # bf16 floor
    err = ((out1.float() - out2.float() @ R).abs().max() / denom).item()
    assert noise < 0.1, f'bf16 path drifted from fp32: {noise}'
    assert err < max(4 * noise, 2e-2), \
        f'bf16 fused equivariance {err} vs noise floor {noise}'


@needs_gpu
def test_equivariance_num_degrees_4_f64_cuda():
    """Degree-4 float64 equivariance ON DEVICE (1e-8), mirroring the CPU
    test at tests/test_equivariance.py (reference strictest regime)."""
    from se3_transformer_amd.ops.wigner import rot
    from se3_transformer_amd.utils import torch_default_dtype
    with torch_default_dtype(torch.float64):
        torch.manual_seed(0)
        model = SE3Transformer(dim=8, heads=2, dim_head=4, depth=1,
                               attend_self=True, num_neighbors=4,
                               num_degrees=4, output_degrees=2).to('cuda')
        feats = torch.randn(1, 12, 8, device='cuda')
        coors = torch.randn(1, 12, 3, device='cuda')
        mask = torch.ones(1, 12, dtype=torch.bool, device='cuda')
        R = rot(23., 117., 195.).to(torch.float64).to('cuda')
        out1 = model(feats, coors @ R, mask, return_type=1)
        out2 = model(feats, coors, mask, return_type=1) @ R
        diff = (out1 - out2).abs().max().item()
        assert diff < 1e-8, f'degree-4 f64 equivariance on GPU: {diff}'


@needs_gpu
def test_hipgraph_full_step_capture_replay():
    """Capture one fused train step (fwd+bwd+SGD) in a hipGraph and replay:
    the mechanism bench.py's single-GPU default relies on. Parameters must
    keep moving across replays and grads stay finite."""
    torch.manual_seed(30)
    model = SE3Transformer(dim=32, heads=2, dim_head=16, depth=1,
                           num_degrees=2, num_neighbors=6,
                           attend_self=True).to('cuda')
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    feats = torch.randn(1, 48, 32, device='cuda')
    coors = torch.randn(1, 48, 3, device='cuda')
    mask = torch.ones(1, 48, dtype=torch.bool, device='cuda')

    def step():
        opt.zero_grad(set_to_none=False)
        with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
            out = model(feats, coors, mask, return_type=0)
            loss = out.float().pow(2).mean()
        loss.backward()
        opt.step()

    for _ in range(2):        # allocator warmup
        step()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step()
    p0 = next(model.parameters())
    before = p0.detach().clone()
    g.replay()
    torch.cuda.synchronize()
    after1 = p0.detach().clone()
    g.replay()
    torch.cuda.synchronize()
    after2 = p0.detach().clone()
    assert not torch.equal(before, after1), 'replay 1 did not update params'
    assert not torch.equal(after1, after2), 'replay 2 did not update params'
    assert all(torch.isfinite(p).all() for p in model.parameters())
