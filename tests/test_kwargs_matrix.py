"""CPU sweep over reference constructor-kwarg combinations (reference README
examples + se3_transformer_pytorch.py:937-982): every configuration must
build, run forward, and backprop finite gradients."""
import pytest
import torch

from se3_transformer_amd import SE3Transformer

CASES = {
    'fourier_dist': dict(fourier_encode_dist=True, rel_dist_num_fourier_features=2),
    'norm_out_reduce': dict(norm_out=True, reduce_dim_out=True),
    'conv_layers': dict(num_conv_layers=1),
    # dim_head must be divisible by 2*num_rotaries (same constraint as the
    # reference's SinusoidalEmbeddings splitting)
    'null_global_rotary': dict(use_null_kv=True, global_feats_dim=8,
                               rotary_position=True, rotary_rel_dist=True,
                               dim_head=8),
    'tie_kv': dict(tie_key_values=True),
    'linear_proj_keys': dict(linear_proj_keys=True),
    'gated_norm': dict(norm_gated_scale=True),
    'hidden_fiber_dict': dict(num_degrees=None, hidden_fiber_dict={0: 12, 1: 8}),
    'out_fiber_dict': dict(out_fiber_dict={0: 6, 1: 4}, num_degrees=2),
    'edge_tokens': dict(num_edge_tokens=4, edge_dim=3),
    'tokens_positions': dict(num_tokens=11, num_positions=64),
    'one_headed_kv': dict(one_headed_key_values=True),
    'causal': dict(causal=True),
    'input_degrees_2': dict(input_degrees=2, dim_in=(12, 6)),
}


@pytest.mark.parametrize('name', sorted(CASES))
def test_kwarg_combo(name):
    kw = dict(dim=12, heads=2, dim_head=6, depth=1, num_degrees=2,
              num_neighbors=4, attend_self=True)
    kw.update(CASES[name])
    torch.manual_seed(0)
    model = SE3Transformer(**kw)

    b, n = 1, 16
    if name == 'tokens_positions':
        feats = torch.randint(0, 11, (b, n))
    elif name == 'input_degrees_2':
        feats = {'0': torch.randn(b, n, 12, 1), '1': torch.randn(b, n, 6, 3)}
    else:
        feats = torch.randn(b, n, 12)
    coors = torch.randn(b, n, 3)
    mask = torch.ones(b, n, dtype=torch.bool)
    extra = {}
    if name == 'edge_tokens':
        extra['edges'] = torch.randint(0, 4, (b, n, n))
    if name == 'null_global_rotary':
        extra['global_feats'] = torch.randn(b, 2, 8)

    out = model(feats, coors, mask, return_type=0, **extra)
    out.pow(2).mean().backward()
    assert torch.isfinite(out).all()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)
