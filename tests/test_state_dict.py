"""State-dict layout parity with the reference module tree.

The north star fixes the state_dict layout as API: names come
from the reference module tree (se3_transformer_pytorch.py). These expected
names are hand-derived from the reference source, cited per group.
"""
import torch

from se3_transformer_amd import SE3Transformer


def _names(model):
    return set(model.state_dict().keys())


def test_attention_model_state_dict_layout():
    model = SE3Transformer(dim=8, depth=2, num_degrees=2, num_neighbors=4,
                           heads=2, dim_head=4, attend_self=True,
                           use_null_kv=True, num_conv_layers=1,
                           num_tokens=10, num_positions=64,
                           edge_dim=3, num_edge_tokens=5,
                           num_adj_degrees=2, adj_dim=4,
                           output_degrees=2, reduce_dim_out=True, norm_out=True)
    names = _names(model)
    expected = [
        # embeddings (reference :991-1039)
        'token_emb.weight', 'pos_emb.weight', 'edge_emb.weight', 'adj_emb.weight',
        # conv_in: kernel_unary.(di,do).rp.net.{0,1,3,4,6} (reference :188-193, :287-295)
        'conv_in.kernel_unary.(0,0).rp.net.0.weight',
        'conv_in.kernel_unary.(0,0).rp.net.0.bias',
        'conv_in.kernel_unary.(0,0).rp.net.1.weight',
        'conv_in.kernel_unary.(0,0).rp.net.3.weight',
        'conv_in.kernel_unary.(0,0).rp.net.4.bias',
        'conv_in.kernel_unary.(0,0).rp.net.6.weight',
        'conv_in.kernel_unary.(0,1).rp.net.6.bias',
        # self-interaction LinearSE3 (reference :198-201, :85-89)
        'conv_in.self_interact.weights.0',  # intersection of in/out fibers: degree 0 only
        # pre-conv stack (reference :1078-1083)
        'convs.0.0.kernel_unary.(1,0).rp.net.0.weight',
        'convs.0.1.transform.0.scale',
        # attention trunk (reference :1099-1109, :387-519, :656-683)
        'net.blocks.0.0.prenorm.transform.0.scale',
        'net.blocks.0.0.prenorm.transform.1.scale',
        'net.blocks.0.0.attn.to_q.weights.0',
        'net.blocks.0.0.attn.to_q.weights.1',
        'net.blocks.0.0.attn.to_v.kernel_unary.(0,1).rp.net.6.weight',
        'net.blocks.0.0.attn.to_k.kernel_unary.(1,1).rp.net.3.weight',
        'net.blocks.0.0.attn.to_out.weights.0',
        'net.blocks.0.0.attn.null_keys.0', 'net.blocks.0.0.attn.null_values.1',
        'net.blocks.0.0.attn.to_self_k.weights.0',
        'net.blocks.0.0.attn.to_self_v.weights.1',
        # feed-forward block (reference :347-383)
        'net.blocks.1.1.prenorm.transform.0.scale',
        'net.blocks.1.1.feedforward.project_in.weights.0',
        'net.blocks.1.1.feedforward.nonlin.transform.1.scale',
        'net.blocks.1.1.feedforward.project_out.weights.1',
        # output heads (reference :1113-1122)
        'conv_out.kernel_unary.(1,1).rp.net.6.weight',
        'norm.transform.0.scale',
        'linear_out.weights.0', 'linear_out.weights.1',
    ]
    missing = [n for n in expected if n not in names]
    assert not missing, f'missing state-dict entries: {missing}'
    # shape spot checks (radial net: Linear(edge_dim+1 -> 128); edge_dim = 3 + adj 4 = 7)
    sd = model.state_dict()
    assert sd['conv_in.kernel_unary.(0,0).rp.net.0.weight'].shape == (128, 8)
    assert sd['conv_in.kernel_unary.(0,1).rp.net.6.weight'].shape == (1 * 8 * 8, 128)
    assert sd['net.blocks.0.0.attn.to_q.weights.0'].shape == (8, 2 * 4)
    assert sd['net.blocks.0.0.attn.null_keys.1'].shape == (2, 4, 3)
    assert sd['linear_out.weights.1'].shape == (8, 1)


def test_reversible_state_dict_layout():
    model = SE3Transformer(dim=8, depth=1, num_degrees=2, num_neighbors=4,
                           heads=2, dim_head=4, attend_self=True,
                           output_degrees=2, reversible=True)
    names = _names(model)
    for n in [
        # reversible blocks wrap f/g in Deterministic -> .f.net / .g.net
        # (reference reversible.py:93-97, 200-203)
        'net.blocks.0.f.net.attn.to_q.weights.0',
        'net.blocks.0.f.net.prenorm.transform.1.scale',
        'net.blocks.0.g.net.feedforward.project_in.weights.0',
        # reversible forces norm_out (reference :1115)
        'norm.transform.0.scale',
    ]:
        assert n in names, n


def test_egnn_state_dict_layout():
    model = SE3Transformer(dim=8, depth=2, num_degrees=2, num_neighbors=4,
                           output_degrees=2, use_egnn=True, egnn_feedforward=True)
    names = _names(model)
    for n in [
        # EGNN trunk (reference :867-884, :707-762)
        'net.layers.0.0.node_norm.weight', 'net.layers.0.0.node_norm.bias',
        'net.layers.0.0.edge_mlp.0.weight', 'net.layers.0.0.edge_mlp.2.weight',
        'net.layers.0.0.htype_norms.1.scale', 'net.layers.0.0.htype_norms.1.bias',
        'net.layers.0.0.htype_gating.1.weight',
        'net.layers.0.0.htypes_mlp.0.weight', 'net.layers.0.0.htypes_mlp.2.weight',
        'net.layers.0.0.node_mlp.0.weight', 'net.layers.0.0.node_mlp.2.weight',
        'net.layers.1.1.feedforward.project_in.weights.0',
    ]:
        assert n in names, n


def test_gated_scale_state_dict():
    model = SE3Transformer(dim=8, depth=1, num_degrees=2, num_neighbors=4,
                           output_degrees=2, norm_gated_scale=True)
    names = _names(model)
    assert 'net.blocks.0.0.prenorm.transform.0.w_gate' in names
    assert 'net.blocks.0.0.prenorm.transform.0.scale' not in names


def test_rotary_inv_freq_buffer():
    model = SE3Transformer(dim=8, depth=1, num_degrees=2, num_neighbors=4,
                           dim_head=8, output_degrees=2,
                           rotary_position=True, rotary_rel_dist=True)
    assert 'rotary_pos_emb.inv_freq' in _names(model)
    assert model.state_dict()['rotary_pos_emb.inv_freq'].shape == (2,)


def test_checkpoint_roundtrip(tmp_path):
    import torch
    from se3_transformer_amd import SE3Transformer
    torch.manual_seed(0)
    m1 = SE3Transformer(dim=16, heads=2, dim_head=8, depth=1, num_degrees=2,
                        num_neighbors=4)
    feats = torch.randn(1, 12, 16)
    coors = torch.randn(1, 12, 3)
    mask = torch.ones(1, 12, dtype=torch.bool)
    ref = m1(feats, coors, mask, return_type=0)
    torch.save(m1.state_dict(), tmp_path / 'ckpt.pt')

    torch.manual_seed(123)
    m2 = SE3Transformer(dim=16, heads=2, dim_head=8, depth=1, num_degrees=2,
                        num_neighbors=4)
    missing, unexpected = m2.load_state_dict(
        torch.load(tmp_path / 'ckpt.pt', weights_only=True))
    assert not missing and not unexpected
    out = m2(feats, coors, mask, return_type=0)
    assert torch.allclose(out, ref, atol=1e-6)
