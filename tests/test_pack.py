"""CPU checks that the weight-packing permutations match the HIP kernels'
per-lane fragment indexing (pairconv.hip / pairconv_bwd.hip)."""
import torch

from se3_transformer_amd.ops.fused import _pack_w_dh, _pack_w_fwd


def test_pack_w_fwd_layout():
    mo, miF, K = 16, 64, 128
    W = torch.arange(mo * miF * K, dtype=torch.float32).view(mo * miF, K)
    P = _pack_w_fwd(W, mo, miF).reshape(-1)
    miFcb = miF // 32
    for mob in range(mo // 8):
        for cb in range(miFcb):
            for wm in range(4):
                for mf in range(4):
                    for kit in range(4):
                        for lane in (0, 17, 63):
                            l15, l4 = lane & 15, lane >> 4
                            r = wm * 64 + mf * 16 + l15
                            n = (mob * 8 + (r >> 5)) * miF + cb * 32 + (r & 31)
                            k = kit * 32 + l4 * 8
                            base = (((((mob * miFcb + cb) * 4 + wm) * 4 + mf) * 4 + kit) * 64 + lane) * 8
                            got = P[base:base + 8]
                            want = W[n, k:k + 8]
                            assert torch.equal(got, want), (mob, cb, wm, mf, kit, lane)


def test_pack_w_dh_layout():
    mo, miF, K = 16, 64, 128
    W = torch.arange(mo * miF * K, dtype=torch.float32).view(mo * miF, K)
    P = _pack_w_dh(W, mo, miF).reshape(-1)
    miFcb = miF // 32
    for mob in range(mo // 8):
        for cb in range(miFcb):
            for wk in range(2):
                for kf in range(4):
                    for ns in range(8):
                        for lane in (0, 17, 63):
                            l15, l4 = lane & 15, lane >> 4
                            # B-frag: k = wk*64+kf*16+l15, n-run = ns*32 + l4*8 + j
                            n = (mob * 8 + ns) * miF + cb * 32 + l4 * 8
                            k = wk * 64 + kf * 16 + l15
                            base = (((((mob * miFcb + cb) * 2 + wk) * 4 + kf) * 8 + ns) * 64 + lane) * 8
                            got = P[base:base + 8]
                            want = W[n:n + 8, k]
                            assert torch.equal(got, want), (mob, cb, wk, kf, ns, lane)
