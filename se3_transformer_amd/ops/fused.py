"""Fused-kernel dispatch layer (MI355X HIP extension, se3_transformer_amd._C).

Autograd Functions wrapping the hand-written CDNA4 kernels in csrc/:
pairwise convolution (fwd + the three dedicated backward kernels + the
one-pass dual-layout weight pack — the per-edge radial output R of
reference se3_transformer_pytorch.py:297-343 never touches HBM in either
direction), the radial trunk, the basis×features precontraction, NormSE3,
neighbor attention v2 (online softmax, in-kernel rotary, HIP backward)
and the EGNN higher-type ops. A chunked library-GEMM backward for the
pairconv remains behind SE3_TORCH_BWD as the parity oracle.
"""
from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from se3_transformer_amd import _C as ext
            _EXT = ext
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = e
    return _EXT


def ext_available() -> bool:
    return _load_ext() is not None


def require_ext():
    """On a CUDA/ROCm device the HIP extension must be present — fail loudly
    rather than silently falling back to eager."""
    if not ext_available():
        raise RuntimeError(
            'se3_transformer_amd._C HIP extension is not built but a GPU is '
            'present; run `python scripts/build_ext.py` '
            f'(import error: {_EXT_ERR})')


def fused_shapes_ok(mo: int, miF: int, O: int, mid_dim: int) -> bool:
    return mid_dim == 128 and miF % 32 == 0 and mo % 8 == 0 and O in (1, 3, 5, 7)


def _pack_w_fwd(W16, mo, miF):
    """Rearrange W (mo*miF, 128) bf16 into the forward/du kernels' per-lane
    fragment order [mo/8][miF/32][wm4][mf4][kit4][lane64][j8] so each wave's
    MFMA A-fragment load is one contiguous 1 KiB read."""
    v = W16.view(mo // 8, 4, 2, miF // 32, 2, 16, 4, 4, 8)
    return v.permute(0, 3, 1, 2, 4, 6, 7, 5, 8).contiguous()


def _pack_w_dh(W16, mo, miF):
    """Fragment order for the dH kernel's B-operand:
    [mo/8][miF/32][wk2][kf4][ns8][lane64][j8]."""
    v = W16.view(mo // 8, 8, miF // 32, 4, 8, 2, 4, 16)
    return v.permute(0, 2, 5, 6, 1, 3, 7, 4).contiguous()


class _FusedPairConv(torch.autograd.Function):
    """out[e,mo,o] = sum_{c,h} (H[e,h] W[(mo,c),h] + bias[(mo,c)]) * Ut[c,o,e]

    H  (E,128) bf16 | W (mo*miF,128) fp32/bf16 param | bias (mo*miF,) fp32
    Ut (miF,O,E) bf16 | returns (E, mo, O) fp32.
    """

    @staticmethod
    def forward(ctx, H, W, bias, Ut, mo):
        ext = _load_ext()
        E = H.shape[0]
        miF, O, _ = Ut.shape
        H16 = H.contiguous().to(torch.bfloat16)
        Ut16 = Ut.contiguous().to(torch.bfloat16)
        # bias term: out0[e,mo,o] = sum_c bias[mo,c] Ut[c,o,e]
        b16 = bias.detach().to(torch.bfloat16).view(mo, miF)
        out = (b16 @ Ut16.reshape(miF, O * E)).view(mo, O, E) \
            .permute(2, 0, 1).contiguous().float()
        hip_bwd = os.environ.get('SE3_TORCH_BWD') != '1'
        # SE3_LOWMEM_PACK=1 trades ~6% step time for memory: save the
        # torch-layout bf16 W (1x) and re-pack in backward, instead of
        # holding both packed fragment layouts (2x) through autograd —
        # frees ~36 GB at the headline 18.2B-param config (enables larger
        # neighbor counts before the activation-memory wall).
        lowmem = os.environ.get('SE3_LOWMEM_PACK') == '1'
        if hip_bwd and not lowmem and hasattr(ext, 'pack_w_both'):
            # one-pass pack kernel: W read once, both fragment layouts
            # written; saved for backward so nothing re-packs per step
            Wd = W.detach().contiguous()
            n128 = mo * miF * 128
            Pf = torch.empty(n128, dtype=torch.bfloat16, device=Wd.device)
            Pdh = torch.empty(n128, dtype=torch.bfloat16, device=Wd.device)
            ext.pack_w_both(Wd, Pf, Pdh, mo)
            ext.pairconv_fwd(H16, Pf, Ut16, out, mo)
            ctx.save_for_backward(H16, Pf, Pdh, Ut16, b16)
            ctx.packed = True
        else:
            W16 = W.detach().contiguous().to(torch.bfloat16)
            ext.pairconv_fwd(H16, _pack_w_fwd(W16, mo, miF), Ut16, out, mo)
            ctx.save_for_backward(H16, W16, Ut16, b16)
            ctx.packed = False
        ctx.mo = mo
        ctx.w_dtype = W.dtype
        ctx.b_dtype = bias.dtype
        return out

    @staticmethod
    def backward(ctx, grad_out):
        mo = ctx.mo
        if ctx.packed:
            H16, Pf, Pdh, Ut16, b16 = ctx.saved_tensors
        else:
            H16, W16, Ut16, b16 = ctx.saved_tensors
        E, K = H16.shape
        miF, O, _ = Ut16.shape
        g = grad_out.contiguous()                      # (E, mo, O) fp32
        g16 = g.to(torch.bfloat16)

        need_H, need_W, need_b, need_u = ctx.needs_input_grad[:4]

        ext = _load_ext()
        if ext is not None and os.environ.get('SE3_TORCH_BWD') != '1':
            dH = dW = db = dUt = None
            g_t = g16.permute(1, 2, 0).contiguous()    # (mo, O, E) bf16
            P1 = Pu = None
            if ctx.packed:
                P1, Pu = Pdh, Pf
            elif (need_H or need_u) and hasattr(ext, 'pack_w_both'):
                # SE3_LOWMEM_PACK path: re-derive both fragment layouts in
                # one kernel pass from the saved torch-layout W
                n128 = mo * miF * K
                Pu = torch.empty(n128, dtype=torch.bfloat16,
                                 device=H16.device)
                P1 = torch.empty_like(Pu)
                ext.pack_w_both(W16, Pu, P1, mo)
            if need_H:
                if P1 is None:
                    P1 = _pack_w_dh(W16, mo, miF)
                dH = torch.zeros(E, K, dtype=torch.float32, device=H16.device)
                ext.pairconv_bwd_dh(g_t, Ut16, P1, dH, mo)
            P1 = None
            if need_W:
                Ht = H16.t().contiguous()              # (128, E)
                dW = torch.empty(mo * miF, K, dtype=torch.float32,
                                 device=H16.device)
                ext.pairconv_bwd_dw(g_t, Ut16, Ht, dW, mo)
                dW = dW.to(ctx.w_dtype)
                del Ht
            if need_b:
                # db[(m,c)] = sum_{e,o} g[e,m,o] u[c,o,e] — one GEMM, both
                # operands are free (mo,O*E)/(miF,O*E) views
                db = (g_t.reshape(mo, O * E) @ Ut16.reshape(miF, O * E).t()) \
                    .reshape(mo * miF).to(ctx.b_dtype)
            if need_u:
                dUt = torch.empty(miF, O, E, dtype=torch.float32,
                                  device=H16.device)
                if Pu is None:
                    Pu = _pack_w_fwd(W16, mo, miF)
                ext.pairconv_bwd_du(H16, Pu, b16.float().reshape(-1),
                                    g_t, dUt, mo)
            return dH, dW, db, dUt, None

        if ctx.packed:   # pragma: no cover - env toggled between fwd and bwd
            raise RuntimeError('SE3_TORCH_BWD enabled after a packed forward; '
                               'set it before the forward pass')
        u_eco = Ut16.permute(2, 0, 1)                  # (E, miF, O) view
        dH = dW = db = dUt = None
        if need_H:
            dH = torch.zeros(E, K, dtype=torch.float32, device=H16.device)
        if need_W:
            dW = torch.empty(mo * miF, K, dtype=torch.float32, device=H16.device)
        if need_b:
            db = torch.empty(mo * miF, dtype=torch.float32, device=H16.device)
        if need_u:
            dUt = torch.zeros(miF, O, E, dtype=torch.float32, device=H16.device)

        # chunk over mo so the dR slab stays ~<= 1 GiB
        per_mo = E * miF * 2
        cm = max(8, min(mo, (1 << 30) // max(per_mo, 1) // 8 * 8))
        for m0 in range(0, mo, cm):
            m1 = min(mo, m0 + cm)
            # dR[e,(m,c)] = sum_o g[e,m,o] * u[e,c,o]
            dR16 = torch.einsum('emo,eco->emc', g16[:, m0:m1], u_eco) \
                .reshape(E, (m1 - m0) * miF)
            Wslab = W16[m0 * miF:m1 * miF]             # ((m1-m0)*miF, K)
            if need_H:
                dH += (dR16 @ Wslab).float()
            if need_W:
                dW[m0 * miF:m1 * miF] = (dR16.t() @ H16).float()
            if need_b:
                db[m0 * miF:m1 * miF] = dR16.float().sum(dim=0)
            if need_u:
                # R[e,(m,c)] = H @ W^T + bias ; du[c,o,e] += sum_m R * g
                R = (H16 @ Wslab.t()).view(E, m1 - m0, miF).float() \
                    + b16[m0:m1].float().unsqueeze(0)
                dUt += torch.einsum('emc,emo->coe', R, g[:, m0:m1])
            del dR16

        if need_W:
            dW = dW.to(ctx.w_dtype)
        if need_b:
            db = db.to(ctx.b_dtype)
        return dH, dW, db, dUt, None


def fused_pairconv(H, W, bias, Ut, mo):
    return _FusedPairConv.apply(H, W, bias, Ut, mo)


class _NormSE3Fn(torch.autograd.Function):
    """Fused equivariant norm-gate (csrc/norm_se3.hip), scale path."""

    @staticmethod
    def forward(ctx, t, scale, eps):
        ext = _load_ext()
        s32 = scale.detach().reshape(-1).float().contiguous()
        out = torch.empty_like(t)
        ext.norm_se3_fwd(t, s32, out, eps)
        ctx.save_for_backward(t, s32)
        ctx.eps = eps
        ctx.scale_shape = scale.shape
        ctx.scale_dtype = scale.dtype
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _load_ext()
        t, s32 = ctx.saved_tensors
        dt = torch.empty_like(t)
        dscale = torch.zeros_like(s32)
        ext.norm_se3_bwd(t, s32, dout.contiguous(), dt, dscale, ctx.eps)
        return dt, dscale.view(ctx.scale_shape).to(ctx.scale_dtype), None


def norm_se3(t, scale, eps):
    return _NormSE3Fn.apply(t, scale, eps)


class _AttnFn(torch.autograd.Function):
    """Fused neighbor attention v2 (csrc/attn2.hip): rows (b,h,i), keys in
    64-wide online-softmax tiles (J unbounded), rotary embeddings applied to
    q/k/v in-kernel on degree 0, HIP backward via the saved logsumexp.
    qf/kf are the RAW rotary frequency tables (or None); gradients do not
    flow into them (the wrapper gates on their requires_grad)."""

    @staticmethod
    def forward(ctx, q, k, v, mask_u8, qf, kf, n, heads, scale, kv_one=False):
        ext = _load_ext()
        R, DM = q.shape
        dev = q.device
        out = torch.empty(R, DM, dtype=torch.float32, device=dev)
        lse = torch.empty(R, dtype=torch.float32, device=dev)
        m = mask_u8 if mask_u8 is not None else \
            torch.empty(0, dtype=torch.uint8, device=dev)
        qf_ = qf if qf is not None else torch.empty(0, device=dev)
        kf_ = kf if kf is not None else torch.empty(0, device=dev)
        jr = kf.shape[1] if kf is not None else 0
        rot = (qf.shape[-1] if qf is not None
               else (kf.shape[-1] if kf is not None else 0))
        ext.attn2_fwd(q, k, v, m, qf_, kf_, out, lse,
                      n, heads, scale, kv_one, jr, rot)
        ctx.save_for_backward(q, k, v, mask_u8, qf, kf, out, lse)
        ctx.dims = (n, heads, scale, kv_one, jr, rot)
        return out

    @staticmethod
    def backward(ctx, g):
        ext = _load_ext()
        q, k, v, mask_u8, qf, kf, out, lse = ctx.saved_tensors
        n, heads, scale, kv_one, jr, rot = ctx.dims
        R, DM = q.shape
        dev = q.device
        m = mask_u8 if mask_u8 is not None else \
            torch.empty(0, dtype=torch.uint8, device=dev)
        qf_ = qf if qf is not None else torch.empty(0, device=dev)
        kf_ = kf if kf is not None else torch.empty(0, device=dev)
        dq = torch.empty(R, DM, dtype=torch.float32, device=dev)
        if kv_one:
            # heads share KV rows: the kernel accumulates atomically
            dk = torch.zeros(k.shape, dtype=torch.float32, device=dev)
            dv = torch.zeros(v.shape, dtype=torch.float32, device=dev)
        else:
            dk = torch.empty(k.shape, dtype=torch.float32, device=dev)
            dv = torch.empty(v.shape, dtype=torch.float32, device=dev)
        ext.attn2_bwd(q, k, v, m, qf_, kf_, out, lse,
                      g.contiguous().float(), dq, dk, dv,
                      n, heads, scale, kv_one, jr, rot)
        return (dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype),
                None, None, None, None, None, None, None)


def fused_attention(q, k, v, mask_u8, n, heads, scale, kv_one=False,
                    qf=None, kf=None):
    return _AttnFn.apply(q, k, v, mask_u8, qf, kf, n, heads, scale, kv_one)


class _UBuildFn(torch.autograd.Function):
    """Basis-feature precontraction (csrc/ubuild.hip):
    Ut[(c f), o, e] = sum_i B[e,o,i,f] x[e,c,i], emitted directly in the
    e-contiguous layout the pairconv kernels consume. dB is not produced
    (the caller gates on basis.requires_grad)."""

    @staticmethod
    def forward(ctx, x, B, O, I, F):
        ext = _load_ext()
        E, C, _ = x.shape
        Ut = torch.empty(C * F, O, E, dtype=torch.bfloat16, device=x.device)
        ext.ubuild_fwd(B, x.contiguous(), Ut, O, I, F)
        ctx.save_for_backward(B)
        ctx.meta = (x.dtype, x.shape, O, I, F)
        return Ut

    @staticmethod
    def backward(ctx, dUt):
        ext = _load_ext()
        (B,) = ctx.saved_tensors
        dtype, (E, C, I_), O, I, F = ctx.meta
        dX = torch.empty(E, C, I, dtype=dtype, device=B.device)
        ext.ubuild_bwd_dx(B, dUt.contiguous().float(), dX, O, I, F)
        return dX, None, None, None, None


def ubuild_ok(B, C, O, I, F) -> bool:
    ext = _load_ext()
    return (ext is not None and hasattr(ext, 'ubuild_fwd')
            and B.dtype == torch.float32 and not B.requires_grad
            and C % 32 == 0 and O * I * F <= 343)


def ubuild(x, B, O, I, F):
    return _UBuildFn.apply(x, B, O, I, F)


class _HtypeRelDistFn(torch.autograd.Function):
    """EGNN neighbor rel-htype distances (csrc/egnn.hip): straight from the
    gathered neighbor indices — the O(n^2 d m) pairwise rel tensor of the
    eager path never exists."""

    @staticmethod
    def forward(ctx, ht, idx):
        ext = _load_ext()
        b, n, d, m = ht.shape
        k = idx.shape[2]
        dist = torch.empty(b, n, k, d, dtype=torch.float32, device=ht.device)
        htc = ht.contiguous()
        ext.egnn_rel_dist_fwd(htc, idx.contiguous(), dist)
        ctx.save_for_backward(htc, idx)
        return dist

    @staticmethod
    def backward(ctx, gdist):
        ext = _load_ext()
        ht, idx = ctx.saved_tensors
        dht = torch.zeros(ht.shape, dtype=torch.float32, device=ht.device)
        ext.egnn_rel_dist_bwd(ht, idx, gdist.contiguous().float(), dht)
        return dht.to(ht.dtype), None


def htype_rel_dist(ht, idx):
    return _HtypeRelDistFn.apply(ht, idx)


class _HtypeUpdateFn(torch.autograd.Function):
    """EGNN htype update (csrc/egnn.hip): HtypesNorm of the neighbor
    rel-htypes + weighted sum over neighbors, fused and gather-based."""

    @staticmethod
    def forward(ctx, ht, idx, w, scale, bias, eps):
        ext = _load_ext()
        b, n, d, m = ht.shape
        htc = ht.contiguous()
        wc = w.detach().float().contiguous()
        s32 = scale.detach().reshape(-1).float().contiguous()
        b32 = bias.detach().reshape(-1).float().contiguous()
        upd = torch.empty(b, n, d, m, dtype=torch.float32, device=ht.device)
        ext.egnn_htype_update_fwd(htc, idx.contiguous(), wc, s32, b32,
                                  upd, eps)
        ctx.save_for_backward(htc, idx, wc, s32, b32)
        ctx.meta = (eps, w.dtype, scale.shape, scale.dtype)
        return upd

    @staticmethod
    def backward(ctx, g):
        ext = _load_ext()
        ht, idx, wc, s32, b32 = ctx.saved_tensors
        eps, w_dtype, p_shape, p_dtype = ctx.meta
        dht = torch.zeros(ht.shape, dtype=torch.float32, device=ht.device)
        dw = torch.empty(wc.shape, dtype=torch.float32, device=ht.device)
        dsc = torch.zeros_like(s32)
        dbi = torch.zeros_like(b32)
        ext.egnn_htype_update_bwd(ht, idx, wc, s32, b32,
                                  g.contiguous().float(), dht, dw, dsc, dbi,
                                  eps)
        return (dht.to(ht.dtype), None, dw.to(w_dtype),
                dsc.view(p_shape).to(p_dtype), dbi.view(p_shape).to(p_dtype),
                None)


def htype_update(ht, idx, w, scale, bias, eps):
    return _HtypeUpdateFn.apply(ht, idx, w, scale, bias, eps)


def egnn_kernels_ok(m: int) -> bool:
    ext = _load_ext()
    return (ext is not None and hasattr(ext, 'egnn_htype_update_fwd')
            and m <= 7 and os.environ.get('SE3_EAGER_EGNN') != '1')


RADIAL_TRUNK_DIMS = (1, 2, 3, 5, 9, 17, 21, 25)   # instantiated in csrc/radial.hip


def radial_trunk_ok(in_dim: int, mid_dim: int) -> bool:
    ext = _load_ext()
    return (mid_dim == 128 and in_dim in RADIAL_TRUNK_DIMS
            and ext is not None and hasattr(ext, 'radial_trunk_fwd'))


class _RadialTrunkFn(torch.autograd.Function):
    """Fused RadialFunc trunk (csrc/radial.hip): Linear(D->128) + LN + GELU
    + Linear(128->128) + LN + GELU in one kernel each way. Mirrors the
    autocast-bf16 eager numerics (GEMMs bf16/fp32-acc, LN+GELU fp32)."""

    @staticmethod
    def forward(ctx, x, w0, b0, g0, be0, w3, b3, g3, be3, eps):
        ext = _load_ext()
        E, D = x.shape
        x16 = x.contiguous().to(torch.bfloat16)
        w0_16 = w0.detach().contiguous().to(torch.bfloat16)
        w3_16 = w3.detach().contiguous().to(torch.bfloat16)
        p0 = torch.cat([b0.detach().float(), g0.detach().float(),
                        be0.detach().float()]).contiguous()
        p3 = torch.cat([b3.detach().float(), g3.detach().float(),
                        be3.detach().float()]).contiguous()
        H = torch.empty(E, 128, dtype=torch.bfloat16, device=x.device)
        yh0 = torch.empty_like(H)
        yh3 = torch.empty_like(H)
        rs = torch.empty(2, E, dtype=torch.float32, device=x.device)
        ext.radial_trunk_fwd(x16, w0_16, p0, w3_16, p3, H, yh0, yh3, rs, eps)
        ctx.save_for_backward(x16, w0_16, p0, w3_16, p3, yh0, yh3, rs)
        ctx.eps = eps
        ctx.x_dtype = x.dtype
        return H

    @staticmethod
    def backward(ctx, dH):
        ext = _load_ext()
        x16, w0_16, p0, w3_16, p3, yh0, yh3, rs = ctx.saved_tensors
        E, D = x16.shape
        dev = x16.device
        dH16 = dH.contiguous().to(torch.bfloat16)
        dW0 = torch.zeros(128, D, dtype=torch.float32, device=dev)
        dp0 = torch.zeros(3 * 128, dtype=torch.float32, device=dev)
        dW3 = torch.zeros(128, 128, dtype=torch.float32, device=dev)
        dp3 = torch.zeros(3 * 128, dtype=torch.float32, device=dev)
        need_x = ctx.needs_input_grad[0]
        dX = (torch.zeros(E, D, dtype=torch.float32, device=dev)
              if need_x else torch.empty(0, device=dev))
        w3t = w3_16.t().contiguous()
        ext.radial_trunk_bwd(dH16, x16, w0_16, p0, w3_16, w3t, p3,
                             yh0, yh3, rs, dW0, dp0, dW3, dp3, dX, ctx.eps)
        return ((dX.to(ctx.x_dtype) if need_x else None),
                dW0, dp0[:128], dp0[128:256], dp0[256:],
                dW3, dp3[:128], dp3[128:256], dp3[256:], None)


def radial_trunk(x, w0, b0, g0, be0, w3, b3, g3, be3, eps=1e-5):
    return _RadialTrunkFn.apply(x, w0, b0, g0, be0, w3, b3, g3, be3, eps)
