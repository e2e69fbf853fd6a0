// Fused pairwise-convolution forward for the SE(3)-Transformer TFN layer.
//
// Reference computation (se3_transformer_pytorch.py:301-343 + :251): per edge e,
//   R[e, mo, mi, f] = radial_net(edge_feats)   (the net.6 GEMM: H[e,:] @ W)
//   K[e] = sum_f R[...,f] * B[...,f]           (per-edge kernel matrix)
//   out  = K[e] @ x_gathered[e]                (matvec)
// materializing R (E x mo*mi*F, ~67 GB at the headline config) and K per edge.
//
// MI355X-native restructure: precontract u[e, (mi,f), o] = sum_i B[e,o,i,f] x[e,mi,i]
// (small), then
//   out[e, mo, o] += sum_{mi,f} (H[e,:] . W[:, (mo,mi,f)]) * u[e, (mi,f), o]
// i.e. a bf16 MFMA GEMM over (edges x radial-output-columns) with the
// (mi,f)->o contraction fused into the epilogue. R never touches HBM.
// The bias term sum_c bias[mo,c] u[e,c,o] is pre-added into `out` by the host.
//
// Layouts:
//   H   (E, 128)      bf16   radial trunk activations (k-contiguous)
//   W   (mo*miF, 128) bf16   net.6 weight, torch Linear layout (out,in)
//   Ut  (miF, O, E)   bf16   basis-contracted features, e-contiguous
//   out (E, mo, O)    f32    pre-initialized with the bias term; kernel adds
//
// Tile: block = 512 threads (8 waves) owns (64 edges) x (8 mo); loops over
// miF in chunks of 32; per chunk an MFMA GEMM of (256 n-rows x 64 e-cols, K=128)
// with n = 8mo x 32urow, then a VALU contraction against u_lds into
// per-lane register accumulators (the cross-lane l4 reduce is deferred to
// after the chunk loop). Each (e, mo) output is owned by exactly one block:
// no atomics anywhere.
//
// Software pipeline (round 2): the next chunk's u tile is loaded into
// registers while the current chunk's MFMA + epilogue run, so the
// u-stage's HBM/L2 latency hides under compute instead of serializing
// each chunk (the round-1 PMC analysis measured ~3.5k-cycle wave stalls
// per chunk from the serialized u-stage -> W-frags round trips).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BLK_E 64
#define BLK_MO 8
#define UCHUNK 32
#define KDIM 128
#define NTHREADS 512

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

// MB2 = mo-blocks (of 8) processed per workgroup against ONE staged u chunk.
// MB2=2 halves the u-chunk HBM traffic (u is re-read per mo-block otherwise)
// while keeping 2-blocks/CU residency: the two sub-blocks run serially, so
// the MFMA accumulator registers are reused, only the LDS `part` doubles
// (O=7: 16K h + 28K u + 28K part = 72 KiB <= 80 KiB per block).
// WP = W-fragment software pipeline: the next sub-block's 16 A-fragments
// are prefetched into registers during the current sub-block's epilogue
// (where the MFMA accumulators are dead), so each MFMA phase starts with
// its operands already in flight instead of paying a fresh L2/HBM round
// trip (T14 applied to the W stream).
// ES = epilogue split: process the padded o range in two passes with half
// the persistent accumulators (frees VGPRs for the WP prefetch at O>=5)
template <int O, int UU, int MB2, int WP, int ES>   // UU = 16B u-units register-staged per thread
__global__ void __launch_bounds__(NTHREADS, 4)   // cap VGPR<=128: 2 blocks/CU
pairconv_fwd_kernel(const __bf16* __restrict__ H,
                    const __bf16* __restrict__ P,   // packed W: [mo/8][miF/32][wm4][mf4][kit4][lane64][8]
                    const __bf16* __restrict__ Ut,
                    float* __restrict__ out,
                    int E, int mo, int miF, int nmemb, int coh) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    // carve: H tile | u chunk | partial accumulator. The u chunk is stored
    // [urow][e][o] with o PADDED to 8 so the epilogue reads one 16B vector
    // per (urow, e) instead of O scalar ds_read_u16s — the epilogue VALU/LDS
    // issue rate was co-limiting with the MFMA pipe (measured ~240 TF/s vs
    // the ~380 TF/s W-stream roofline).
    __bf16* h_lds = reinterpret_cast<__bf16*>(smem);                       // [64][128] swizzled, 16 KiB
    __bf16* u_lds = reinterpret_cast<__bf16*>(smem + 16384);               // [32][64][8] 32 KiB
    float* part = reinterpret_cast<float*>(smem + 16384 + UCHUNK * BLK_E * 8 * 2); // [64][8*MB2][O]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;          // 0..7
    const int wm = wid >> 1;           // 0..3  (n-rows wm*64 .. +64)
    const int we = wid & 1;            // 0..1  (e-cols we*32 .. +32)
    const int l15 = lane & 15;
    const int l4 = lane >> 4;          // 0..3

    // cohort mapping: all concurrently-resident blocks on one XCD share the
    // same mo-block (=> the same packed-W slice stays in that XCD's L2).
    int eb, mb;
    if (coh) {
        int x = blockIdx.x & 7, r = blockIdx.x >> 3;
        eb = r % nmemb;
        mb = x + 8 * (r / nmemb);
    } else {
        eb = blockIdx.x % nmemb;
        mb = blockIdx.x / nmemb;
    }
    const int e0 = eb * BLK_E;
    const int mo0 = mb * BLK_MO * MB2;

    // ---- stage H tile (64 x 128 bf16), XOR-swizzled 16B slots within each row
    {
        for (int i = tid; i < (BLK_E * KDIM) / 8; i += NTHREADS) {  // 16B units
            int e = i >> 4;             // 16 units per row
            int k16 = i & 15;           // 16B slot
            int dst = e * 256 + ((k16 * 16) ^ ((e & 15) << 4));
            bf16x8 v;
            if (e0 + e < E) {
                v = *reinterpret_cast<const bf16x8*>(H + (size_t)(e0 + e) * KDIM + k16 * 8);
            } else {
                v = bf16x8(0);
            }
            *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(h_lds) + dst) = v;
        }
        // zero partial accumulator + the u pad lanes (o in [O, 8) are never
        // overwritten by staging, so one upfront clear keeps them zero)
        for (int i = tid; i < BLK_E * BLK_MO * MB2 * O; i += NTHREADS) part[i] = 0.f;
        for (int i = tid; i < UCHUNK * BLK_E * 8 / 8; i += NTHREADS)
            *reinterpret_cast<bf16x8*>(u_lds + (size_t)i * 8) = bf16x8(0);
    }

    constexpr int UTOT = (UCHUNK * O * BLK_E) / 8;   // u chunk in 16B units
    bf16x8 u_reg[UU > 0 ? UU : 1];
    auto load_u = [&](int c) {
        const int uc0 = c * UCHUNK;
#pragma unroll
        for (int t = 0; t < UU; ++t) {
            int i = tid + t * NTHREADS;
            if (i < UTOT) {
                int ro = i >> 3;            // (urow*O + o)
                int eu = (i & 7) * 8;       // e offset within 64
                const __bf16* src = Ut + ((size_t)(uc0 + (ro / O)) * O + (ro % O)) * E + e0 + eu;
                if (e0 + eu + 8 <= E) {
                    u_reg[t] = *reinterpret_cast<const bf16x8*>(src);
                } else {
                    bf16x8 v(0);
                    for (int j = 0; j < 8; ++j)
                        if (e0 + eu + j < E) v[j] = src[j];
                    u_reg[t] = v;
                }
            }
        }
    };

    load_u(0);

    const int nchunks = miF / UCHUNK;
    // W-pipeline state: the 16 A-fragments of the NEXT (chunk, sub) pair
    auto pb_of = [&](int cc, int ss) {
        return P + ((((size_t)(mb * MB2 + ss) * (miF / 32) + cc) * 4 + wm) * 4) * 4 * 64 * 8
               + (size_t)lane * 8;
    };
    bf16x8 a_pre[WP ? 16 : 1];
    if (WP) {
#pragma unroll
        for (int f = 0; f < 16; ++f)
            a_pre[f] = *reinterpret_cast<const bf16x8*>(pb_of(0, 0) + (size_t)f * 512);
    }
    __syncthreads();

    for (int c = 0; c < nchunks; ++c) {
        // ---- commit the staged u chunk, scattered to [urow][e][o(pad 8)]
#pragma unroll
        for (int t = 0; t < UU; ++t) {
            int i = tid + t * NTHREADS;
            if (i < UTOT) {
                int ro = i >> 3, eu = (i & 7) * 8;
                const int ur = ro / O, o = ro % O;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    u_lds[((size_t)ur * BLK_E + eu + j) * 8 + o] = u_reg[t][j];
            }
        }
        // tail units beyond the register budget: direct load (UU*NT >= UTOT
        // for the shipped instantiations — loop compiles away)
        for (int i = tid + UU * NTHREADS; i < UTOT; i += NTHREADS) {
            int ro = i >> 3, eu = (i & 7) * 8;
            const __bf16* src = Ut + ((size_t)(c * UCHUNK + (ro / O)) * O + (ro % O)) * E + e0 + eu;
            bf16x8 v;
            if (e0 + eu + 8 <= E) {
                v = *reinterpret_cast<const bf16x8*>(src);
            } else {
                v = bf16x8(0);
                for (int j = 0; j < 8; ++j)
                    if (e0 + eu + j < E) v[j] = src[j];
            }
            const int ur = ro / O, o = ro % O;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                u_lds[((size_t)ur * BLK_E + eu + j) * 8 + o] = v[j];
        }
        if (c + 1 < nchunks) load_u(c + 1);   // issue next chunk's loads early
        __syncthreads();

        // ---- per mo-sub-block: GEMM + epilogue against the SAME u chunk.
        // Unrolled, but a sched_barrier between the sub-blocks keeps the
        // second sub-block's W-fragment loads from being hoisted into the
        // first (which would double the live A-operand registers and spill).
#pragma unroll
        for (int sub = 0; sub < MB2; ++sub) {
        if (MB2 > 1 && sub > 0) __builtin_amdgcn_sched_barrier(0);
        // ---- GEMM: R^T tile (256 n x 64 e), K=128
        f32x4 acc[4][2];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int ef = 0; ef < 2; ++ef) acc[mf][ef] = f32x4(0.f);

        const __bf16* pbase = pb_of(c, sub);
#pragma unroll
        for (int kit = 0; kit < 4; ++kit) {
            const int k0 = kit * 32 + l4 * 8;
            bf16x8 a[4], b[2];
#pragma unroll
            for (int mf = 0; mf < 4; ++mf) {
                a[mf] = WP ? a_pre[mf * 4 + kit]
                           : *reinterpret_cast<const bf16x8*>(
                                 pbase + ((size_t)mf * 4 + kit) * 64 * 8);
            }
#pragma unroll
            for (int ef = 0; ef < 2; ++ef) {
                int e = we * 32 + ef * 16 + l15;
                int byte = e * 256 + ((k0 * 2) ^ ((e & 15) << 4));
                b[ef] = *reinterpret_cast<const bf16x8*>(reinterpret_cast<char*>(h_lds) + byte);
            }
#pragma unroll
            for (int mf = 0; mf < 4; ++mf)
#pragma unroll
                for (int ef = 0; ef < 2; ++ef)
                    acc[mf][ef] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mf], b[ef], acc[mf][ef], 0, 0, 0);
        }
        if (WP) {
            // prefetch the NEXT sub-block/chunk's fragments now — their
            // L2/HBM latency hides under the epilogue below (acc is the
            // only other register consumer left, and it dies there)
            int ns = sub + 1, nc2 = c;
            if (ns == MB2) { ns = 0; ++nc2; }
            if (nc2 == nchunks) { ns = sub; nc2 = c; }   // tail: benign refetch
            const __bf16* pn = pb_of(nc2, ns);
#pragma unroll
            for (int f = 0; f < 16; ++f)
                a_pre[f] = *reinterpret_cast<const bf16x8*>(pn + (size_t)f * 512);
        }

        // ---- epilogue: contract acc against u_lds into s[ef][moi][o-pairs].
        // One ds_read_b128 per (row, e) covers all 8 padded o; the o-pair
        // float2 accumulation maps onto v_pk_fma_f32. With ES (epilogue
        // split), the o range is processed in NH sequential halves with
        // only PH o-pairs of accumulators live (re-reads u but halves the
        // persistent s footprint — funds WP at the big orders); a
        // sched_barrier keeps the halves from being re-merged.
        typedef __attribute__((ext_vector_type(2))) float f32x2e;
        constexpr int PH = ES ? 2 : 4;   // o-pairs held per pass
        constexpr int NH = 4 / PH;
#pragma unroll
        for (int oh = 0; oh < NH; ++oh) {
        if (NH > 1 && oh > 0) __builtin_amdgcn_sched_barrier(0);
        f32x2e s[2][2][PH];
#pragma unroll
        for (int ef = 0; ef < 2; ++ef)
#pragma unroll
            for (int mi_ = 0; mi_ < 2; ++mi_)
#pragma unroll
                for (int p_ = 0; p_ < PH; ++p_) s[ef][mi_][p_] = f32x2e{0.f, 0.f};

#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {   // mf and mf+2 share urow (rows r, r+32)
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                const int r = wm * 64 + mf * 16 + l4 * 4 + reg;
                const int urow = r & 31;
#pragma unroll
                for (int ef = 0; ef < 2; ++ef) {
                    const int e = we * 32 + ef * 16 + l15;
                    const float rv0 = acc[mf][ef][reg];
                    const float rv1 = acc[mf + 2][ef][reg];
                    bf16x8 uv8 = *reinterpret_cast<const bf16x8*>(
                        u_lds + ((size_t)urow * BLK_E + e) * 8);
#pragma unroll
                    for (int p_ = 0; p_ < PH; ++p_) {
                        const int pp = oh * PH + p_;
                        f32x2e u2 = {(float)uv8[2 * pp], (float)uv8[2 * pp + 1]};
                        s[ef][0][p_] += rv0 * u2;
                        s[ef][1][p_] += rv1 * u2;
                    }
                }
            }
        }

        // cross-lane reduce over l4 groups (rows), then accumulate into LDS partial
#pragma unroll
        for (int ef = 0; ef < 2; ++ef)
#pragma unroll
            for (int mi_ = 0; mi_ < 2; ++mi_)
#pragma unroll
                for (int p_ = 0; p_ < PH; ++p_) {
#pragma unroll
                    for (int c2 = 0; c2 < 2; ++c2) {
                        if (2 * (oh * PH + p_) + c2 >= O) break;
                        float v = s[ef][mi_][p_][c2];
                        v += __shfl_xor(v, 16);
                        v += __shfl_xor(v, 32);
                        s[ef][mi_][p_][c2] = v;
                    }
                }
        if (l4 == 0) {
#pragma unroll
            for (int ef = 0; ef < 2; ++ef) {
                const int e = we * 32 + ef * 16 + l15;
#pragma unroll
                for (int mi_ = 0; mi_ < 2; ++mi_) {
                    const int moi = sub * BLK_MO + wm * 2 + mi_;
#pragma unroll
                    for (int o = 2 * oh * PH;
                         o < 2 * (oh + 1) * PH && o < O; ++o) {
                        float* p = part + ((size_t)e * (BLK_MO * MB2) + moi) * O + o;
                        *p += s[ef][mi_][(o >> 1) - oh * PH][o & 1];
                    }
                }
            }
        }
        }   // oh
        }   // sub
        __syncthreads();
    }

    // ---- write out: out[e0+e][mo0+moi][o] += partial
    for (int i = tid; i < BLK_E * BLK_MO * MB2 * O; i += NTHREADS) {
        int o = i % O;
        int moi = (i / O) % (BLK_MO * MB2);
        int e = i / (O * BLK_MO * MB2);
        if (e0 + e < E) {
            float* p = out + ((size_t)(e0 + e) * mo + mo0 + moi) * O + o;
            *p += part[((size_t)e * (BLK_MO * MB2) + moi) * O + o];
        }
    }
}

template <int O>
static void launch_fwd(const torch::Tensor& H, const torch::Tensor& W,
                       const torch::Tensor& Ut, torch::Tensor& out,
                       int E, int mo, int miF) {
    int nmemb = (E + BLK_E - 1) / BLK_E;
    auto stream = at::cuda::getCurrentHIPStream();
    // UU: 16B u-units register-staged per thread for the next-chunk pipeline.
    // MEASURED NEGATIVE on MI355X for the hot O=7 shape (UU=0: 239.6 TF/s,
    // UU=1: 217, UU=2: 221, UU=4: 153 — the VGPR spill the staging forces
    // under the 2-blocks/CU cap costs more than the hidden u latency, and
    // the co-resident block already covers most of the stall). Default 0;
    // SE3_FWD_UU kept for A/B runs.
    const char* uu_env = getenv("SE3_FWD_UU");
    int uu_sel = uu_env ? atoi(uu_env) : 0;
    constexpr int UUfull = (UCHUNK * O * BLK_E / 8 + NTHREADS - 1) / NTHREADS;
    int uu = uu_sel < 0 ? 0 : uu_sel;
    if (uu > 2) uu = 2;
    if (uu > UUfull) uu = UUfull;
    // MB2: mo-blocks per workgroup sharing one staged u chunk (halves u
    // traffic). Default 2 where mo allows; SE3_FWD_MB2=1 forces the old
    // one-block grid for A/B runs.
    const char* mb_env = getenv("SE3_FWD_MB2");
    int mb2 = (mb_env ? atoi(mb_env) : 2);
    if (mb2 != 1 && mo % (BLK_MO * 2) != 0) mb2 = 1;
    if (mb2 != 1) mb2 = 2;
    // WP: W-fragment software pipeline (prefetch next sub-block's A-frags
    // under the epilogue). Only instantiated with UU=0. Measured on MI355X:
    // +7% at O=5 (347.9 -> 372 TF/s at the (2,2) shape), -9% at O=7 (the
    // held prefetch registers push the epilogue to ~108 B/lane spill), so
    // the default is per-O; SE3_FWD_WP overrides for A/B runs.
    const char* wp_env = getenv("SE3_FWD_WP");
    int wp = wp_env ? atoi(wp_env) : ((O == 3 || O == 5) ? 1 : 0);
    if (wp) uu = 0;
    // ES (epilogue split) — A/B knob for the large orders; see the kernel
    const char* es_env = getenv("SE3_FWD_ES");
    int es = es_env ? atoi(es_env) : 0;
    if (O < 5) es = 0;
    if (es) uu = 0;
    int ng = mo / (BLK_MO * mb2);
    int coh = (ng % 8 == 0) ? 1 : 0;
    dim3 grid(nmemb * ng);
    size_t lds = 16384 + (size_t)UCHUNK * BLK_E * 8 * 2
                 + (size_t)BLK_E * BLK_MO * mb2 * O * 4;
#define LAUNCH_FWD(UU, MB2, WPv, ESv)                                                        \
    hipLaunchKernelGGL(HIP_KERNEL_NAME(pairconv_fwd_kernel<O, UU, MB2, WPv, ESv>), grid,     \
                       dim3(NTHREADS), lds, stream,                                          \
                       reinterpret_cast<const __bf16*>(H.data_ptr()),                        \
                       reinterpret_cast<const __bf16*>(W.data_ptr()),                        \
                       reinterpret_cast<const __bf16*>(Ut.data_ptr()),                       \
                       out.data_ptr<float>(), E, mo, miF, nmemb, coh)  /* W arg = packed P */
    if (mb2 == 2) {
        if (es && wp) { LAUNCH_FWD(0, 2, 1, 1); }
        else if (es) { LAUNCH_FWD(0, 2, 0, 1); }
        else if (wp) { LAUNCH_FWD(0, 2, 1, 0); }
        else switch (uu) {
            case 0: LAUNCH_FWD(0, 2, 0, 0); break;
            case 1: LAUNCH_FWD(1, 2, 0, 0); break;
            default: LAUNCH_FWD(2, 2, 0, 0); break;
        }
    } else {
        if (es && wp) { LAUNCH_FWD(0, 1, 1, 1); }
        else if (es) { LAUNCH_FWD(0, 1, 0, 1); }
        else if (wp) { LAUNCH_FWD(0, 1, 1, 0); }
        else switch (uu) {
            case 0: LAUNCH_FWD(0, 1, 0, 0); break;
            case 1: LAUNCH_FWD(1, 1, 0, 0); break;
            default: LAUNCH_FWD(2, 1, 0, 0); break;
        }
    }
#undef LAUNCH_FWD
}

void pairconv_fwd(torch::Tensor H, torch::Tensor W, torch::Tensor Ut,
                  torch::Tensor out, int64_t mo_) {
    TORCH_CHECK(H.is_cuda() && W.is_cuda() && Ut.is_cuda() && out.is_cuda());
    TORCH_CHECK(H.dtype() == torch::kBFloat16 && W.dtype() == torch::kBFloat16 &&
                Ut.dtype() == torch::kBFloat16 && out.dtype() == torch::kFloat32);
    TORCH_CHECK(H.is_contiguous() && W.is_contiguous() && Ut.is_contiguous() &&
                out.is_contiguous());
    int E = H.size(0);
    int mo = (int)mo_;
    int miF = Ut.size(0);
    int O = Ut.size(1);
    TORCH_CHECK(H.size(1) == KDIM, "radial hidden dim must be 128");
    TORCH_CHECK(W.numel() == (int64_t)mo * miF * KDIM, "expect packed W");
    TORCH_CHECK(Ut.size(2) == E);
    TORCH_CHECK(out.size(0) == E && out.size(1) == mo && out.size(2) == O);
    TORCH_CHECK(miF % UCHUNK == 0, "miF must be a multiple of 32");
    TORCH_CHECK(mo % BLK_MO == 0, "mo must be a multiple of 8");
    switch (O) {
        case 1: launch_fwd<1>(H, W, Ut, out, E, mo, miF); break;
        case 3: launch_fwd<3>(H, W, Ut, out, E, mo, miF); break;
        case 5: launch_fwd<5>(H, W, Ut, out, E, mo, miF); break;
        case 7: launch_fwd<7>(H, W, Ut, out, E, mo, miF); break;
        default: TORCH_CHECK(false, "unsupported output order ", O);
    }
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "pairconv_fwd launch failed: ", hipGetErrorString(err));
}

void sh_basis_fwd(torch::Tensor rel, torch::Tensor qcat, torch::Tensor normtab,
                  torch::Tensor meta, torch::Tensor out, int64_t L);
void knn_graph(torch::Tensor coors, torch::Tensor nmask, torch::Tensor allow,
               torch::Tensor sparse, torch::Tensor idx,
               torch::Tensor dist, torch::Tensor rel, torch::Tensor m,
               int64_t k, double radius, bool causal);
void attn2_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
               torch::Tensor mask, torch::Tensor qf, torch::Tensor kf,
               torch::Tensor out, torch::Tensor lse,
               int64_t n, int64_t heads, double scale, bool kv_one,
               int64_t jr, int64_t rot);
void attn2_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
               torch::Tensor mask, torch::Tensor qf, torch::Tensor kf,
               torch::Tensor out, torch::Tensor lse, torch::Tensor g,
               torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
               int64_t n, int64_t heads, double scale, bool kv_one,
               int64_t jr, int64_t rot);
void norm_se3_fwd(torch::Tensor t, torch::Tensor scale, torch::Tensor out, double eps);
void norm_se3_bwd(torch::Tensor t, torch::Tensor scale, torch::Tensor dout,
                  torch::Tensor dt, torch::Tensor dscale, double eps);
void pairconv_bwd_dh(torch::Tensor G, torch::Tensor Ut, torch::Tensor Wt,
                     torch::Tensor dH, int64_t mo_);
void pairconv_bwd_dw(torch::Tensor G, torch::Tensor Ut, torch::Tensor Ht,
                     torch::Tensor dW, int64_t mo_);
void pairconv_bwd_du(torch::Tensor H, torch::Tensor W, torch::Tensor bias,
                     torch::Tensor G, torch::Tensor dU, int64_t mo_);
void pack_w_both(torch::Tensor W, torch::Tensor Pf, torch::Tensor Pdh, int64_t mo_);
void ubuild_fwd(torch::Tensor B, torch::Tensor X, torch::Tensor Ut,
                int64_t O, int64_t I, int64_t F);
void ubuild_bwd_dx(torch::Tensor B, torch::Tensor dU, torch::Tensor dX,
                   int64_t O, int64_t I, int64_t F);
void egnn_rel_dist_fwd(torch::Tensor ht, torch::Tensor idx, torch::Tensor dist);
void egnn_rel_dist_bwd(torch::Tensor ht, torch::Tensor idx, torch::Tensor gdist,
                       torch::Tensor dht);
void egnn_htype_update_fwd(torch::Tensor ht, torch::Tensor idx, torch::Tensor w,
                           torch::Tensor sc, torch::Tensor bi,
                           torch::Tensor upd, double eps);
void egnn_htype_update_bwd(torch::Tensor ht, torch::Tensor idx, torch::Tensor w,
                           torch::Tensor sc, torch::Tensor bi, torch::Tensor g,
                           torch::Tensor dht, torch::Tensor dw,
                           torch::Tensor dsc, torch::Tensor dbi, double eps);
void radial_trunk_fwd(torch::Tensor X, torch::Tensor W0, torch::Tensor p0,
                      torch::Tensor W3, torch::Tensor p3,
                      torch::Tensor H, torch::Tensor yh0, torch::Tensor yh3,
                      torch::Tensor rs01, double eps);
void radial_trunk_bwd(torch::Tensor dH, torch::Tensor X, torch::Tensor W0,
                      torch::Tensor p0, torch::Tensor W3, torch::Tensor W3t,
                      torch::Tensor p3, torch::Tensor yh0, torch::Tensor yh3,
                      torch::Tensor rs01, torch::Tensor dW0, torch::Tensor dp0,
                      torch::Tensor dW3, torch::Tensor dp3, torch::Tensor dX,
                      double eps);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("pairconv_fwd", &pairconv_fwd,
          "fused radial-GEMM + basis contraction forward (MI355X)");
    m.def("pairconv_bwd_dh", &pairconv_bwd_dh, "dH backward");
    m.def("pairconv_bwd_dw", &pairconv_bwd_dw, "dW backward");
    m.def("pairconv_bwd_du", &pairconv_bwd_du, "dU backward");
    m.def("knn_graph", &knn_graph, "on-device kNN graph build");
    m.def("attn2_fwd", &attn2_fwd,
          "fused neighbor attention fwd (online softmax, in-kernel rotary)");
    m.def("attn2_bwd", &attn2_bwd, "fused neighbor attention backward");
    m.def("norm_se3_fwd", &norm_se3_fwd, "fused NormSE3 forward");
    m.def("norm_se3_bwd", &norm_se3_bwd, "fused NormSE3 backward");
    m.def("sh_basis_fwd", &sh_basis_fwd,
          "fused spherical-harmonics + equivariant basis (MI355X)");
    m.def("pack_w_both", &pack_w_both,
          "one-pass pack of net.6 W into both MFMA fragment layouts");
    m.def("radial_trunk_fwd", &radial_trunk_fwd,
          "fused radial trunk (Linear-LN-GELU x2) forward");
    m.def("radial_trunk_bwd", &radial_trunk_bwd,
          "fused radial trunk backward");
    m.def("ubuild_fwd", &ubuild_fwd,
          "basis x features precontraction (e-contiguous out)");
    m.def("ubuild_bwd_dx", &ubuild_bwd_dx, "ubuild dX backward");
    m.def("egnn_rel_dist_fwd", &egnn_rel_dist_fwd,
          "EGNN neighbor rel-htype distances (no n^2 intermediate)");
    m.def("egnn_rel_dist_bwd", &egnn_rel_dist_bwd, "its backward");
    m.def("egnn_htype_update_fwd", &egnn_htype_update_fwd,
          "EGNN htype norm+weighted neighbor sum (no n^2 intermediate)");
    m.def("egnn_htype_update_bwd", &egnn_htype_update_bwd, "its backward");
}
