// Backward kernels for the fused pairwise convolution (see pairconv.hip).
//
//   R[e,n]  = H[e,:] . W[n,:] + bias[n],   n = (mo, c),  c = (mi,f) "urow"
//   out[e,mo,o] = sum_c R[e,(mo,c)] * u[c,o,e]
//
// Gradients (g = dL/dout, (E,mo,O) bf16):
//   dR[e,n]   = sum_o g[e,mo,o] * u[c,o,e]          (VALU, recomputed on the fly)
//   dH[e,k]   = sum_n dR[e,n] * W[n,k]              (kernel B1, MFMA, needs W^T)
//   dW[n,k]   = sum_e dR[e,n] * H[e,k]              (kernel B2, MFMA, needs H^T)
//   db[n]     = sum_e dR[e,n]                       (folded into B2)
//   du[c,o,e] = sum_mo R[e,(mo,c)] * g[e,mo,o]      (kernel B3, recomputes R)
//
// dR / R are never written to global memory anywhere.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define KDIM 128
#define NT 512

__device__ __forceinline__ float b2f(__bf16 x) { return (float)x; }

typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

__device__ __forceinline__ f32x2 b2f2(bf16x2 v) {
    return f32x2{(float)v[0], (float)v[1]};
}

// ---------------------------------------------------------------------------
// B1: dH[e,k] = sum_n dR[e,n] W[n,k]
// block: 64 e x 128 k, loops (mo-block 8) x (urow-chunk 32) over all n.
// Wt is W transposed: (128, mo*miF) bf16, so the MFMA B-operand
// (8 consecutive n at fixed k) is a contiguous 16B load.
// ---------------------------------------------------------------------------
template <int O>
__global__ void __launch_bounds__(NT)
pairconv_bwd_dh_kernel(const __bf16* __restrict__ Gt,  // (mo, O, E) bf16
                       const __bf16* __restrict__ Ut,  // (miF, O, E) bf16
                       const __bf16* __restrict__ P1,  // packed W: [mo/8][miF/32][wk2][kf4][ns8][lane64][8]
                       float* __restrict__ dH,         // (E, 128) f32 (zeroed)
                       int E, int mo, int miF, int nsplit, int nmemb, int coh) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    // u/g tiles are [row][e][o PADDED to 8]: the dR build then reads ONE
    // 16B vector per (row, e) and contracts with v_dot2_f32_bf16 — no
    // scalar LDS reads, no bf16->f32 conversion ops.
    __bf16* dr_lds = reinterpret_cast<__bf16*>(smem);                 // [64e][256n] 32 KiB
    __bf16* u_lds = reinterpret_cast<__bf16*>(smem + 32768);          // [32][64][8] 32 KiB
    __bf16* g_lds = reinterpret_cast<__bf16*>(smem + 65536);          // [8][64][8] 8 KiB

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;
    const int we = wid >> 1;          // 0..3: e-group of 16
    const int wk = wid & 1;           // 0..1: k-group of 64
    // L3-panel mapping (see pairconv.hip): PS split-slices x PE e-blocks
    const int PS = 4, PE = 128;
    int panels_x = (nmemb + PE - 1) / PE;
    int within = blockIdx.x % (PS * PE);
    int panel = blockIdx.x / (PS * PE);
    int eb = (panel % panels_x) * PE + within % PE;
    int sp = (panel / panels_x) * PS + within / PE;
    if (eb >= nmemb || sp >= nsplit) return;
    const int e0 = eb * 64;

    f32x4 acc[4];                      // 16 e x 64 k per wave
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

    const int nmo = mo / 8, nuc = miF / 32;
    const int mb_lo = (nmo / nsplit) * sp;
    const int mb_hi = mb_lo + nmo / nsplit;

    // T14 register staging: next g tile (every mb) and a slice of the next
    // u chunk (every cb) load under the dR + MFMA phases.
    constexpr int GTOT = (8 * O * 64) / 8;     // g tile, 16B units (<= NT)
    constexpr int UTOTd = (32 * O * 64) / 8;   // u chunk, 16B units
    constexpr int UUD = 2;                     // staged u units per thread
    bf16x8 g_reg, u_reg[UUD];
    auto load_g = [&](int mb) {
        if (tid < GTOT) {
            int ro = tid >> 3, eu = (tid & 7) * 8;   // ro = m*O+o
            const __bf16* src = Gt + ((size_t)(mb * 8 + ro / O) * O + (ro % O)) * E + e0 + eu;
            if (e0 + eu + 8 <= E) g_reg = *reinterpret_cast<const bf16x8*>(src);
            else {
                bf16x8 v(0);
                for (int j = 0; j < 8; ++j) if (e0 + eu + j < E) v[j] = src[j];
                g_reg = v;
            }
        }
    };
    auto load_u = [&](int cb) {
#pragma unroll
        for (int t = 0; t < UUD; ++t) {
            int i = tid + t * NT;
            if (i < UTOTd) {
                int ro = i >> 3, eu = (i & 7) * 8;
                const __bf16* src = Ut + ((size_t)(cb * 32 + ro / O) * O + (ro % O)) * E + e0 + eu;
                if (e0 + eu + 8 <= E) u_reg[t] = *reinterpret_cast<const bf16x8*>(src);
                else {
                    bf16x8 v(0);
                    for (int j = 0; j < 8; ++j) if (e0 + eu + j < E) v[j] = src[j];
                    u_reg[t] = v;
                }
            }
        }
    };
    load_u(0);
    load_g(mb_lo);
    // zero the o-pad lanes once (staging never overwrites o in [O, 8))
    for (int i = tid; i < (32 + 8) * 64; i += NT)
        *reinterpret_cast<bf16x8*>(u_lds + (size_t)i * 8) = bf16x8(0);

    for (int cb = 0; cb < nuc; ++cb) {
        // commit the staged u chunk, scattered to [urow][e][o(pad 8)]
        __syncthreads();
#pragma unroll
        for (int t = 0; t < UUD; ++t) {
            int i = tid + t * NT;
            if (i < UTOTd) {
                int ro = i >> 3, eu = (i & 7) * 8;
                const int ur = ro / O, o = ro % O;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    u_lds[((size_t)ur * 64 + eu + j) * 8 + o] = u_reg[t][j];
            }
        }
        for (int i = tid + UUD * NT; i < UTOTd; i += NT) {   // unstaged tail
            int ro = i >> 3, eu = (i & 7) * 8;
            const __bf16* src = Ut + ((size_t)(cb * 32 + ro / O) * O + (ro % O)) * E + e0 + eu;
            bf16x8 v;
            if (e0 + eu + 8 <= E) v = *reinterpret_cast<const bf16x8*>(src);
            else { for (int j = 0; j < 8; ++j) v[j] = (e0 + eu + j < E) ? src[j] : (__bf16)0.f; }
            const int ur = ro / O, o = ro % O;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                u_lds[((size_t)ur * 64 + eu + j) * 8 + o] = v[j];
        }
        if (cb + 1 < nuc) load_u(cb + 1);
        for (int mb = mb_lo; mb < mb_hi; ++mb) {
            // commit the staged g tile, scattered to [m][e][o(pad 8)]
            if (tid < GTOT) {
                int ro = tid >> 3, eu = (tid & 7) * 8;
                const int mr = ro / O, o = ro % O;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    g_lds[((size_t)mr * 64 + eu + j) * 8 + o] = g_reg[j];
            }
            load_g(mb + 1 < mb_hi ? mb + 1 : mb_lo);   // next mb (or next cb's first)
            __syncthreads();
            // cooperative dR tile [64e][256n]: thread owns (e, run of 8 n);
            // per n: one 16B g vector (shared over the run) dot one 16B u
            // vector via 4 v_dot2_f32_bf16; one swizzled b128 write per run.
            for (int i = tid; i < (64 * 256) / 8; i += NT) {
                int e = i & 63, n8 = i >> 6;
                int m = n8 >> 2;                      // (n8*8)>>5
                bf16x8 g8 = *reinterpret_cast<const bf16x8*>(
                    g_lds + ((size_t)m * 64 + e) * 8);
                const bf16x2* g2 = reinterpret_cast<const bf16x2*>(&g8);
                __bf16 v0[8];
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    int c = (n8 * 8 + j) & 31;
                    bf16x8 u8 = *reinterpret_cast<const bf16x8*>(
                        u_lds + ((size_t)c * 64 + e) * 8);
                    const bf16x2* u2 = reinterpret_cast<const bf16x2*>(&u8);
                    float acc = 0.f;
#pragma unroll
                    for (int p = 0; p < 4; ++p)
                        acc = __builtin_amdgcn_fdot2_f32_bf16(g2[p], u2[p], acc, false);
                    v0[j] = (__bf16)acc;
                }
                *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(dr_lds)
                    + e * 512 + (((n8 ^ (e & 15)) << 4))) = *reinterpret_cast<bf16x8*>(v0);
            }
            __syncthreads();
            // MFMA: dH_tile += dR(64e x 256n) @ W(256n x 128k)
            const __bf16* pbase = P1 + ((((size_t)mb * (miF / 32) + cb) * 2 + wk) * 4) * 8 * 64 * 8
                                  + (size_t)lane * 8;
#pragma unroll
            for (int ns = 0; ns < 8; ++ns) {      // 8 n-steps of 32
                // A: dR rows e = we*16+l15, n = ns*32 + l4*8.. (mo-major inside tile)
                int ntile = ns * 32 + l4 * 8;     // 0..255 (this is (m*32+c) packed)
                int e_row = we * 16 + l15;
                bf16x8 a = *reinterpret_cast<const bf16x8*>(
                    reinterpret_cast<char*>(dr_lds) + e_row * 512
                    + ((((ntile >> 3) ^ (e_row & 15)) << 4)));
                // careful: dR tile n index = m*32+c, global n = (mb*8+m)*miF + cb*32 + c
#pragma unroll
                for (int kf = 0; kf < 4; ++kf) {
                    bf16x8 b = *reinterpret_cast<const bf16x8*>(
                        pbase + ((size_t)kf * 8 + ns) * 64 * 8);
                    acc[kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(b, a, acc[kf], 0, 0, 0);
                }
            }
        }
    }
    // D layout from mfma(b, a): D[row=k-ish?]; operands: A-arg=b (k rows), B-arg=a (e cols)
    // mfma(X, Y, acc): D[i][j] = sum_k X[i][k(contraction)] ... X supplies M rows.
    // Here X = b (Wt fragment: rows k, contraction n), Y = a (dR: contraction n, cols e)
    // => D rows = k, cols = e: row = l4*4+reg (k), col = l15 (e).
    {
        for (int kf = 0; kf < 4; ++kf) {
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                int k = wk * 64 + kf * 16 + l4 * 4 + reg;
                int e = e0 + we * 16 + l15;
                if (e < E) atomicAdd(&dH[(size_t)e * KDIM + k], acc[kf][reg]);
            }
        }
    }
}

// ---------------------------------------------------------------------------
// B2: dW[n,k] = sum_e dR[e,n] H[e,k];  db[n] = sum_e dR[e,n]
// block: n-tile 128 (4 mo x 32 urow) x 128 k, loops e in chunks of 32.
// Ht is H transposed: (128, E) bf16.
// ---------------------------------------------------------------------------
template <int O>
__global__ void __launch_bounds__(NT)
pairconv_bwd_dw_kernel(const __bf16* __restrict__ Gt,  // (mo, O, E)
                       const __bf16* __restrict__ Ut,  // (miF, O, E)
                       const __bf16* __restrict__ Ht,  // (128, E)
                       float* __restrict__ dW,         // (mo*miF, 128) f32
                       int E, int mo, int miF) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    // NOTE: dw keeps the e-pair packed-fma dR build — the o-padded dot2
    // layout (as in dh) was tried and measured SLOWER here (234 -> 187
    // TF/s): dw stages per 32-EDGE chunk, so the scatter commit the padded
    // layout needs runs every chunk and dominates.
    __bf16* dr_lds = reinterpret_cast<__bf16*>(smem);                  // [128n][32e] 8 KiB
    __bf16* h_lds = reinterpret_cast<__bf16*>(smem + 8192);            // [128k][32e] 8 KiB
    __bf16* u_lds = reinterpret_cast<__bf16*>(smem + 16384);           // [32][O][32]
    __bf16* g_lds = reinterpret_cast<__bf16*>(smem + 16384 + 32 * O * 32 * 2); // [4][O][32]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;
    const int wn = wid >> 1;          // 0..3: n-group of 32
    const int wk = wid & 1;           // 0..1: k-group of 64
    // n-tile: blockIdx.x over N/128; tile covers mo-block of 4, urow-chunk of 32
    const int mb = blockIdx.x / (miF / 32);     // mo-block (4 mo each)
    const int cb = blockIdx.x % (miF / 32);     // urow chunk

    f32x4 acc[2][4];                  // wave: 32 n x 64 k
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // T14 register-staged pipeline: chunk ec+1's global loads are issued
    // right after chunk ec's LDS image is written, so HBM/L2 latency hides
    // under the dR + MFMA phase instead of serializing each chunk.
    constexpr int UU = (32 * O * 32 / 8 + NT - 1) / NT;   // u 16B units/thread
    constexpr int GU = 1;                                  // g units/thread
    bf16x8 u_reg[UU], g_reg[GU], h_reg;
    const int nec = (E + 31) / 32;

    auto load_chunk = [&](int ec) {
        const int e0 = ec * 32;
#pragma unroll
        for (int t = 0; t < UU; ++t) {
            int i = tid + t * NT;
            u_reg[t] = bf16x8(0);
            if (i < (32 * O * 32) / 8) {
                int ro = i >> 2, eu = (i & 3) * 8;
                const __bf16* src = Ut + ((size_t)(cb * 32 + ro / O) * O + (ro % O)) * E + e0 + eu;
                if (e0 + eu + 8 <= E) u_reg[t] = *reinterpret_cast<const bf16x8*>(src);
                else { for (int j = 0; j < 8; ++j) u_reg[t][j] = (e0 + eu + j < E) ? src[j] : (__bf16)0.f; }
            }
        }
        {
            int i = tid;
            g_reg[0] = bf16x8(0);
            if (i < (4 * O * 32) / 8) {
                int ro = i >> 2, eu = (i & 3) * 8;
                const __bf16* src = Gt + ((size_t)(mb * 4 + ro / O) * O + (ro % O)) * E + e0 + eu;
                if (e0 + eu + 8 <= E) g_reg[0] = *reinterpret_cast<const bf16x8*>(src);
                else { for (int j = 0; j < 8; ++j) g_reg[0][j] = (e0 + eu + j < E) ? src[j] : (__bf16)0.f; }
            }
        }
        {
            int k = tid >> 2, eu = (tid & 3) * 8;
            const __bf16* src = Ht + (size_t)k * E + e0 + eu;
            h_reg = bf16x8(0);
            if (e0 + eu + 8 <= E) h_reg = *reinterpret_cast<const bf16x8*>(src);
            else { for (int j = 0; j < 8; ++j) h_reg[j] = (e0 + eu + j < E) ? src[j] : (__bf16)0.f; }
        }
    };

    load_chunk(0);
    for (int ec = 0; ec < nec; ++ec) {
        __syncthreads();   // previous MFMA done reading the LDS images
        // write the staged registers for chunk ec
#pragma unroll
        for (int t = 0; t < UU; ++t) {
            int i = tid + t * NT;
            if (i < (32 * O * 32) / 8) {
                int ro = i >> 2, eu = (i & 3) * 8;
                *reinterpret_cast<bf16x8*>(u_lds + (size_t)ro * 32 + eu) = u_reg[t];
            }
        }
        if (tid < (4 * O * 32) / 8) {
            int ro = tid >> 2, eu = (tid & 3) * 8;
            *reinterpret_cast<bf16x8*>(g_lds + (size_t)ro * 32 + eu) = g_reg[0];
        }
        {
            int k = tid >> 2, eu = (tid & 3) * 8;
            *reinterpret_cast<bf16x8*>(h_lds + (size_t)k * 32 + eu) = h_reg;
        }
        if (ec + 1 < nec) load_chunk(ec + 1);   // issue next loads early
        __syncthreads();
        // cooperative dR^T tile [128n][32e] (e-pairs, packed)
        for (int i = tid; i < (128 * 32) / 2; i += NT) {
            int e = (i & 15) * 2, n = i >> 4;
            int m = n >> 5, c = n & 31;
            f32x2 acc2 = {0.f, 0.f};
#pragma unroll
            for (int o = 0; o < O; ++o) {
                f32x2 gv = b2f2(*reinterpret_cast<const bf16x2*>(g_lds + (m * O + o) * 32 + e));
                f32x2 uv = b2f2(*reinterpret_cast<const bf16x2*>(u_lds + (c * O + o) * 32 + e));
                acc2[0] = fmaf(gv[0], uv[0], acc2[0]);
                acc2[1] = fmaf(gv[1], uv[1], acc2[1]);
            }
            *reinterpret_cast<bf16x2*>(dr_lds + (size_t)n * 32 + e) =
                bf16x2{(__bf16)acc2[0], (__bf16)acc2[1]};
        }
        __syncthreads();
        // MFMA: dW_tile += dR^T(128n x 32e) @ H(32e x 128k)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                dr_lds + (size_t)(wn * 32 + nf * 16 + l15) * 32 + l4 * 8);
#pragma unroll
            for (int kf = 0; kf < 4; ++kf) {
                bf16x8 b = *reinterpret_cast<const bf16x8*>(
                    h_lds + (size_t)(wk * 64 + kf * 16 + l15) * 32 + l4 * 8);
                acc[nf][kf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nf][kf], 0, 0, 0);
            }
        }
    }
    __syncthreads();
    // write dW: D rows = n ((l4*4+reg within frag)), cols = k (l15)
    const size_t nbase = (size_t)(mb * 4) * miF + cb * 32;
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
        for (int kf = 0; kf < 4; ++kf) {
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                int ntile = wn * 32 + nf * 16 + l4 * 4 + reg;
                int m = ntile >> 5, c = ntile & 31;
                int k = wk * 64 + kf * 16 + l15;
                dW[(nbase + (size_t)m * miF + c) * KDIM + k] = acc[nf][kf][reg];
            }
        }
    }
}

// ---------------------------------------------------------------------------
// B3: du[c,o,e] = sum_mo R[e,(mo,c)] g[e,mo,o],  R = H @ W^T + bias
// block: 64 e x 32 urow, loops mo in blocks of 8; R tile recomputed by MFMA
// exactly as the forward kernel, bounced through LDS, contracted vs g.
// ---------------------------------------------------------------------------
template <int O, int WP>   // WP: prefetch next mo-block's W frags under the
                           // contraction phase (see pairconv.hip fwd)
__global__ void __launch_bounds__(NT, 4)   // keep 2 blocks/CU under WP
pairconv_bwd_du_kernel(const __bf16* __restrict__ H,   // (E,128)
                       const __bf16* __restrict__ P,   // packed W as forward
                       const float* __restrict__ bias, // (mo*miF,)
                       const __bf16* __restrict__ Gt,  // (mo,O,E)
                       float* __restrict__ dU,         // (miF, O, E) f32
                       int E, int mo, int miF, int nmemb, int coh) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __bf16* h_lds = reinterpret_cast<__bf16*>(smem);                   // [64][128] swizzled 16K
    __bf16* r_lds = reinterpret_cast<__bf16*>(smem + 16384);           // [256n][64e] 32K
    __bf16* g_lds = reinterpret_cast<__bf16*>(smem + 49152);           // [8][O][64]
    float* du_acc = reinterpret_cast<float*>(smem + 49152 + ((8 * O * 64 * 2 + 15) & ~15)); // [32][O][64]
    float* bias_lds = reinterpret_cast<float*>(
        smem + 49152 + ((8 * O * 64 * 2 + 15) & ~15) + 32 * O * 64 * 4);  // [256]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;
    const int wm = wid >> 1;
    const int we = wid & 1;
    // cohort mapping: concurrently-resident blocks on one XCD share a urow
    // chunk (the same packed-W column slice stays L2-resident).
    int eb, cb;
    if (coh) {
        int x = blockIdx.x & 7, r = blockIdx.x >> 3;
        eb = r % nmemb;
        cb = x + 8 * (r / nmemb);
    } else {
        eb = blockIdx.x % nmemb;
        cb = blockIdx.x / nmemb;
    }
    const int e0 = eb * 64;
    const int uc0 = cb * 32;          // urow chunk

    // stage H tile swizzled (as forward)
    for (int i = tid; i < (64 * KDIM) / 8; i += NT) {
        int e = i >> 4, k16 = i & 15;
        int dst = e * 256 + ((k16 * 16) ^ ((e & 15) << 4));
        bf16x8 v;
        if (e0 + e < E) v = *reinterpret_cast<const bf16x8*>(H + (size_t)(e0 + e) * KDIM + k16 * 8);
        else v = bf16x8(0);
        *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(h_lds) + dst) = v;
    }
    for (int i = tid; i < 32 * O * 64; i += NT) du_acc[i] = 0.f;

    // T14 register staging: the next mo-block's g tile + bias chunk load
    // while the current block's MFMA + contraction run (g tile fits one
    // 16B unit per thread for O <= 7).
    constexpr int GTOT = (8 * O * 64) / 8;   // g tile in 16B units (<= NT)
    bf16x8 g_reg;
    float bias_reg;
    auto load_gb = [&](int mb) {
        if (tid < GTOT) {
            int ro = tid >> 3, eu = (tid & 7) * 8;
            const __bf16* src = Gt + ((size_t)(mb * 8 + ro / O) * O + (ro % O)) * E + e0 + eu;
            if (e0 + eu + 8 <= E) g_reg = *reinterpret_cast<const bf16x8*>(src);
            else {
                bf16x8 v(0);
                for (int j = 0; j < 8; ++j) if (e0 + eu + j < E) v[j] = src[j];
                g_reg = v;
            }
        }
        if (tid < 256) {
            int m = tid >> 5, c = tid & 31;
            bias_reg = bias[(size_t)(mb * 8 + m) * miF + uc0 + c];
        }
    };
    load_gb(0);

    const int nmo = mo / 8;
    auto pb_of = [&](int mbb) {
        return P + ((((size_t)mbb * (miF / 32) + cb) * 4 + wm) * 4) * 4 * 64 * 8
               + (size_t)lane * 8;
    };
    bf16x8 a_pre[WP ? 16 : 1];
    if (WP) {
#pragma unroll
        for (int f = 0; f < 16; ++f)
            a_pre[f] = *reinterpret_cast<const bf16x8*>(pb_of(0) + (size_t)f * 512);
    }
    __syncthreads();

    for (int mb = 0; mb < nmo; ++mb) {
        // commit the staged g tile [8][O][64] + bias chunk [256]
        if (tid < GTOT) {
            int ro = tid >> 3, eu = (tid & 7) * 8;
            *reinterpret_cast<bf16x8*>(g_lds + (size_t)ro * 64 + eu) = g_reg;
        }
        if (tid < 256) bias_lds[tid] = bias_reg;
        if (mb + 1 < nmo) load_gb(mb + 1);   // issue next block's loads early
        __syncthreads();
        // MFMA R tile: (256n x 64e) like forward
        f32x4 acc[4][2];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int ef = 0; ef < 2; ++ef) acc[mf][ef] = {0.f, 0.f, 0.f, 0.f};
        const __bf16* pbase = pb_of(mb);
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
        if (WP) {   // issue next mo-block's fragment loads under the
                    // R-bounce + contraction phases below
            const __bf16* pn = pb_of(mb + 1 < nmo ? mb + 1 : mb);
#pragma unroll
            for (int f = 0; f < 16; ++f)
                a_pre[f] = *reinterpret_cast<const bf16x8*>(pn + (size_t)f * 512);
        }
        // bounce R (+bias) to LDS [256n][64e]
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int ef = 0; ef < 2; ++ef)
#pragma unroll
                for (int reg = 0; reg < 4; ++reg) {
                    int r = wm * 64 + mf * 16 + l4 * 4 + reg;
                    int e = we * 32 + ef * 16 + l15;
                    r_lds[(size_t)r * 64 + e] = (__bf16)(acc[mf][ef][reg] + bias_lds[r]);
                }
        __syncthreads();
        // contraction: du_acc[c][o][e] += sum_m R[(m,c)][e] * g[m][o][e]
        // partition (c,e) across all 512 threads: 32*64 = 2048 cells, 4 per thread
        for (int i = tid; i < 32 * 64; i += NT) {
            int e = i & 63, c = i >> 6;
#pragma unroll
            for (int o = 0; o < O; ++o) {
                float s = du_acc[(c * O + o) * 64 + e];
#pragma unroll
                for (int m = 0; m < 8; ++m)
                    s = fmaf(b2f(r_lds[(size_t)(m * 32 + c) * 64 + e]),
                             b2f(g_lds[(m * O + o) * 64 + e]), s);
                du_acc[(c * O + o) * 64 + e] = s;
            }
        }
        __syncthreads();
    }
    // write du chunk
    for (int i = tid; i < 32 * O * 64; i += NT) {
        int e = i & 63, ro = i >> 6;  // c*O+o
        if (e0 + e < E)
            dU[((size_t)(uc0 + ro / O) * O + (ro % O)) * E + e0 + e] = du_acc[i];
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
#define DISPATCH_O(O, ...)                                         \
    switch (O) {                                                   \
        case 1: { constexpr int kO = 1; __VA_ARGS__; break; }      \
        case 3: { constexpr int kO = 3; __VA_ARGS__; break; }      \
        case 5: { constexpr int kO = 5; __VA_ARGS__; break; }      \
        case 7: { constexpr int kO = 7; __VA_ARGS__; break; }      \
        default: TORCH_CHECK(false, "unsupported O ", O);          \
    }

void pairconv_bwd_dh(torch::Tensor Gt, torch::Tensor Ut, torch::Tensor Wt,
                     torch::Tensor dH, int64_t mo_) {
    int E = dH.size(0), mo = (int)mo_, miF = Ut.size(0), O = Ut.size(1);
    TORCH_CHECK(Gt.is_contiguous() && Ut.is_contiguous() && Wt.is_contiguous() && dH.is_contiguous());
    TORCH_CHECK(Wt.numel() == (int64_t)mo * miF * KDIM, "expect packed W (P1)");
    auto stream = at::cuda::getCurrentHIPStream();
    int eblk = (E + 63) / 64;
    int nmo = mo / 8;
    int nsplit = (nmo % 16 == 0) ? 16 : ((nmo % 8 == 0) ? 8 : 1);
    while (eblk * nsplit * 2 <= 1024 && nsplit * 2 <= nmo && nmo % (nsplit * 2) == 0)
        nsplit *= 2;
    int coh = 0;
    const int PS = 4, PE = 128;
    int panels = ((eblk + PE - 1) / PE) * ((nsplit + PS - 1) / PS);
    dim3 grid((long)panels * PS * PE);
    DISPATCH_O(O, {
        size_t lds = 32768 + 32768 + 8192;   // dr | u[32][64][8] | g[8][64][8]
        hipLaunchKernelGGL(HIP_KERNEL_NAME(pairconv_bwd_dh_kernel<kO>), grid, dim3(NT), lds, stream,
                           reinterpret_cast<const __bf16*>(Gt.data_ptr()),
                           reinterpret_cast<const __bf16*>(Ut.data_ptr()),
                           reinterpret_cast<const __bf16*>(Wt.data_ptr()),
                           dH.data_ptr<float>(), E, mo, miF, nsplit, eblk, coh);
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "bwd_dh: ", hipGetErrorString(err));
}

void pairconv_bwd_dw(torch::Tensor Gt, torch::Tensor Ut, torch::Tensor Ht,
                     torch::Tensor dW, int64_t mo_) {
    int E = Ht.size(1), mo = (int)mo_, miF = Ut.size(0), O = Ut.size(1);
    TORCH_CHECK(Gt.is_contiguous() && Ut.is_contiguous() && Ht.is_contiguous() &&
                dW.is_contiguous());
    TORCH_CHECK(mo % 4 == 0 && miF % 32 == 0);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((mo / 4) * (miF / 32));
    DISPATCH_O(O, {
        size_t lds = 16384 + (size_t)32 * kO * 32 * 2 + (size_t)4 * kO * 32 * 2;
        hipLaunchKernelGGL(HIP_KERNEL_NAME(pairconv_bwd_dw_kernel<kO>), grid, dim3(NT), lds, stream,
                           reinterpret_cast<const __bf16*>(Gt.data_ptr()),
                           reinterpret_cast<const __bf16*>(Ut.data_ptr()),
                           reinterpret_cast<const __bf16*>(Ht.data_ptr()),
                           dW.data_ptr<float>(), E, mo, miF);
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "bwd_dw: ", hipGetErrorString(err));
}

void pairconv_bwd_du(torch::Tensor H, torch::Tensor W, torch::Tensor bias,
                     torch::Tensor Gt, torch::Tensor dU, int64_t mo_) {
    int E = H.size(0), mo = (int)mo_, miF = dU.size(0), O = dU.size(1);
    TORCH_CHECK(H.is_contiguous() && W.is_contiguous() && Gt.is_contiguous() &&
                bias.is_contiguous() && dU.is_contiguous());
    TORCH_CHECK(W.numel() == (int64_t)mo * miF * KDIM, "expect packed W (P)");
    TORCH_CHECK(bias.dtype() == torch::kFloat32);
    auto stream = at::cuda::getCurrentHIPStream();
    int nmemb = (E + 63) / 64;
    int ncb = miF / 32;
    int coh = (ncb % 8 == 0) ? 1 : 0;
    dim3 grid(nmemb * ncb);
    // WP off by default here: du's contraction phase holds more live state
    // than fwd's epilogue, so the prefetch spills 116-240 B/lane under the
    // 2-blocks/CU cap (vs fwd's 24-36 B at O=3/5). SE3_DU_WP=1 for A/B.
    const char* wp_env = getenv("SE3_DU_WP");
    DISPATCH_O(O, {
        int wp = wp_env ? atoi(wp_env) : 0;
        size_t lds = 49152 + (size_t)((8 * kO * 64 * 2 + 15) & ~15) +
                     (size_t)32 * kO * 64 * 4 + 256 * 4;
        if (wp) {
            hipLaunchKernelGGL(HIP_KERNEL_NAME(pairconv_bwd_du_kernel<kO, 1>),
                               grid, dim3(NT), lds, stream,
                               reinterpret_cast<const __bf16*>(H.data_ptr()),
                               reinterpret_cast<const __bf16*>(W.data_ptr()),
                               bias.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(Gt.data_ptr()),
                               dU.data_ptr<float>(), E, mo, miF, nmemb, coh);
        } else {
            hipLaunchKernelGGL(HIP_KERNEL_NAME(pairconv_bwd_du_kernel<kO, 0>),
                               grid, dim3(NT), lds, stream,
                               reinterpret_cast<const __bf16*>(H.data_ptr()),
                               reinterpret_cast<const __bf16*>(W.data_ptr()),
                               bias.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(Gt.data_ptr()),
                               dU.data_ptr<float>(), E, mo, miF, nmemb, coh);
        }
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "bwd_du: ", hipGetErrorString(err));
}
