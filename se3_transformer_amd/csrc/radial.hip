// Fused radial-trunk kernels for RadialFunc (reference
// se3_transformer_pytorch.py:287-295): per edge,
//   y0 = X @ W0^T + b0        (D -> 128)
//   z0 = LN(y0) * g0 + be0;  a1 = GELU(z0)
//   y3 = a1 @ W3^T + b3       (128 -> 128, MFMA)
//   z3 = LN(y3) * g3 + be3;  H  = GELU(z3)
// replacing the eager 6-op chain (2 GEMMs + 2 LayerNorms + 2 GELUs and
// their intermediate HBM round trips) with one kernel each way. The final
// net.6 Linear is folded into the pairconv kernels (pairconv.hip).
//
// Numerics mirror autocast-bf16 eager: GEMMs in bf16 with fp32
// accumulation, LayerNorm/GELU in fp32 (torch autocasts LN to fp32).
//
// Saved for backward: yhat0, yhat3 (the LN-normalized activations, bf16)
// and rs0, rs3 (reciprocal stddevs, fp32). Everything else (z, a1, GELU
// derivatives) is recomputed on the fly — z = yhat * g + be needs no
// division, so a zero LN gamma cannot poison the recompute.
//
// Block = 32 edges x 128 features, 256 threads (4 waves). W0/W3 are read
// straight from global: one pair's W3 is 32 KiB and is L2-resident across
// all the pair's edge blocks, so LDS is spent on the per-edge tiles
// instead (keeps 4 blocks/CU).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define RT_NT 256
#define RT_E 32          // edges per block
#define RT_K 128         // trunk width (RadialFunc mid_dim)
#define RT_PITCH 132     // fp32 LDS row pitch (bank-spread)

__device__ __forceinline__ float rt_gelu(float x) {
    return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
}
__device__ __forceinline__ float rt_dgelu(float x) {
    const float phi = 0.5f * (1.f + erff(x * 0.70710678118654752f));
    const float pdf = 0.39894228040143268f * __expf(-0.5f * x * x);
    return phi + x * pdf;
}
__device__ __forceinline__ float rt_b2f(__bf16 x) { return (float)x; }

// lane-swizzled bf16 LDS row layout (matches pairconv h_lds): 16B slots
// XORed with the row parity so MFMA fragment reads are conflict-free
__device__ __forceinline__ int rt_swz(int e, int k) {
    return e * 256 + ((((k >> 3) ^ (e & 15)) << 4)) + (k & 7) * 2;  // byte
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------
template <int D>
__global__ void __launch_bounds__(RT_NT)
radial_trunk_fwd_kernel(const __bf16* __restrict__ X,    // (E, D)
                        const __bf16* __restrict__ W0,   // (128, D)
                        const float* __restrict__ p0,    // b0|g0|be0 (3*128)
                        const __bf16* __restrict__ W3,   // (128, 128)
                        const float* __restrict__ p3,    // b3|g3|be3 (3*128)
                        __bf16* __restrict__ H,          // (E, 128)
                        __bf16* __restrict__ yh0,        // (E, 128) saved
                        __bf16* __restrict__ yh3,        // (E, 128) saved
                        float* __restrict__ rs01,        // (2, E) saved rs0,rs3
                        int E, float eps) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __bf16* a1_lds = reinterpret_cast<__bf16*>(smem);                  // swizzled [32][128] 8K
    float* y3_lds = reinterpret_cast<float*>(smem + RT_E * 256);       // [32][132] 16.5K
    float* xe_lds = reinterpret_cast<float*>(smem + RT_E * 256 + RT_E * RT_PITCH * 4); // [32][D]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;        // 0..3
    const int e0 = blockIdx.x * RT_E;

    // stage X tile (fp32 in LDS)
    for (int i = tid; i < RT_E * D; i += RT_NT) {
        int e = i / D, d = i % D;
        xe_lds[e * D + d] = (e0 + e < E) ? rt_b2f(X[(size_t)(e0 + e) * D + d]) : 0.f;
    }
    __syncthreads();

    // ---- phase A: y0 GEMV + LN + GELU, one wave per edge (8 edges/wave)
    // lanes hold features k = lane and k + 64
    for (int le = 0; le < 8; ++le) {
        const int e = w * 8 + le;
        float y[2];
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            float acc = p0[k];                       // b0
#pragma unroll
            for (int d = 0; d < D; ++d)
                acc = fmaf(xe_lds[e * D + d], rt_b2f(W0[(size_t)k * D + d]), acc);
            y[h] = acc;
        }
        float s = y[0] + y[1], q = y[0] * y[0] + y[1] * y[1];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            s += __shfl_xor(s, off);
            q += __shfl_xor(q, off);
        }
        const float mean = s * (1.f / RT_K);
        const float var = q * (1.f / RT_K) - mean * mean;
        const float rs = rsqrtf(var + eps);
        if (lane == 0 && e0 + e < E) rs01[e0 + e] = rs;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float yh = (y[h] - mean) * rs;
            const float z = yh * p0[128 + k] + p0[256 + k];   // g0, be0
            const float a = rt_gelu(z);
            if (e0 + e < E) yh0[(size_t)(e0 + e) * RT_K + k] = (__bf16)yh;
            *reinterpret_cast<__bf16*>(
                reinterpret_cast<char*>(a1_lds) + rt_swz(e, k)) = (__bf16)a;
        }
    }
    __syncthreads();

    // ---- phase B: y3 = a1 @ W3^T + b3 via MFMA; wave = 16e x 64k3
    {
        const int we = w & 1, wk = w >> 1;
        const int l15 = lane & 15, l4 = lane >> 4;
        f32x4 acc[4];
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) acc[nf] = f32x4(0.f);
#pragma unroll
        for (int kit = 0; kit < 4; ++kit) {
            const int k1 = kit * 32 + l4 * 8;
            const int e = we * 16 + l15;
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<char*>(a1_lds) + rt_swz(e, k1));
#pragma unroll
            for (int nf = 0; nf < 4; ++nf) {
                const int k3 = wk * 64 + nf * 16 + l15;
                bf16x8 b = *reinterpret_cast<const bf16x8*>(W3 + (size_t)k3 * RT_K + k1);
                // rows = k3 (from b, arg0), cols = e (from a, arg1)
                acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(b, a, acc[nf], 0, 0, 0);
            }
        }
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                const int k3 = wk * 64 + nf * 16 + l4 * 4 + reg;
                const int e = we * 16 + l15;
                y3_lds[e * RT_PITCH + k3] = acc[nf][reg] + p3[k3];   // + b3
            }
    }
    __syncthreads();

    // ---- phase C: LN3 + GELU -> H
    for (int le = 0; le < 8; ++le) {
        const int e = w * 8 + le;
        float y[2] = {y3_lds[e * RT_PITCH + lane], y3_lds[e * RT_PITCH + lane + 64]};
        float s = y[0] + y[1], q = y[0] * y[0] + y[1] * y[1];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            s += __shfl_xor(s, off);
            q += __shfl_xor(q, off);
        }
        const float mean = s * (1.f / RT_K);
        const float var = q * (1.f / RT_K) - mean * mean;
        const float rs = rsqrtf(var + eps);
        if (e0 + e >= E) continue;
        if (lane == 0) rs01[E + e0 + e] = rs;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float yh = (y[h] - mean) * rs;
            const float z = yh * p3[128 + k] + p3[256 + k];   // g3, be3
            yh3[(size_t)(e0 + e) * RT_K + k] = (__bf16)yh;
            H[(size_t)(e0 + e) * RT_K + k] = (__bf16)rt_gelu(z);
        }
    }
}

// ---------------------------------------------------------------------------
// backward
//   dz3 = dH * gelu'(z3);  LN3-bwd -> dy3 (+ dg3, dbe3)
//   dW3 = dy3^T a1, db3 = sum dy3, da1 = dy3 W3
//   dz0 = da1 * gelu'(z0); LN0-bwd -> dy0 (+ dg0, dbe0)
//   dW0 = dy0^T X, db0 = sum dy0, dX = dy0 W0 (optional)
// Per-block partials land in the fp32 grad buffers via atomicAdd.
// ---------------------------------------------------------------------------
template <int D, int NEEDX>
__global__ void __launch_bounds__(RT_NT)
radial_trunk_bwd_kernel(const __bf16* __restrict__ dHg,  // (E, 128)
                        const __bf16* __restrict__ X,    // (E, D)
                        const __bf16* __restrict__ W0,   // (128, D)
                        const float* __restrict__ p0,    // b0|g0|be0
                        const __bf16* __restrict__ W3,   // (128, 128)
                        const __bf16* __restrict__ W3t,  // (128, 128) = W3^T
                        const float* __restrict__ p3,    // b3|g3|be3
                        const __bf16* __restrict__ yh0,
                        const __bf16* __restrict__ yh3,
                        const float* __restrict__ rs01,
                        float* __restrict__ dW0,         // (128, D)
                        float* __restrict__ dp0,         // db0|dg0|dbe0 (3*128)
                        float* __restrict__ dW3,         // (128, 128)
                        float* __restrict__ dp3,         // db3|dg3|dbe3
                        float* __restrict__ dX,          // (E, D) or null
                        int E, float eps) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    // carve
    __bf16* dy3_e = reinterpret_cast<__bf16*>(smem);                    // swizzled [32e][128k3] 8K
    __bf16* dy3_t = reinterpret_cast<__bf16*>(smem + 8192);             // [128k3][32e] 8K
    __bf16* a1_t = reinterpret_cast<__bf16*>(smem + 16384);             // [128k1][32e] 8K
    __bf16* yh0_l = reinterpret_cast<__bf16*>(smem + 24576);            // [32e][128] 8K
    float* da1_l = reinterpret_cast<float*>(smem + 32768);              // [32][132] 16.5K
    float* dy0_l = reinterpret_cast<float*>(smem + 32768 + RT_E * RT_PITCH * 4); // [32][132] 16.5K
    float* red = reinterpret_cast<float*>(smem + 32768 + 2 * RT_E * RT_PITCH * 4); // 4*128 partials
    float* xe_lds = red + 4 * RT_K;                                     // [32][D]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int w = tid >> 6;
    const int e0 = blockIdx.x * RT_E;

    for (int i = tid; i < 4 * RT_K; i += RT_NT) red[i] = 0.f;
    for (int i = tid; i < RT_E * D; i += RT_NT) {
        int e = i / D, d = i % D;
        xe_lds[e * D + d] = (e0 + e < E) ? rt_b2f(X[(size_t)(e0 + e) * D + d]) : 0.f;
    }
    __syncthreads();   // red zeroed before phase 1's LDS atomics

    // ---- phase 1: dz3, LN3 backward -> dy3 (both layouts); dg3/dbe3 partials
    for (int le = 0; le < 8; ++le) {
        const int e = w * 8 + le;
        const bool ok = (e0 + e < E);
        const float rs3 = ok ? rs01[E + e0 + e] : 0.f;
        float dz[2], yh[2];
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float y = ok ? rt_b2f(yh3[(size_t)(e0 + e) * RT_K + k]) : 0.f;
            const float g = p3[128 + k];
            const float z = y * g + p3[256 + k];
            const float dH = ok ? rt_b2f(dHg[(size_t)(e0 + e) * RT_K + k]) : 0.f;
            dz[h] = dH * rt_dgelu(z);
            yh[h] = y;
            // a1 recompute for the dW3 B-operand (k1-major transpose);
            // out-of-range edges must contribute ZERO to the dW3 contraction
            const float y0 = ok ? rt_b2f(yh0[(size_t)(e0 + e) * RT_K + k]) : 0.f;
            const float z0 = y0 * p0[128 + k] + p0[256 + k];
            a1_t[(size_t)k * RT_E + e] = ok ? (__bf16)rt_gelu(z0) : (__bf16)0.f;
            yh0_l[e * RT_K + k] = (__bf16)y0;
        }
        float s1 = dz[0] * p3[128 + lane] + dz[1] * p3[128 + lane + 64];
        float s2 = dz[0] * p3[128 + lane] * yh[0] + dz[1] * p3[128 + lane + 64] * yh[1];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            s1 += __shfl_xor(s1, off);
            s2 += __shfl_xor(s2, off);
        }
        const float m1 = s1 * (1.f / RT_K), m2 = s2 * (1.f / RT_K);
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float g = p3[128 + k];
            const float dy = rs3 * (dz[h] * g - m1 - yh[h] * m2);
            *reinterpret_cast<__bf16*>(
                reinterpret_cast<char*>(dy3_e) + rt_swz(e, k)) = (__bf16)dy;
            dy3_t[(size_t)k * RT_E + e] = (__bf16)dy;
            // dg3 / dbe3 / db3 partials (db3 == sum dy)
            atomicAdd(&red[k], dz[h] * yh[h]);            // dg3
            atomicAdd(&red[RT_K + k], dz[h]);             // dbe3
            atomicAdd(&red[2 * RT_K + k], dy);            // db3
        }
    }
    __syncthreads();

    // flush dg3/dbe3/db3 partials
    for (int k = tid; k < RT_K; k += RT_NT) {
        atomicAdd(&dp3[RT_K + k], red[k]);        // dg3
        atomicAdd(&dp3[2 * RT_K + k], red[RT_K + k]);   // dbe3
        atomicAdd(&dp3[k], red[2 * RT_K + k]);    // db3
    }
    __syncthreads();
    for (int i = tid; i < 3 * RT_K; i += RT_NT) red[i] = 0.f;

    // ---- phase 2: dW3 += dy3^T @ a1 (M=k3, N=k1, K=32 edges)
    {
        const int l15 = lane & 15, l4 = lane >> 4;
        // wave w covers k3 rows [w*32, w*32+32), all 128 k1 cols
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            const int k3 = w * 32 + mf * 16 + l15;
            bf16x8 a0 = *reinterpret_cast<const bf16x8*>(dy3_t + (size_t)k3 * RT_E + l4 * 8);
#pragma unroll
            for (int nf = 0; nf < 8; ++nf) {
                const int k1 = nf * 16 + l15;
                bf16x8 b0 = *reinterpret_cast<const bf16x8*>(a1_t + (size_t)k1 * RT_E + l4 * 8);
                f32x4 acc = f32x4(0.f);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
                // rows = k3 (l4*4+reg), cols = k1 (l15)
#pragma unroll
                for (int reg = 0; reg < 4; ++reg) {
                    const int k3r = w * 32 + mf * 16 + l4 * 4 + reg;
                    atomicAdd(&dW3[(size_t)k3r * RT_K + nf * 16 + l15], acc[reg]);
                }
            }
        }
    }

    // ---- phase 3: da1 = dy3 @ W3 (rows k1 from W3t frags, cols e)
    {
        const int we = w & 1, wk = w >> 1;
        const int l15 = lane & 15, l4 = lane >> 4;
        f32x4 acc[4];
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) acc[nf] = f32x4(0.f);
#pragma unroll
        for (int kit = 0; kit < 4; ++kit) {
            const int k3 = kit * 32 + l4 * 8;
            const int e = we * 16 + l15;
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<char*>(dy3_e) + rt_swz(e, k3));
#pragma unroll
            for (int nf = 0; nf < 4; ++nf) {
                const int k1 = wk * 64 + nf * 16 + l15;
                bf16x8 b = *reinterpret_cast<const bf16x8*>(W3t + (size_t)k1 * RT_K + k3);
                acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(b, a, acc[nf], 0, 0, 0);
            }
        }
        __syncthreads();   // dy3_e no longer needed; da1_l alias-safe
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                const int k1 = wk * 64 + nf * 16 + l4 * 4 + reg;
                const int e = we * 16 + l15;
                da1_l[e * RT_PITCH + k1] = acc[nf][reg];
            }
    }
    __syncthreads();

    // ---- phase 4: dz0, LN0 backward -> dy0; dg0/dbe0/db0 partials
    for (int le = 0; le < 8; ++le) {
        const int e = w * 8 + le;
        const bool ok = (e0 + e < E);
        const float rs0 = ok ? rs01[e0 + e] : 0.f;
        float dz[2], yh[2];
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float y = rt_b2f(yh0_l[e * RT_K + k]);
            const float z = y * p0[128 + k] + p0[256 + k];
            dz[h] = da1_l[e * RT_PITCH + k] * rt_dgelu(z);
            yh[h] = y;
        }
        float s1 = dz[0] * p0[128 + lane] + dz[1] * p0[128 + lane + 64];
        float s2 = dz[0] * p0[128 + lane] * yh[0] + dz[1] * p0[128 + lane + 64] * yh[1];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            s1 += __shfl_xor(s1, off);
            s2 += __shfl_xor(s2, off);
        }
        const float m1 = s1 * (1.f / RT_K), m2 = s2 * (1.f / RT_K);
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            const int k = lane + h * 64;
            const float g = p0[128 + k];
            const float dy = ok ? rs0 * (dz[h] * g - m1 - yh[h] * m2) : 0.f;
            dy0_l[e * RT_PITCH + k] = dy;
            atomicAdd(&red[k], ok ? dz[h] * yh[h] : 0.f);   // dg0
            atomicAdd(&red[RT_K + k], ok ? dz[h] : 0.f);    // dbe0
            atomicAdd(&red[2 * RT_K + k], dy);              // db0
        }
    }
    __syncthreads();
    for (int k = tid; k < RT_K; k += RT_NT) {
        atomicAdd(&dp0[RT_K + k], red[k]);
        atomicAdd(&dp0[2 * RT_K + k], red[RT_K + k]);
        atomicAdd(&dp0[k], red[2 * RT_K + k]);
    }

    // ---- phase 5: dW0[k][d] += sum_e dy0[e][k] * X[e][d]  (VALU, D small)
    for (int i = tid; i < RT_K * D; i += RT_NT) {
        const int k = i / D, d = i % D;
        float acc = 0.f;
#pragma unroll 4
        for (int e = 0; e < RT_E; ++e)
            acc = fmaf(dy0_l[e * RT_PITCH + k], xe_lds[e * D + d], acc);
        atomicAdd(&dW0[(size_t)k * D + d], acc);
    }

    // ---- phase 6 (optional): dX[e][d] = sum_k dy0[e][k] * W0[k][d]
    if (NEEDX) {
        for (int i = tid; i < RT_E * D; i += RT_NT) {
            const int e = i / D, d = i % D;
            if (e0 + e >= E) continue;
            float acc = 0.f;
            for (int k = 0; k < RT_K; ++k)
                acc = fmaf(dy0_l[e * RT_PITCH + k], rt_b2f(W0[(size_t)k * D + d]), acc);
            dX[(size_t)(e0 + e) * D + d] = acc;
        }
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
#define RT_DISPATCH_D(D, ...)                                       \
    switch (D) {                                                    \
        case 1: { constexpr int kD = 1; __VA_ARGS__; break; }       \
        case 2: { constexpr int kD = 2; __VA_ARGS__; break; }       \
        case 3: { constexpr int kD = 3; __VA_ARGS__; break; }       \
        case 5: { constexpr int kD = 5; __VA_ARGS__; break; }       \
        case 9: { constexpr int kD = 9; __VA_ARGS__; break; }       \
        case 17: { constexpr int kD = 17; __VA_ARGS__; break; }     \
        case 21: { constexpr int kD = 21; __VA_ARGS__; break; }     \
        case 25: { constexpr int kD = 25; __VA_ARGS__; break; }     \
        default: TORCH_CHECK(false, "unsupported edge dim ", D);    \
    }

void radial_trunk_fwd(torch::Tensor X, torch::Tensor W0, torch::Tensor p0,
                      torch::Tensor W3, torch::Tensor p3,
                      torch::Tensor H, torch::Tensor yh0, torch::Tensor yh3,
                      torch::Tensor rs01, double eps) {
    int E = X.size(0), D = X.size(1);
    TORCH_CHECK(X.is_cuda() && X.is_contiguous() && X.dtype() == torch::kBFloat16);
    TORCH_CHECK(W0.is_contiguous() && W3.is_contiguous() &&
                p0.is_contiguous() && p3.is_contiguous());
    TORCH_CHECK(W0.size(0) == RT_K && W0.size(1) == D);
    TORCH_CHECK(W3.size(0) == RT_K && W3.size(1) == RT_K);
    TORCH_CHECK(p0.numel() == 3 * RT_K && p3.numel() == 3 * RT_K);
    TORCH_CHECK(rs01.numel() == 2 * E);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((E + RT_E - 1) / RT_E);
    RT_DISPATCH_D(D, {
        size_t lds = RT_E * 256 + RT_E * RT_PITCH * 4 + RT_E * kD * 4;
        hipLaunchKernelGGL(HIP_KERNEL_NAME(radial_trunk_fwd_kernel<kD>),
                           grid, dim3(RT_NT), lds, stream,
                           reinterpret_cast<const __bf16*>(X.data_ptr()),
                           reinterpret_cast<const __bf16*>(W0.data_ptr()),
                           p0.data_ptr<float>(),
                           reinterpret_cast<const __bf16*>(W3.data_ptr()),
                           p3.data_ptr<float>(),
                           reinterpret_cast<__bf16*>(H.data_ptr()),
                           reinterpret_cast<__bf16*>(yh0.data_ptr()),
                           reinterpret_cast<__bf16*>(yh3.data_ptr()),
                           rs01.data_ptr<float>(), E, (float)eps);
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "radial_trunk_fwd: ", hipGetErrorString(err));
}

void radial_trunk_bwd(torch::Tensor dH, torch::Tensor X, torch::Tensor W0,
                      torch::Tensor p0, torch::Tensor W3, torch::Tensor W3t,
                      torch::Tensor p3, torch::Tensor yh0, torch::Tensor yh3,
                      torch::Tensor rs01, torch::Tensor dW0, torch::Tensor dp0,
                      torch::Tensor dW3, torch::Tensor dp3, torch::Tensor dX,
                      double eps) {
    int E = X.size(0), D = X.size(1);
    TORCH_CHECK(dH.is_contiguous() && dH.dtype() == torch::kBFloat16);
    bool needx = dX.defined() && dX.numel() > 0;
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((E + RT_E - 1) / RT_E);
    RT_DISPATCH_D(D, {
        size_t lds = 32768 + 2 * RT_E * RT_PITCH * 4 + 4 * RT_K * 4 + RT_E * kD * 4;
        if (needx) {
            hipLaunchKernelGGL(HIP_KERNEL_NAME(radial_trunk_bwd_kernel<kD, 1>),
                               grid, dim3(RT_NT), lds, stream,
                               reinterpret_cast<const __bf16*>(dH.data_ptr()),
                               reinterpret_cast<const __bf16*>(X.data_ptr()),
                               reinterpret_cast<const __bf16*>(W0.data_ptr()),
                               p0.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(W3.data_ptr()),
                               reinterpret_cast<const __bf16*>(W3t.data_ptr()),
                               p3.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(yh0.data_ptr()),
                               reinterpret_cast<const __bf16*>(yh3.data_ptr()),
                               rs01.data_ptr<float>(), dW0.data_ptr<float>(),
                               dp0.data_ptr<float>(), dW3.data_ptr<float>(),
                               dp3.data_ptr<float>(), dX.data_ptr<float>(),
                               E, (float)eps);
        } else {
            hipLaunchKernelGGL(HIP_KERNEL_NAME(radial_trunk_bwd_kernel<kD, 0>),
                               grid, dim3(RT_NT), lds, stream,
                               reinterpret_cast<const __bf16*>(dH.data_ptr()),
                               reinterpret_cast<const __bf16*>(X.data_ptr()),
                               reinterpret_cast<const __bf16*>(W0.data_ptr()),
                               p0.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(W3.data_ptr()),
                               reinterpret_cast<const __bf16*>(W3t.data_ptr()),
                               p3.data_ptr<float>(),
                               reinterpret_cast<const __bf16*>(yh0.data_ptr()),
                               reinterpret_cast<const __bf16*>(yh3.data_ptr()),
                               rs01.data_ptr<float>(), dW0.data_ptr<float>(),
                               dp0.data_ptr<float>(), dW3.data_ptr<float>(),
                               dp3.data_ptr<float>(), nullptr, E, (float)eps);
        }
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "radial_trunk_bwd: ", hipGetErrorString(err));
}
