// Basis-feature precontraction ("u build") for the fused pairwise conv:
//
//   Ut[(c*F + f), o, e] = sum_i  B[e, o, i, f] * X[e, c, i]
//
// This is the `torch.einsum('eoif,eci->cfoe')` in PairwiseConv.apply_fused
// (reference contraction se3_transformer_pytorch.py:336-338 restructured,
// see models/core.py) — in eager torch it lowers to permute-copies + bmm +
// an output copy (the largest non-custom slice of the round-1/2 kernel
// stats). Here: one pass, B staged once per edge tile in LDS (it is re-used
// by all C channels), output written directly in the e-contiguous layout
// the pairconv kernels consume.
//
// Backward (dX only): dX[e, c, i] = sum_{f,o} B[e,o,i,f] * dU[(c*F+f), o, e].
// dB (differentiable_coors) stays on the eager path — the Python wrapper
// gates on basis.requires_grad.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define UB_NT 256
#define UB_E 32          // edges per block (fwd)
#define UB_C 32          // channel chunk (fwd)
#define UB_EB 16         // edges per block (bwd)
#define UB_CB 8          // channel chunk (bwd)

typedef __attribute__((ext_vector_type(2))) __bf16 ub_bf16x2;

template <typename TX>
__global__ void __launch_bounds__(UB_NT)
ubuild_fwd_kernel(const float* __restrict__ B,   // (E, O, I, F)
                  const TX* __restrict__ X,      // (E, C, I)
                  __bf16* __restrict__ Ut,       // (C*F, O, E)
                  int E, int C, int O, int I, int F) {
    // b and x tiles are stored [.., i PADDED to 8] bf16 so each output
    // element is two 16B LDS reads + 4 v_dot2_f32_bf16 over i
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __bf16* b_lds = reinterpret_cast<__bf16*>(smem);          // [32e][O*F][8i]
    __bf16* x_lds = b_lds + UB_E * O * F * 8;                 // [32c][32e][8i]

    const int tid = threadIdx.x;
    const int e0 = blockIdx.x * UB_E;
    const int oif = O * I * F;
    const int of = O * F;

    for (int t = tid; t < UB_E * of; t += UB_NT) {   // zero-init b pad lanes
        *reinterpret_cast<bf16x8*>(b_lds + (size_t)t * 8) = bf16x8(0);
    }
    __syncthreads();
    for (int t = tid; t < UB_E * oif; t += UB_NT) {
        int e = t / oif, r = t % oif;
        int o = r / (I * F), rem = r % (I * F);
        int i = rem / F, f = rem % F;
        b_lds[((size_t)e * of + o * F + f) * 8 + i] =
            (__bf16)((e0 + e < E) ? B[(size_t)(e0 + e) * oif + r] : 0.f);
    }

    for (int c0 = 0; c0 < C; c0 += UB_C) {
        __syncthreads();   // previous chunk's compute done with x_lds
        for (int t = tid; t < UB_C * UB_E * 8; t += UB_NT) {
            int c = t / (UB_E * 8), rem = t % (UB_E * 8);
            int e = rem / 8, i = rem % 8;
            x_lds[((size_t)c * UB_E + e) * 8 + i] = (__bf16)(
                (i < I && e0 + e < E)
                    ? (float)X[((size_t)(e0 + e) * C + c0 + c) * I + i] : 0.f);
        }
        __syncthreads();
        // out elements of this chunk: (c 32) x (f F) x (o O) x (e 32), e minor
        const int nel = UB_C * F * O * UB_E;
        for (int t = tid; t < nel; t += UB_NT) {
            int e = t & (UB_E - 1);
            int r = t / UB_E;          // ((c*F + f)*O + o)
            int o = r % O;
            int cf = r / O;
            int f = cf % F, c = cf / F;
            bf16x8 bv = *reinterpret_cast<const bf16x8*>(
                b_lds + ((size_t)e * of + o * F + f) * 8);
            bf16x8 xv = *reinterpret_cast<const bf16x8*>(
                x_lds + ((size_t)c * UB_E + e) * 8);
            const ub_bf16x2* b2 = reinterpret_cast<const ub_bf16x2*>(&bv);
            const ub_bf16x2* x2 = reinterpret_cast<const ub_bf16x2*>(&xv);
            float acc = 0.f;
#pragma unroll
            for (int p = 0; p < 4; ++p)
                acc = __builtin_amdgcn_fdot2_f32_bf16(b2[p], x2[p], acc, false);
            if (e0 + e < E)
                Ut[((size_t)((c0 + c) * F + f) * O + o) * E + e0 + e] = (__bf16)acc;
        }
    }
}

template <typename TX>
__global__ void __launch_bounds__(UB_NT)
ubuild_bwd_dx_kernel(const float* __restrict__ B,   // (E, O, I, F)
                     const float* __restrict__ dU,  // (C*F, O, E) f32
                     TX* __restrict__ dX,           // (E, C, I)
                     int E, int C, int O, int I, int F) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* b_lds = reinterpret_cast<float*>(smem);            // [16e][O*I*F]
    float* g_lds = b_lds + UB_EB * O * I * F;                 // [8c][F*O][16e]

    const int tid = threadIdx.x;
    const int e0 = blockIdx.x * UB_EB;
    const int oif = O * I * F;
    const int fo = F * O;

    for (int t = tid; t < UB_EB * oif; t += UB_NT) {
        int e = t / oif, r = t % oif;
        b_lds[e * oif + r] = (e0 + e < E) ? B[(size_t)(e0 + e) * oif + r] : 0.f;
    }

    for (int c0 = 0; c0 < C; c0 += UB_CB) {
        __syncthreads();
        for (int t = tid; t < UB_CB * fo * UB_EB; t += UB_NT) {
            int e = t % UB_EB;
            int r = t / UB_EB;         // c*F*O + (f*O + o)
            int c = r / fo, fo_i = r % fo;
            int f = fo_i / O, o = fo_i % O;
            g_lds[(c * fo + fo_i) * UB_EB + e] = (e0 + e < E)
                ? dU[((size_t)((c0 + c) * F + f) * O + o) * E + e0 + e] : 0.f;
        }
        __syncthreads();
        // dX elements: (c 8) x (i I) x (e 16), e minor
        const int nel = UB_CB * I * UB_EB;
        for (int t = tid; t < nel; t += UB_NT) {
            int e = t % UB_EB;
            int r = t / UB_EB;
            int i = r % I, c = r / I;
            float acc = 0.f;
            const float* g = g_lds + (c * fo) * UB_EB + e;
            const float* b = b_lds + e * oif + i * F;
            for (int f = 0; f < F; ++f)
                for (int o = 0; o < O; ++o)
                    acc = fmaf(b[(size_t)o * I * F + f], g[(f * O + o) * UB_EB], acc);
            if (e0 + e < E)
                dX[((size_t)(e0 + e) * C + c0 + c) * I + i] = (TX)acc;
        }
    }
}

void ubuild_fwd(torch::Tensor B, torch::Tensor X, torch::Tensor Ut,
                int64_t O, int64_t I, int64_t F) {
    TORCH_CHECK(B.is_cuda() && B.is_contiguous() && B.dtype() == torch::kFloat32);
    TORCH_CHECK(X.is_contiguous() && Ut.is_contiguous() &&
                Ut.dtype() == torch::kBFloat16);
    int E = B.size(0);
    int C = X.size(1);
    TORCH_CHECK(C % UB_C == 0, "channels must be a multiple of 32");
    TORCH_CHECK(O * I * F <= 343, "degree pair too large for LDS staging");
    TORCH_CHECK(I <= 8, "I (2*d_in+1) must be <= 8");
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((E + UB_E - 1) / UB_E);
    size_t lds = (size_t)UB_E * O * F * 8 * 2 + (size_t)UB_C * UB_E * 8 * 2;
    if (X.dtype() == torch::kFloat32) {
        hipLaunchKernelGGL(HIP_KERNEL_NAME(ubuild_fwd_kernel<float>), grid,
                           dim3(UB_NT), lds, stream, B.data_ptr<float>(),
                           X.data_ptr<float>(),
                           reinterpret_cast<__bf16*>(Ut.data_ptr()),
                           E, C, (int)O, (int)I, (int)F);
    } else {
        TORCH_CHECK(X.dtype() == torch::kBFloat16);
        hipLaunchKernelGGL(HIP_KERNEL_NAME(ubuild_fwd_kernel<__bf16>), grid,
                           dim3(UB_NT), lds, stream, B.data_ptr<float>(),
                           reinterpret_cast<const __bf16*>(X.data_ptr()),
                           reinterpret_cast<__bf16*>(Ut.data_ptr()),
                           E, C, (int)O, (int)I, (int)F);
    }
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "ubuild_fwd: ", hipGetErrorString(err));
}

void ubuild_bwd_dx(torch::Tensor B, torch::Tensor dU, torch::Tensor dX,
                   int64_t O, int64_t I, int64_t F) {
    TORCH_CHECK(B.is_cuda() && B.is_contiguous() && B.dtype() == torch::kFloat32);
    TORCH_CHECK(dU.is_contiguous() && dU.dtype() == torch::kFloat32 &&
                dX.is_contiguous());
    int E = B.size(0);
    int C = dX.size(1);
    TORCH_CHECK(C % UB_CB == 0, "channels must be a multiple of 8");
    TORCH_CHECK(O * I * F <= 343);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((E + UB_EB - 1) / UB_EB);
    size_t lds = (size_t)UB_EB * O * I * F * 4 + (size_t)UB_CB * F * O * UB_EB * 4;
    if (dX.dtype() == torch::kFloat32) {
        hipLaunchKernelGGL(HIP_KERNEL_NAME(ubuild_bwd_dx_kernel<float>), grid,
                           dim3(UB_NT), lds, stream, B.data_ptr<float>(),
                           dU.data_ptr<float>(), dX.data_ptr<float>(),
                           E, C, (int)O, (int)I, (int)F);
    } else {
        TORCH_CHECK(dX.dtype() == torch::kBFloat16);
        hipLaunchKernelGGL(HIP_KERNEL_NAME(ubuild_bwd_dx_kernel<__bf16>), grid,
                           dim3(UB_NT), lds, stream, B.data_ptr<float>(),
                           dU.data_ptr<float>(),
                           reinterpret_cast<__bf16*>(dX.data_ptr()),
                           E, C, (int)O, (int)I, (int)F);
    }
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "ubuild_bwd_dx: ", hipGetErrorString(err));
}
