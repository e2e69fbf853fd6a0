// Fused weight-pack kernel: one read of the radial net.6 weight (torch
// Linear layout, (mo*miF, 128), fp32 or bf16) produces BOTH per-lane
// fragment-order packed layouts the pairconv kernels consume:
//
//   P_fwd [mo/8][miF/32][wm4][mf4][kit4][lane64][j8]   (fwd + bwd_du A-operand)
//   P_dh  [mo/8][miF/32][wk2][kf4][ns8][lane64][j8]    (bwd_dh B-operand)
//
// Replaces the per-step host-side chain (W.to(bf16) copy + two
// permute().contiguous() passes in forward and two more in backward =
// ~6% of the round-1 step, r01_final_kernel_stats.csv 'direct_copy' row)
// with a single kernel: W is read once, both outputs written once.
//
// Block = one (m, cb) tile: 32 rows (c) x 128 cols (k), staged in LDS as
// bf16 with a 16B-slot XOR swizzle so phase-2 reads are conflict-light.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define NTP 256

template <typename TIN>
__global__ void __launch_bounds__(NTP)
pack_w_kernel(const TIN* __restrict__ W, __bf16* __restrict__ Pf,
              __bf16* __restrict__ Pdh, int mo, int miF) {
    __shared__ __bf16 lds[32 * 128];
    const int nc = miF / 32;
    const int m = blockIdx.x / nc, cb = blockIdx.x % nc;
    const int tid = threadIdx.x;

    // phase 1: load + convert the (32 c) x (128 k) tile, coalesced per row
    for (int i = tid; i < 512; i += NTP) {            // 512 vec8 units
        int c32 = i >> 4, k8 = i & 15;
        const TIN* src = W + ((size_t)m * miF + (size_t)cb * 32 + c32) * 128 + k8 * 8;
        __bf16 v[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = (__bf16)(float)src[j];
        *reinterpret_cast<bf16x8*>(&lds[c32 * 128 + ((k8 ^ (c32 & 15)) * 8)]) =
            *reinterpret_cast<bf16x8*>(v);
    }
    __syncthreads();

    // phase 2a: P_fwd. m = a0*8 + a1*2 + a2; c = cb*32 + a4*16 + a5;
    // k = a6*32 + a7*8 + a8; out dims [a0][cb][a1][a2|a4][a6][a7|a5][a8].
    const int a0 = m >> 3, a1 = (m >> 1) & 3, a2 = m & 1;
    for (int t = tid; t < 512; t += NTP) {
        int a5 = t & 15, a7 = (t >> 4) & 3, a6 = (t >> 6) & 3, a4 = t >> 8;
        int c32 = a4 * 16 + a5;
        int k8 = a6 * 4 + a7;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            &lds[c32 * 128 + ((k8 ^ (c32 & 15)) * 8)]);
        size_t off = (((((size_t)a0 * nc + cb) * 4 + a1) * 4 + (a2 * 2 + a4)) * 4 + a6) * 512
                     + (a7 * 16 + a5) * 8;
        *reinterpret_cast<bf16x8*>(Pf + off) = v;
    }

    // phase 2b: P_dh. m = b0*8 + b1; c = cb*32 + b3*8 + b4;
    // k = b5*64 + b6*16 + b7; out dims [b0][cb][b5][b6][b1][b3|b7][b4].
    const int b0 = m >> 3, b1 = m & 7;
    for (int t = tid; t < 512; t += NTP) {
        int b7 = t & 15, b3 = (t >> 4) & 3, b6 = (t >> 6) & 3, b5 = t >> 8;
        int k = b5 * 64 + b6 * 16 + b7;
        int k8 = k >> 3, kj = k & 7;
        __bf16 v[8];
#pragma unroll
        for (int b4 = 0; b4 < 8; ++b4) {
            int c32 = b3 * 8 + b4;
            v[b4] = lds[c32 * 128 + ((k8 ^ (c32 & 15)) * 8) + kj];
        }
        size_t off = (((((size_t)b0 * nc + cb) * 2 + b5) * 4 + b6) * 8 + b1) * 512
                     + (b3 * 16 + b7) * 8;
        *reinterpret_cast<bf16x8*>(Pdh + off) = *reinterpret_cast<bf16x8*>(v);
    }
}

void pack_w_both(torch::Tensor W, torch::Tensor Pf, torch::Tensor Pdh, int64_t mo_) {
    TORCH_CHECK(W.is_cuda() && W.is_contiguous());
    TORCH_CHECK(Pf.is_cuda() && Pf.is_contiguous() && Pf.dtype() == torch::kBFloat16);
    TORCH_CHECK(Pdh.is_cuda() && Pdh.is_contiguous() && Pdh.dtype() == torch::kBFloat16);
    int mo = (int)mo_;
    int64_t N = W.size(0);
    TORCH_CHECK(W.dim() == 2 && W.size(1) == 128, "expect (N, 128) weight");
    TORCH_CHECK(N % mo == 0, "rows must split into mo blocks");
    int miF = (int)(N / mo);
    TORCH_CHECK(mo % 8 == 0 && miF % 32 == 0);
    TORCH_CHECK(Pf.numel() == N * 128 && Pdh.numel() == N * 128);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((long)mo * (miF / 32));
    if (W.dtype() == torch::kFloat32) {
        hipLaunchKernelGGL(HIP_KERNEL_NAME(pack_w_kernel<float>), grid, dim3(NTP), 0, stream,
                           W.data_ptr<float>(),
                           reinterpret_cast<__bf16*>(Pf.data_ptr()),
                           reinterpret_cast<__bf16*>(Pdh.data_ptr()), mo, miF);
    } else {
        TORCH_CHECK(W.dtype() == torch::kBFloat16);
        hipLaunchKernelGGL(HIP_KERNEL_NAME(pack_w_kernel<__bf16>), grid, dim3(NTP), 0, stream,
                           reinterpret_cast<const __bf16*>(W.data_ptr()),
                           reinterpret_cast<__bf16*>(Pf.data_ptr()),
                           reinterpret_cast<__bf16*>(Pdh.data_ptr()), mo, miF);
    }
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "pack_w_both: ", hipGetErrorString(err));
}
