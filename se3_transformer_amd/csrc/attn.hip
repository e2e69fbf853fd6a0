// Fused per-degree equivariant neighbor attention (reference
// se3_transformer_pytorch.py:508-517): logits over the k-NN neighborhood
// (incl. prepended self/null/global keys), masked softmax, weighted value
// sum — one kernel instead of einsum/masked_fill/softmax/einsum.
//
// Layouts (prepared by the Python wrapper):
//   q    (R, DM)        R = b*h*n rows, DM = dim_head * (2l+1)
//   k, v (R, J, DM)     J = neighbors + prefix keys (<= 64)
//   mask (b, n, J) u8   1 = attend (broadcast over heads); absent => all 1
//   out  (R, DM)
// One wave per query row: the DM axis is cycled over the 64 lanes; logits,
// softmax and the attention weights live one-per-lane (lane j <-> key j).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#define NTA 256  // 4 waves per block

template <typename T, int DT>  // DT = ceil(DM/64) upper bound actually used
__global__ void __launch_bounds__(NTA)
attn_fwd_kernel(const T* __restrict__ q, const T* __restrict__ k,
                const T* __restrict__ v, const unsigned char* __restrict__ mask,
                float* __restrict__ out, long R, int J, int DM, int n,
                int heads, float scale, int kv_one) {
    const int lane = threadIdx.x & 63;
    const long r = ((long)blockIdx.x * (NTA / 64)) + (threadIdx.x >> 6);
    if (r >= R) return;

    // load q row into per-lane chunks
    float qv[DT];
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        qv[t] = (d < DM) ? (float)q[r * DM + d] : 0.f;
    }

    // one-headed KV: all heads of a (b, i) row share one key/value row
    long b0_ = r / ((long)heads * n);
    long i0_ = r % n;
    const long kvr = kv_one ? (b0_ * n + i0_) : r;

    // logits: lane j holds logit_j
    float logit = -3.0e38f;
    const unsigned char* mrow = nullptr;
    if (mask != nullptr) {
        // rows are ordered (b, h, i): b = r / (heads*n), i = r % n
        long b_ = r / ((long)heads * n);
        long i_ = r % n;
        mrow = mask + (b_ * n + i_) * J;
    }
    for (int j = 0; j < J; ++j) {
        float part = 0.f;
#pragma unroll
        for (int t = 0; t < DT; ++t) {
            int d = t * 64 + lane;
            part += qv[t] * ((d < DM) ? (float)k[(kvr * J + j) * DM + d] : 0.f);
        }
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
            part += __shfl_xor(part, off);
        // every lane now has the full dot; lane j keeps it
        float l = part * scale;
        if (mrow != nullptr && mrow[j] == 0) l = -3.0e38f;
        if (lane == j) logit = l;
    }

    // softmax across lanes 0..J-1
    float m = (lane < J) ? logit : -3.0e38f;
    float mx = m;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
    float e = (lane < J) ? __expf(m - mx) : 0.f;
    float se = e;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        se += __shfl_xor(se, off);
    float a = e / se;   // lane j: attention weight for key j

    // out = sum_j a_j * v_j
    float ov[DT];
#pragma unroll
    for (int t = 0; t < DT; ++t) ov[t] = 0.f;
    for (int j = 0; j < J; ++j) {
        float aj = __shfl(a, j);
#pragma unroll
        for (int t = 0; t < DT; ++t) {
            int d = t * 64 + lane;
            if (d < DM)
                ov[t] = fmaf(aj, (float)v[(kvr * J + j) * DM + d], ov[t]);
        }
    }
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        if (d < DM) out[r * DM + d] = ov[t];
    }
}

#define DISPATCH_DT(DT, ...)                                      \
    if (DT <= 1) { constexpr int kDT = 1; __VA_ARGS__; }          \
    else if (DT <= 2) { constexpr int kDT = 2; __VA_ARGS__; }     \
    else if (DT <= 4) { constexpr int kDT = 4; __VA_ARGS__; }     \
    else if (DT <= 7) { constexpr int kDT = 7; __VA_ARGS__; }     \
    else { TORCH_CHECK(false, "DM too large"); }

void attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
              torch::Tensor mask, torch::Tensor out,
              int64_t n, int64_t heads, double scale, bool kv_one) {
    TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                v.is_contiguous() && out.is_contiguous());
    long R = q.size(0);
    int DM = q.size(1), J = k.size(1);
    TORCH_CHECK(J <= 64, "fused attention supports <= 64 keys per query");
    TORCH_CHECK(out.dtype() == torch::kFloat32);
    const unsigned char* mptr = nullptr;
    if (mask.defined() && mask.numel() > 0) {
        TORCH_CHECK(mask.dtype() == torch::kUInt8 && mask.is_contiguous());
        mptr = mask.data_ptr<unsigned char>();
    }
    int DT = (DM + 63) / 64;
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((R + (NTA / 64) - 1) / (NTA / 64));
    DISPATCH_DT(DT, {
        if (q.dtype() == torch::kFloat32) {
            hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<float, kDT>),
                               grid, dim3(NTA), 0, stream,
                               q.data_ptr<float>(), k.data_ptr<float>(),
                               v.data_ptr<float>(), mptr, out.data_ptr<float>(),
                               R, J, DM, (int)n, (int)heads, (float)scale,
                               kv_one ? 1 : 0);
        } else {
            TORCH_CHECK(q.dtype() == torch::kBFloat16);
            hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<__hip_bfloat16, kDT>),
                               grid, dim3(NTA), 0, stream,
                               reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                               reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                               reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                               mptr, out.data_ptr<float>(),
                               R, J, DM, (int)n, (int)heads, (float)scale,
                               kv_one ? 1 : 0);
        }
    });
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "attn_fwd: ", hipGetErrorString(err));
}
