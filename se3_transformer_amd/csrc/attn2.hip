// Fused per-degree equivariant neighbor attention, v2 (reference
// se3_transformer_pytorch.py:488-517): masked softmax over the key axis
// (prefix null/global keys + self + k-NN neighbors), weighted value sum,
// with rotary position embeddings folded in for degree 0 — q, k AND v are
// rotated in-registers (the reference rotates all three, :488-494), so the
// Python side no longer materializes rotated copies.
//
// v2 over v1 (attn.hip): keys are processed in 64-wide tiles with an
// online (flash-style) softmax, so J is unbounded (radius-graph
// attention, num_neighbors=inf); the per-row logsumexp is saved and the
// backward is a HIP kernel too (one recompute pass; dv/dk written
// directly — per-row key tensors are exclusive — or atomically for the
// shared-KV one-headed variant).
//
// Layouts (prepared by ops/fused.py):
//   q       (R, DM)       R = b*h*n rows, DM = dim_head * (2l+1)
//   k, v    (Rkv, J, DM)  Rkv = R, or b*n when kv_one
//   mask    (b, n, J) u8  1 = attend; absent => all ones
//   qf      (b*n, rot)    rotary frequencies for queries (or empty)
//   kf      (b*n, jr, rot) rotary freqs for the LAST jr keys (or empty)
//   out     (R, DM) f32;  lse (R,) f32 saved for backward
// One wave per query row; lane j <-> key j within each 64-key tile; the
// DM axis cycles over lanes in DT chunks of 64.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#define NTA2 256  // 4 waves (= 4 query rows) per block

// rotary pairing of se3 rotary.py rotate_half (reference rotary.py:15-24):
//   out[d] = t[d] cos(f[d]) + sgn(d) * t[P(d)] sin(f[d])
//   P(d) = d < rot/2 ? 2d+1 : 2(d - rot/2);  sgn = -1 for d < rot/2 else +1
// transpose (for gradients):
//   dt[d] = dout[d] cos(f[d]) + sgnT(d) * dout[Q(d)] sin(f[Q(d)])
//   Q(d) = d odd ? (d-1)/2 : (d + rot)/2;   sgnT = -1 for d odd else +1
__device__ __forceinline__ int rope_P(int d, int rot) {
    return d < rot / 2 ? 2 * d + 1 : 2 * d - rot;
}
__device__ __forceinline__ int rope_Q(int d, int rot) {
    return (d & 1) ? (d - 1) / 2 : (d + rot) / 2;
}

// rotate the lane-held element x (dim d = lane, valid for d < rot <= 64)
// using wave shuffles; f = freq for this lane's dim, fq = freq at Q(d)
template <bool TRANS>
__device__ __forceinline__ float rope_apply(float x, int lane, int rot,
                                            float f, float fq) {
    float partner = __shfl(x, TRANS ? rope_Q(lane, rot) : rope_P(lane, rot));
    if (lane >= rot) return x;
    float sgn = TRANS ? ((lane & 1) ? -1.f : 1.f)
                      : ((lane < rot / 2) ? -1.f : 1.f);
    return x * __cosf(f) + sgn * partner * __sinf(TRANS ? fq : f);
}

template <typename T, int DT>
__global__ void __launch_bounds__(NTA2)
attn2_fwd_kernel(const T* __restrict__ q, const T* __restrict__ k,
                 const T* __restrict__ v, const unsigned char* __restrict__ mask,
                 const float* __restrict__ qf, const float* __restrict__ kf,
                 float* __restrict__ out, float* __restrict__ lse,
                 long R, int J, int DM, int n, int heads, float scale,
                 int kv_one, int jr, int rot) {
    const int lane = threadIdx.x & 63;
    const long r = ((long)blockIdx.x * (NTA2 / 64)) + (threadIdx.x >> 6);
    if (r >= R) return;

    const long b_ = r / ((long)heads * n);
    const long i_ = r % n;
    const long bi = b_ * n + i_;
    const long kvr = kv_one ? bi : r;
    const unsigned char* mrow = mask ? mask + bi * J : nullptr;

    // q row in per-lane chunks (+ rotary on chunk 0)
    float qv[DT];
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        qv[t] = (d < DM) ? (float)q[r * DM + d] : 0.f;
    }
    if (qf != nullptr) {
        float f = (lane < rot) ? qf[bi * rot + lane] : 0.f;
        qv[0] = rope_apply<false>(qv[0], lane, rot, f, 0.f);
    }

    float m = -3.0e38f, s = 0.f;
    float ov[DT];
#pragma unroll
    for (int t = 0; t < DT; ++t) ov[t] = 0.f;

    for (int j0 = 0; j0 < J; j0 += 64) {
        const int tj = min(64, J - j0);
        // key rotary freq for THIS lane's key (lane j <-> key j0+lane)
        // logits: lane j holds logit for key j0+j
        float logit = -3.0e38f;
        for (int jj = 0; jj < tj; ++jj) {
            const int j = j0 + jj;
            float kfre = 0.f;
            const bool roped = (kf != nullptr) && (j >= J - jr);
            float part = 0.f;
#pragma unroll
            for (int t = 0; t < DT; ++t) {
                int d = t * 64 + lane;
                float kv_ = (d < DM) ? (float)k[(kvr * J + j) * DM + d] : 0.f;
                if (t == 0 && roped) {
                    float f = (lane < rot)
                        ? kf[(bi * jr + (j - (J - jr))) * rot + lane] : 0.f;
                    kv_ = rope_apply<false>(kv_, lane, rot, f, 0.f);
                }
                part += qv[t] * kv_;
            }
#pragma unroll
            for (int off = 32; off > 0; off >>= 1)
                part += __shfl_xor(part, off);
            float l = part * scale;
            if (mrow != nullptr && mrow[j] == 0) l = -3.0e38f;
            if (lane == jj) logit = l;
        }
        // online softmax update over this tile
        float tm = (lane < tj) ? logit : -3.0e38f;
        float mx = tm;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
            mx = fmaxf(mx, __shfl_xor(mx, off));
        const float mn = fmaxf(m, mx);
        const float alpha = __expf(m - mn);
        float p = (lane < tj) ? __expf(tm - mn) : 0.f;
        float ps = p;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
            ps += __shfl_xor(ps, off);
        s = s * alpha + ps;
        m = mn;
#pragma unroll
        for (int t = 0; t < DT; ++t) ov[t] *= alpha;
        for (int jj = 0; jj < tj; ++jj) {
            const int j = j0 + jj;
            const float aj = __shfl(p, jj);
            if (aj == 0.f) continue;
            const bool roped = (kf != nullptr) && (j >= J - jr);
#pragma unroll
            for (int t = 0; t < DT; ++t) {
                int d = t * 64 + lane;
                if (d >= DM) continue;
                float vv = (float)v[(kvr * J + j) * DM + d];
                if (t == 0 && roped) {
                    float f = (lane < rot)
                        ? kf[(bi * jr + (j - (J - jr))) * rot + lane] : 0.f;
                    vv = rope_apply<false>(vv, lane, rot, f, 0.f);
                }
                ov[t] = fmaf(aj, vv, ov[t]);
            }
        }
    }

    const float inv_s = 1.f / s;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        if (d < DM) out[r * DM + d] = ov[t] * inv_s;
    }
    if (lane == 0 && lse != nullptr) lse[r] = m + __logf(s);
}

// ---------------------------------------------------------------------------
// backward: one recompute pass per row using the saved logsumexp.
//   a_j = exp(sim_j - lse); delta = dot(g, out)
//   dv_j = a_j g;  ds_j = a_j (dot(g, v_j) - delta)
//   dq += scale * sum_j ds_j k_j;  dk_j = scale * ds_j q
// (all in the ROTATED frame; dq/dk/dv are un-rotated — transpose map —
// before being written)
// ---------------------------------------------------------------------------
template <typename T, int DT, bool ATOMIC>
__global__ void __launch_bounds__(NTA2)
attn2_bwd_kernel(const T* __restrict__ q, const T* __restrict__ k,
                 const T* __restrict__ v, const unsigned char* __restrict__ mask,
                 const float* __restrict__ qf, const float* __restrict__ kf,
                 const float* __restrict__ out, const float* __restrict__ lse,
                 const float* __restrict__ g,
                 float* __restrict__ dq, float* __restrict__ dk,
                 float* __restrict__ dv,
                 long R, int J, int DM, int n, int heads, float scale,
                 int kv_one, int jr, int rot) {
    const int lane = threadIdx.x & 63;
    const long r = ((long)blockIdx.x * (NTA2 / 64)) + (threadIdx.x >> 6);
    if (r >= R) return;

    const long b_ = r / ((long)heads * n);
    const long i_ = r % n;
    const long bi = b_ * n + i_;
    const long kvr = kv_one ? bi : r;
    const unsigned char* mrow = mask ? mask + bi * J : nullptr;
    const float l_row = lse[r];

    float qv[DT], gv[DT], dqv[DT];
    float delta = 0.f;
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        qv[t] = (d < DM) ? (float)q[r * DM + d] : 0.f;
        gv[t] = (d < DM) ? g[r * DM + d] : 0.f;
        dqv[t] = 0.f;
        delta += gv[t] * ((d < DM) ? out[r * DM + d] : 0.f);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        delta += __shfl_xor(delta, off);
    float fq_lane = 0.f;
    if (qf != nullptr) {
        fq_lane = (lane < rot) ? qf[bi * rot + lane] : 0.f;
        qv[0] = rope_apply<false>(qv[0], lane, rot, fq_lane, 0.f);
    }

    for (int j = 0; j < J; ++j) {
        const bool masked = (mrow != nullptr && mrow[j] == 0);
        const bool roped = (kf != nullptr) && (j >= J - jr);
        float fk = 0.f;
        if (roped && lane < rot)
            fk = kf[(bi * jr + (j - (J - jr))) * rot + lane];
        // sim_j and dot(g, v_j)
        float part = 0.f, gvdot = 0.f;
        float kvv[DT], vvv[DT];
#pragma unroll
        for (int t = 0; t < DT; ++t) {
            int d = t * 64 + lane;
            kvv[t] = (d < DM) ? (float)k[(kvr * J + j) * DM + d] : 0.f;
            vvv[t] = (d < DM) ? (float)v[(kvr * J + j) * DM + d] : 0.f;
            if (t == 0 && roped) {
                kvv[0] = rope_apply<false>(kvv[0], lane, rot, fk, 0.f);
                vvv[0] = rope_apply<false>(vvv[0], lane, rot, fk, 0.f);
            }
            part += qv[t] * kvv[t];
            gvdot += gv[t] * vvv[t];
        }
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            part += __shfl_xor(part, off);
            gvdot += __shfl_xor(gvdot, off);
        }
        float sim = part * scale;
        if (masked) sim = -3.0e38f;
        const float aj = __expf(sim - l_row);
        const float dsj = aj * (gvdot - delta) * scale;
        // dq accumulation (rotated frame)
#pragma unroll
        for (int t = 0; t < DT; ++t) dqv[t] = fmaf(dsj, kvv[t], dqv[t]);
        // dk_j / dv_j: un-rotate, then write
        float dkj[DT], dvj[DT];
#pragma unroll
        for (int t = 0; t < DT; ++t) {
            dkj[t] = dsj * qv[t];
            dvj[t] = aj * gv[t];
        }
        if (roped) {
            float fqk = (rope_Q(lane, rot) < rot && lane < rot)
                ? kf[(bi * jr + (j - (J - jr))) * rot + rope_Q(lane, rot)] : 0.f;
            dkj[0] = rope_apply<true>(dkj[0], lane, rot, fk, fqk);
            dvj[0] = rope_apply<true>(dvj[0], lane, rot, fk, fqk);
        }
#pragma unroll
        for (int t = 0; t < DT; ++t) {
            int d = t * 64 + lane;
            if (d >= DM) continue;
            if (ATOMIC) {
                atomicAdd(&dk[(kvr * J + j) * DM + d], dkj[t]);
                atomicAdd(&dv[(kvr * J + j) * DM + d], dvj[t]);
            } else {
                dk[(kvr * J + j) * DM + d] = dkj[t];
                dv[(kvr * J + j) * DM + d] = dvj[t];
            }
        }
    }

    // un-rotate dq and write
    if (qf != nullptr) {
        float fqq = (rope_Q(lane, rot) < rot && lane < rot)
            ? qf[bi * rot + rope_Q(lane, rot)] : 0.f;
        dqv[0] = rope_apply<true>(dqv[0], lane, rot, fq_lane, fqq);
    }
#pragma unroll
    for (int t = 0; t < DT; ++t) {
        int d = t * 64 + lane;
        if (d < DM) dq[r * DM + d] = dqv[t];
    }
}

// ---------------------------------------------------------------------------
#define A2_DISPATCH(T, DT, ...)                                        \
    if (DT <= 1) { constexpr int kDT = 1; __VA_ARGS__; }               \
    else if (DT <= 2) { constexpr int kDT = 2; __VA_ARGS__; }          \
    else if (DT <= 4) { constexpr int kDT = 4; __VA_ARGS__; }          \
    else if (DT <= 7) { constexpr int kDT = 7; __VA_ARGS__; }          \
    else { TORCH_CHECK(false, "DM too large for attn2"); }

static const unsigned char* opt_mask(const torch::Tensor& m) {
    if (m.defined() && m.numel() > 0) {
        TORCH_CHECK(m.dtype() == torch::kUInt8 && m.is_contiguous());
        return m.data_ptr<unsigned char>();
    }
    return nullptr;
}
static const float* opt_f32(const torch::Tensor& t) {
    if (t.defined() && t.numel() > 0) {
        TORCH_CHECK(t.dtype() == torch::kFloat32 && t.is_contiguous());
        return t.data_ptr<float>();
    }
    return nullptr;
}

void attn2_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
               torch::Tensor mask, torch::Tensor qf, torch::Tensor kf,
               torch::Tensor out, torch::Tensor lse,
               int64_t n, int64_t heads, double scale, bool kv_one,
               int64_t jr, int64_t rot) {
    TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                v.is_contiguous() && out.is_contiguous());
    long R = q.size(0);
    int DM = q.size(1), J = k.size(1);
    TORCH_CHECK(out.dtype() == torch::kFloat32 && lse.numel() == R);
    TORCH_CHECK(rot <= 64 && rot % 2 == 0, "rotary dims must be even and <= 64");
    int DT = (DM + 63) / 64;
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((R + (NTA2 / 64) - 1) / (NTA2 / 64));
    const unsigned char* mp = opt_mask(mask);
    const float* qfp = opt_f32(qf);
    const float* kfp = opt_f32(kf);
#define A2_LAUNCH_F(T)                                                          \
    A2_DISPATCH(T, DT, {                                                        \
        hipLaunchKernelGGL(HIP_KERNEL_NAME(attn2_fwd_kernel<T, kDT>),           \
                           grid, dim3(NTA2), 0, stream,                         \
                           reinterpret_cast<const T*>(q.data_ptr()),            \
                           reinterpret_cast<const T*>(k.data_ptr()),            \
                           reinterpret_cast<const T*>(v.data_ptr()),            \
                           mp, qfp, kfp, out.data_ptr<float>(),                 \
                           lse.data_ptr<float>(), R, J, DM, (int)n,             \
                           (int)heads, (float)scale, kv_one ? 1 : 0,            \
                           (int)jr, (int)rot);                                  \
    })
    if (q.dtype() == torch::kFloat32) { A2_LAUNCH_F(float); }
    else {
        TORCH_CHECK(q.dtype() == torch::kBFloat16);
        A2_LAUNCH_F(__hip_bfloat16);
    }
#undef A2_LAUNCH_F
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "attn2_fwd: ", hipGetErrorString(err));
}

void attn2_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
               torch::Tensor mask, torch::Tensor qf, torch::Tensor kf,
               torch::Tensor out, torch::Tensor lse, torch::Tensor g,
               torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
               int64_t n, int64_t heads, double scale, bool kv_one,
               int64_t jr, int64_t rot) {
    long R = q.size(0);
    int DM = q.size(1), J = k.size(1);
    TORCH_CHECK(g.is_contiguous() && g.dtype() == torch::kFloat32);
    TORCH_CHECK(dq.dtype() == torch::kFloat32 && dk.dtype() == torch::kFloat32 &&
                dv.dtype() == torch::kFloat32);
    int DT = (DM + 63) / 64;
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((R + (NTA2 / 64) - 1) / (NTA2 / 64));
    const unsigned char* mp = opt_mask(mask);
    const float* qfp = opt_f32(qf);
    const float* kfp = opt_f32(kf);
#define A2_LAUNCH_B(T, AT)                                                      \
    A2_DISPATCH(T, DT, {                                                        \
        hipLaunchKernelGGL(HIP_KERNEL_NAME(attn2_bwd_kernel<T, kDT, AT>),       \
                           grid, dim3(NTA2), 0, stream,                         \
                           reinterpret_cast<const T*>(q.data_ptr()),            \
                           reinterpret_cast<const T*>(k.data_ptr()),            \
                           reinterpret_cast<const T*>(v.data_ptr()),            \
                           mp, qfp, kfp, out.data_ptr<float>(),                 \
                           lse.data_ptr<float>(), g.data_ptr<float>(),          \
                           dq.data_ptr<float>(), dk.data_ptr<float>(),          \
                           dv.data_ptr<float>(), R, J, DM, (int)n,              \
                           (int)heads, (float)scale, kv_one ? 1 : 0,            \
                           (int)jr, (int)rot);                                  \
    })
    if (q.dtype() == torch::kFloat32) {
        if (kv_one) { A2_LAUNCH_B(float, true); }
        else { A2_LAUNCH_B(float, false); }
    } else {
        TORCH_CHECK(q.dtype() == torch::kBFloat16);
        if (kv_one) { A2_LAUNCH_B(__hip_bfloat16, true); }
        else { A2_LAUNCH_B(__hip_bfloat16, false); }
    }
#undef A2_LAUNCH_B
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "attn2_bwd: ", hipGetErrorString(err));
}
