// Fused spherical-harmonics + equivariant-basis kernel.
//
// Replaces the reference's per-(l,m) recursive SH evaluation + per-(J, pair)
// Y_J @ Q_J^T einsum chain (spherical_harmonics.py:35-123, basis.py:140-198)
// with ONE pass per edge: cartesian -> (ct, st, cp, sp) -> all Y_l^m via
// Legendre/Chebyshev recurrences (Y staged in LDS as per-thread indexable
// scratch) -> every degree-pair's packed basis K[o,i,f] = sum_m Y_J[m] Q_J^T[m,(o,i)]
// written in the (E, O, I, F) layout the fused conv kernels consume.
//
// Convention parity with ops/sh.py (reference basis.py:57-95, irr_repr.py:103):
//   x_sh, y_sh, z_sh = z, x, y; ct = -z_sh/r; st = rho/r; cp = x_sh/rho; sp = y_sh/rho.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

#define NTSH 256

__global__ void __launch_bounds__(NTSH)
sh_basis_kernel(const float* __restrict__ rel,     // (E, 3)
                const float* __restrict__ qcat,    // concatenated Q_J^T tables
                const float* __restrict__ normtab, // [(L+1)(L+2)/2] norm consts
                const int4* __restrict__ meta,     // per pair: (di, do, off_out, off_q)
                int npairs, float* __restrict__ out, int E, int TOT, int L) {
    extern __shared__ __attribute__((aligned(16))) float y_lds[]; // [(L+1)^2][NTSH]
    const int tid = threadIdx.x;
    const int e = blockIdx.x * NTSH + tid;
    if (e >= E) return;

    const float eps2 = 1e-24f;
    float x_sh = rel[(size_t)e * 3 + 2];
    float y_sh = rel[(size_t)e * 3 + 0];
    float z_sh = rel[(size_t)e * 3 + 1];
    float rho2 = x_sh * x_sh + y_sh * y_sh;
    float r2 = rho2 + z_sh * z_sh;
    float rho = sqrtf(fmaxf(rho2, eps2));
    float r = sqrtf(fmaxf(r2, eps2));
    bool tiny_rho = rho2 <= eps2, tiny_r = r2 <= eps2;
    float ct = tiny_r ? -1.f : -z_sh / r;
    float st = tiny_r ? 0.f : rho / r;
    float cp = tiny_rho ? 1.f : x_sh / rho;
    float sp = tiny_rho ? 0.f : y_sh / rho;

    // all Y_l^m -> y_lds[s][tid], s = l*l + l + m
    float pmm = 1.f, cm = 1.f, sm = 0.f;
    for (int m = 0; m <= L; ++m) {
        if (m > 0) {
            pmm *= st * (float)(-(2 * m - 1));
            float cm_new = cm * cp - sm * sp;
            sm = sm * cp + cm * sp;
            cm = cm_new;
        }
        // upward recursion in l at fixed m: P_m^m, P_{m+1}^m, ...
        float p_prev2 = 0.f, p_prev = pmm;
        for (int l = m; l <= L; ++l) {
            float p;
            if (l == m) p = pmm;
            else if (l == m + 1) p = (float)(2 * m + 1) * ct * pmm;
            else p = ((float)(2 * l - 1) * ct * p_prev - (float)(l + m - 1) * p_prev2)
                     / (float)(l - m);
            if (l > m) { p_prev2 = p_prev; p_prev = p; }
            float n = normtab[l * (l + 1) / 2 + m];
            if (m == 0) {
                y_lds[(l * l + l) * NTSH + tid] = n * p;
            } else {
                y_lds[(l * l + l + m) * NTSH + tid] = n * p * cm;
                y_lds[(l * l + l - m) * NTSH + tid] = n * p * sm;
            }
        }
    }

    // per-pair basis: K[o,i,f] = sum_mj Y[J^2 + mj] * q[mj*(O*I) + o*I + i]
    float* orow = out + (size_t)e * TOT;
    for (int p = 0; p < npairs; ++p) {
        int4 mt = meta[p];
        int di = mt.x, dout = mt.y;
        int O = 2 * dout + 1, I = 2 * di + 1, OI = O * I;
        int Jlo = abs(di - dout), Jhi = di + dout;
        int F = Jhi - Jlo + 1;
        const float* q = qcat + mt.w;
        float* ob = orow + mt.z;
        for (int J = Jlo; J <= Jhi; ++J) {
            int f = J - Jlo;
            int nm = 2 * J + 1;
            for (int oi = 0; oi < OI; ++oi) {
                float acc = 0.f;
                for (int mj = 0; mj < nm; ++mj)
                    acc = fmaf(y_lds[(J * J + mj) * NTSH + tid], q[mj * OI + oi], acc);
                ob[oi * F + f] = acc;
            }
            q += nm * OI;
        }
    }
}

void sh_basis_fwd(torch::Tensor rel, torch::Tensor qcat, torch::Tensor normtab,
                  torch::Tensor meta, torch::Tensor out, int64_t L) {
    TORCH_CHECK(rel.is_cuda() && rel.dtype() == torch::kFloat32 && rel.is_contiguous());
    TORCH_CHECK(qcat.is_contiguous() && normtab.is_contiguous() &&
                meta.is_contiguous() && out.is_contiguous());
    TORCH_CHECK(meta.dtype() == torch::kInt32 && meta.size(1) == 4);
    int E = rel.size(0);
    int TOT = out.size(1);
    int npairs = meta.size(0);
    TORCH_CHECK(out.size(0) == E);
    int S = (int)((L + 1) * (L + 1));
    size_t lds = (size_t)S * NTSH * sizeof(float);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((E + NTSH - 1) / NTSH);
    hipLaunchKernelGGL(sh_basis_kernel, grid, dim3(NTSH), lds, stream,
                       rel.data_ptr<float>(), qcat.data_ptr<float>(),
                       normtab.data_ptr<float>(),
                       reinterpret_cast<const int4*>(meta.data_ptr<int>()),
                       npairs, out.data_ptr<float>(), E, TOT, (int)L);
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "sh_basis: ", hipGetErrorString(err));
}
