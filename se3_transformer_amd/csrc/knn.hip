// On-device k-NN graph construction (reference se3_transformer_pytorch.py
// :1221-1294): replaces the dense (b,n,n,3) rel-pos tensor, the self-removal
// gathers, the norm, the masked topk and the three neighbor gathers with one
// kernel. One wave per query node: each lane keeps a sorted top-L list of
// its strided j-candidates in LDS (L = min(k, ceil(n/64)) — a lane can
// never contribute more than its own candidate count), then the 64 lists
// are merged with a k-round cross-lane argmin; the k winners' relative
// geometry and validity mask are written directly.
//
// Selection semantics match the eager path exactly (:1250-1283):
//   * self excluded; causal keeps only j < i (overrides everything);
//   * `allow` (the user neighbor_mask, (b,n,n)) removes candidates from
//     selection (eager gives them +max distance => only unmaskable
//     padding, which comes out mask=0 either way);
//   * `sparse` (attend_sparse_neighbors adjacency, (b,n,n)) gives its
//     candidates priority distance 0 and exempts them from the radius
//     test — and from `allow`, as in the eager fill order;
//   * the node validity mask ANDs into the output mask only.
// Invalid slots emit index 0/1 with ZERO geometry (consumers that ignore
// the mask see no fabricated edge).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

#define MAXK2 64

__global__ void __launch_bounds__(64)
knn_kernel(const float* __restrict__ coors,        // (b, n, 3)
           const unsigned char* __restrict__ nmask, // (b, n) or nullptr
           const unsigned char* __restrict__ allow, // (b, n, n) or nullptr
           const unsigned char* __restrict__ sparse,// (b, n, n) or nullptr
           long* __restrict__ out_idx,              // (b, n, k)
           float* __restrict__ out_dist,            // (b, n, k)
           float* __restrict__ out_rel,             // (b, n, k, 3)
           unsigned char* __restrict__ out_m,       // (b, n, k)
           int b, int n, int k, int L, float radius, int causal) {
    extern __shared__ __attribute__((aligned(8))) char smem[];
    float* ld = reinterpret_cast<float*>(smem);            // [L][64]
    int* li = reinterpret_cast<int*>(smem + (size_t)L * 64 * 4); // [L][64]
    __shared__ int sel[MAXK2];

    const int lane = threadIdx.x & 63;
    const long q = blockIdx.x;                   // query row = bi*n + i
    if (q >= (long)b * n) return;
    const int bi = (int)(q / n), i = (int)(q % n);

    const float xi = coors[q * 3], yi = coors[q * 3 + 1], zi = coors[q * 3 + 2];
    const unsigned char* arow = allow ? allow + q * n : nullptr;
    const unsigned char* srow = sparse ? sparse + q * n : nullptr;

    int cnt = 0;  // entries in this lane's list (sorted ascending by eff)
    for (int s = 0; s < L; ++s) { ld[(size_t)s * 64 + lane] = 3.0e38f; li[(size_t)s * 64 + lane] = 0; }
    for (int j = lane; j < n; j += 64) {
        if (j == i) continue;
        if (causal && j >= i) continue;
        const bool sp = srow && srow[j];
        if (!sp && arow && arow[j] == 0) continue;
        float eff = 0.f;
        if (!sp) {
            float dx = xi - coors[((long)bi * n + j) * 3];
            float dy = yi - coors[((long)bi * n + j) * 3 + 1];
            float dz = zi - coors[((long)bi * n + j) * 3 + 2];
            eff = dx * dx + dy * dy + dz * dz;
        }
        if (cnt == L && eff >= ld[(size_t)(L - 1) * 64 + lane]) continue;
        // insertion sort (ascending)
        int p = (cnt < L) ? cnt : L - 1;
        while (p > 0 && ld[(size_t)(p - 1) * 64 + lane] > eff) {
            ld[(size_t)p * 64 + lane] = ld[(size_t)(p - 1) * 64 + lane];
            li[(size_t)p * 64 + lane] = li[(size_t)(p - 1) * 64 + lane];
            --p;
        }
        ld[(size_t)p * 64 + lane] = eff;
        li[(size_t)p * 64 + lane] = j;
        if (cnt < L) ++cnt;
    }

    // merge the 64 sorted lists: k rounds of cross-lane argmin over heads
    int head = 0;
    for (int r = 0; r < k; ++r) {
        float v = (head < cnt) ? ld[(size_t)head * 64 + lane] : 3.0e38f;
        int l = lane;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            float ov = __shfl_xor(v, off);
            int ol = __shfl_xor(l, off);
            if (ov < v || (ov == v && ol < l)) { v = ov; l = ol; }
        }
        if (lane == l) {
            // v == +inf means no candidate anywhere (e.g. causal row 0):
            // emit -1 so the slot comes out invalid.
            sel[r] = (v < 3.0e38f) ? li[(size_t)head * 64 + lane] : -1;
            ++head;
        }
    }
    __builtin_amdgcn_wave_barrier();

    // lanes 0..k-1: write the selected neighbors' geometry + validity
    if (lane < k) {
        int j = sel[lane];
        bool valid = j >= 0;
        if (!valid) j = (i == 0) ? (n > 1 ? 1 : 0) : 0;  // any real node != i
        float dx = 0.f, dy = 0.f, dz = 0.f, d = 0.f;
        bool sp = false;
        if (valid) {
            dx = xi - coors[((long)bi * n + j) * 3];
            dy = yi - coors[((long)bi * n + j) * 3 + 1];
            dz = zi - coors[((long)bi * n + j) * 3 + 2];
            d = sqrtf(dx * dx + dy * dy + dz * dz);
            sp = srow && srow[j];
        }
        long o = q * k + lane;
        out_idx[o] = j;
        out_dist[o] = d;
        out_rel[o * 3] = dx;
        out_rel[o * 3 + 1] = dy;
        out_rel[o * 3 + 2] = dz;
        unsigned char ok = valid && (sp || d <= radius);
        if (nmask != nullptr)
            ok = ok & nmask[(long)bi * n + i] & nmask[(long)bi * n + j];
        out_m[o] = ok;
    }
}

static const unsigned char* knn_opt_u8(const torch::Tensor& t) {
    if (t.defined() && t.numel() > 0) {
        TORCH_CHECK(t.dtype() == torch::kUInt8 && t.is_contiguous());
        return t.data_ptr<unsigned char>();
    }
    return nullptr;
}

void knn_graph(torch::Tensor coors, torch::Tensor nmask, torch::Tensor allow,
               torch::Tensor sparse, torch::Tensor idx,
               torch::Tensor dist, torch::Tensor rel, torch::Tensor m,
               int64_t k, double radius, bool causal) {
    TORCH_CHECK(coors.is_cuda() && coors.dtype() == torch::kFloat32 &&
                coors.is_contiguous());
    int b = coors.size(0), n = coors.size(1);
    TORCH_CHECK(k >= 1 && k <= MAXK2 && k <= n - 1);
    TORCH_CHECK(idx.is_contiguous() && dist.is_contiguous() &&
                rel.is_contiguous() && m.is_contiguous());
    int L = (int)std::min<long>(k, (n + 63) / 64);
    auto stream = at::cuda::getCurrentHIPStream();
    dim3 grid((long)b * n);
    size_t lds = (size_t)L * 64 * 8;
    hipLaunchKernelGGL(knn_kernel, grid, dim3(64), lds, stream,
                       coors.data_ptr<float>(), knn_opt_u8(nmask),
                       knn_opt_u8(allow), knn_opt_u8(sparse),
                       idx.data_ptr<long>(),
                       dist.data_ptr<float>(), rel.data_ptr<float>(),
                       m.data_ptr<unsigned char>(), b, n, (int)k, L,
                       (float)radius, causal ? 1 : 0);
    hipError_t err = hipGetLastError();
    TORCH_CHECK(err == hipSuccess, "knn_graph: ", hipGetErrorString(err));
}
