// Fused classifier head for the batched vision models (K6+K7 of SURVEY.md
// §2b): AdaptiveAvgPool(1) -> flatten -> per-client Linear in one kernel
// per direction (reference: src/models/resnet.py:148-151,
// src/models/conv.py:57-59).  Replaces pool + view + batched-GEMM (+ its
// transpose glue) with one launch each way — the head is tiny (C<=512,
// J<=100 classes, N<=64), so launch count is the cost, not FLOPs.
#include "common.h"

// feat (N, R*C, H, W) -> pooled (N, R, C) [saved for bwd], scores (N, R, J)
// scores = pooled @ W[r]^T + b[r];  W (R, J, C) fp32, b (R, J) fp32.
// N is tiled over blockIdx.y (TN rows per block) so the (TN, C) LDS pool
// slab stays within limits at any batch size — the sBN statistics / eval
// paths run this head at N=500, not just the train batch.
template <typename T>
__global__ void __launch_bounds__(256)
head_fwd_kernel(const T* __restrict__ feat, const float* __restrict__ w,
                const float* __restrict__ b, float* __restrict__ pooled,
                float* __restrict__ scores, int N, int R, int C, int HW,
                int J, int TN) {
    const int r = blockIdx.x;
    const int n0 = blockIdx.y * TN;
    const int nt = min(TN, N - n0);
    extern __shared__ float pool_lds[];  // (TN, C)
    const float inv_hw = 1.f / HW;
    // pool: each thread owns (n, c) pairs of this n-tile
    for (int e = threadIdx.x; e < nt * C; e += blockDim.x) {
        const int n = n0 + e / C, c = e - (e / C) * C;
        const T* src = feat + ((long)n * R * C + (long)r * C + c) * HW;
        float s = 0.f;
        for (int i = 0; i < HW; ++i) s += ld_f32(src + i);
        const float m = s * inv_hw;
        pool_lds[e] = m;
        pooled[((long)n * R + r) * C + c] = m;
    }
    __syncthreads();
    // scores: thread owns (n, j)
    for (int e = threadIdx.x; e < nt * J; e += blockDim.x) {
        const int n = n0 + e / J, j = e - (e / J) * J;
        const float* wr = w + ((long)r * J + j) * C;
        const float* pr = pool_lds + (n - n0) * C;
        float s = b ? b[r * J + j] : 0.f;
        for (int c = 0; c < C; ++c) s += pr[c] * wr[c];
        scores[((long)n * R + r) * J + j] = s;
    }
}

// dscores (N, R, J) -> dW (R, J, C), db (R, J), dfeat (N, R*C, H, W)
template <typename T>
__global__ void __launch_bounds__(256)
head_bwd_kernel(const float* __restrict__ dscores,
                const float* __restrict__ pooled,
                const float* __restrict__ w, float* __restrict__ dw,
                float* __restrict__ db, T* __restrict__ dfeat, int N, int R,
                int C, int HW, int J) {
    const int r = blockIdx.x;
    extern __shared__ float ds_lds[];  // (N, J)
    for (int e = threadIdx.x; e < N * J; e += blockDim.x)
        ds_lds[e] = dscores[((long)(e / J) * R + r) * J + (e - (e / J) * J)];
    __syncthreads();
    // dW[j][c] = sum_n ds[n][j] * pooled[n][c];  db[j] = sum_n ds[n][j]
    for (int e = threadIdx.x; e < J * C; e += blockDim.x) {
        const int j = e / C, c = e - j * C;
        float s = 0.f;
        for (int n = 0; n < N; ++n)
            s += ds_lds[n * J + j] * pooled[((long)n * R + r) * C + c];
        dw[((long)r * J + j) * C + c] = s;
    }
    if (db) {
        for (int j = threadIdx.x; j < J; j += blockDim.x) {
            float s = 0.f;
            for (int n = 0; n < N; ++n) s += ds_lds[n * J + j];
            db[r * J + j] = s;
        }
    }
    // dfeat[n][r*C+c][hw] = (sum_j ds[n][j] * W[j][c]) / HW
    const float inv_hw = 1.f / HW;
    for (int e = threadIdx.x; e < N * C; e += blockDim.x) {
        const int n = e / C, c = e - n * C;
        const float* wr = w + (long)r * J * C + c;
        float s = 0.f;
        for (int j = 0; j < J; ++j) s += ds_lds[n * J + j] * wr[(long)j * C];
        const float g = s * inv_hw;
        T* dst = dfeat + ((long)n * R * C + (long)r * C + c) * HW;
        for (int i = 0; i < HW; ++i) st_f32(dst + i, g);
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_HT(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> head_fwd(at::Tensor feat, at::Tensor w, at::Tensor b,
                                 int64_t R) {
    TORCH_CHECK(feat.is_cuda() && feat.is_contiguous() && w.is_contiguous());
    const int N = feat.size(0);
    const int C = (int)(feat.size(1) / R);
    const int HW = feat.numel() / (N * feat.size(1));
    const int J = w.size(1);
    auto opts = feat.options().dtype(at::kFloat);
    auto pooled = at::empty({N, (long)R, C}, opts);
    auto scores = at::empty({N, (long)R, J}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    // n-tile size: (TN, C) fp32 slab capped at 32 KB LDS
    const int TN = std::max(1, std::min(N, 8192 / C));
    const int lds = TN * C * sizeof(float);
    const int ny = (N + TN - 1) / TN;
    DISPATCH_HT(feat.scalar_type(), {
        hipLaunchKernelGGL(head_fwd_kernel<scalar_t>, dim3((int)R, ny),
                           dim3(256), lds, stream,
                           (const scalar_t*)feat.data_ptr(),
                           w.data_ptr<float>(),
                           b.defined() ? b.data_ptr<float>() : nullptr,
                           pooled.data_ptr<float>(), scores.data_ptr<float>(),
                           N, (int)R, C, HW, J, TN);
    });
    return {scores, pooled};
}

std::vector<at::Tensor> head_bwd(at::Tensor dscores, at::Tensor pooled,
                                 at::Tensor w, int64_t R, int64_t H,
                                 int64_t W, bool bf16_feat, bool want_db) {
    const auto feat_dtype = bf16_feat ? at::kBFloat16 : at::kFloat;
    const int N = dscores.size(0);
    const int J = dscores.size(2);
    const int C = pooled.size(2);
    const int HW = (int)(H * W);
    auto dsc = dscores.contiguous();
    auto opts = pooled.options();
    auto dw = at::empty({(long)R, J, C}, opts);
    auto db = want_db ? at::empty({(long)R, J}, opts) : at::Tensor();
    auto dfeat = at::empty({N, (long)R * C, H, W},
                           pooled.options().dtype(feat_dtype));
    auto stream = at::hip::getCurrentHIPStream();
    const int lds = N * J * sizeof(float);
    // fail loudly instead of poisoning the queue with an over-LDS launch
    // (the training path keeps N small; fwd tiles N, bwd does not)
    TORCH_CHECK(lds <= 64 * 1024,
                "head_bwd: N*J too large for the LDS stage (N=", N,
                ", J=", J, ")");
    DISPATCH_HT(feat_dtype, {
        hipLaunchKernelGGL(head_bwd_kernel<scalar_t>, dim3((int)R), dim3(256),
                           lds, stream, dsc.data_ptr<float>(),
                           pooled.data_ptr<float>(), w.data_ptr<float>(),
                           dw.data_ptr<float>(),
                           want_db ? db.data_ptr<float>() : nullptr,
                           (scalar_t*)dfeat.data_ptr(), N, (int)R, C, HW, J);
    });
    return {dw, db, dfeat};
}
