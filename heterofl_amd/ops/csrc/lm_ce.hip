// Fused masked-LM cross-entropy over the full vocabulary (K10 LM flavor of
// SURVEY.md §2b; reference: src/models/transformer.py:156-161 — vocab-masked
// logits then CE over every position).  logits (ROWS, V) with
// ROWS = R*B*S and V ~ 33k; one block per row computes the masked
// max/log-sum-exp, a tiny reduce averages per client, and the backward is a
// single elementwise-with-row-state kernel.  Replaces the eager fp32-cast +
// masked_fill + log_softmax + gather + mean chain (~12 launches on a 42 MB
// tensor) with 2 forward + 1 backward launches, logits staying bf16.
#include "common.h"

template <typename T>
__global__ void __launch_bounds__(256)
lm_ce_row_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                 const float* __restrict__ mask, float* __restrict__ row_nll,
                 float* __restrict__ row_lse, int rows_per_client, int V) {
    const int row = blockIdx.x;
    const int r = row / rows_per_client;
    const T* s = logits + (long)row * V;
    const float* mk = mask ? mask + (long)r * V : nullptr;
    float mx = -1e30f;
    for (int j = threadIdx.x; j < V; j += blockDim.x) {
        float v = ld_f32(s + j);
        if (mk) v = mk[j] != 0.f ? v : 0.f;
        mx = fmaxf(mx, v);
    }
    __shared__ float scratch[256 / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_down(mx, off, WAVE));
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = mx;
    __syncthreads();
    if (threadIdx.x < 256 / WAVE) mx = scratch[threadIdx.x];
    for (int off = 2; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_down(mx, off, WAVE));
    mx = __shfl(mx, 0, WAVE);
    if (threadIdx.x == 0) scratch[0] = mx;
    __syncthreads();
    mx = scratch[0];
    __syncthreads();
    float sum = 0.f;
    for (int j = threadIdx.x; j < V; j += blockDim.x) {
        float v = ld_f32(s + j);
        if (mk) v = mk[j] != 0.f ? v : 0.f;
        sum += __expf(v - mx);
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        sum += __shfl_down(sum, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = sum;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.f;
        for (int w = 0; w < 256 / WAVE; ++w) t += scratch[w];
        const float lse = mx + __logf(t);
        const long y = labels[row];
        float vy = ld_f32(s + y);
        if (mk) vy = mk[y] != 0.f ? vy : 0.f;
        row_nll[row] = lse - vy;
        row_lse[row] = lse;
    }
}

__global__ void __launch_bounds__(64)
lm_ce_reduce_kernel(const float* __restrict__ row_nll,
                    float* __restrict__ losses, int rows_per_client) {
    const int r = blockIdx.x;
    float s = 0.f;
    for (int i = threadIdx.x; i < rows_per_client; i += 64)
        s += row_nll[(long)r * rows_per_client + i];
    for (int off = WAVE / 2; off > 0; off >>= 1)
        s += __shfl_down(s, off, WAVE);
    if (threadIdx.x == 0) losses[r] = s / rows_per_client;
}

template <typename T>
__global__ void __launch_bounds__(256)
lm_ce_bwd_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                 const float* __restrict__ mask,
                 const float* __restrict__ row_lse,
                 const float* __restrict__ up, T* __restrict__ dlogits,
                 int rows_per_client, int V) {
    const int row = blockIdx.x;
    const int r = row / rows_per_client;
    const T* s = logits + (long)row * V;
    T* ds = dlogits + (long)row * V;
    const float* mk = mask ? mask + (long)r * V : nullptr;
    const float lse = row_lse[row];
    const long y = labels[row];
    const float scale = up[r] / rows_per_client;
    for (int j = threadIdx.x; j < V; j += blockDim.x) {
        float v = ld_f32(s + j);
        bool live = true;
        if (mk) live = mk[j] != 0.f;
        v = live ? v : 0.f;
        float g = __expf(v - lse) - (j == y ? 1.f : 0.f);
        st_f32(ds + j, live ? scale * g : 0.f);
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_CT(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> lm_ce_fwd(at::Tensor logits, at::Tensor labels,
                                  at::Tensor mask, int64_t R) {
    TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
    const int V = logits.size(-1);
    const long rows = logits.numel() / V;
    TORCH_CHECK(rows % R == 0);
    auto opts = logits.options().dtype(at::kFloat);
    auto row_nll = at::empty({rows}, opts);
    auto row_lse = at::empty({rows}, opts);
    auto losses = at::empty({R}, opts);
    auto lab = labels.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_CT(logits.scalar_type(), {
        hipLaunchKernelGGL(lm_ce_row_kernel<scalar_t>, dim3((int)rows),
                           dim3(256), 0, stream,
                           (const scalar_t*)logits.data_ptr(),
                           lab.data_ptr<long>(),
                           mask.defined() ? mask.data_ptr<float>() : nullptr,
                           row_nll.data_ptr<float>(),
                           row_lse.data_ptr<float>(), (int)(rows / R), V);
    });
    hipLaunchKernelGGL(lm_ce_reduce_kernel, dim3((int)R), dim3(64), 0,
                       stream, row_nll.data_ptr<float>(),
                       losses.data_ptr<float>(), (int)(rows / R));
    return {losses, row_lse};
}

at::Tensor lm_ce_bwd(at::Tensor logits, at::Tensor labels, at::Tensor mask,
                     at::Tensor row_lse, at::Tensor up, int64_t R) {
    const int V = logits.size(-1);
    const long rows = logits.numel() / V;
    auto dlogits = at::empty_like(logits);
    auto lab = labels.contiguous();
    auto upc = up.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_CT(logits.scalar_type(), {
        hipLaunchKernelGGL(lm_ce_bwd_kernel<scalar_t>, dim3((int)rows),
                           dim3(256), 0, stream,
                           (const scalar_t*)logits.data_ptr(),
                           lab.data_ptr<long>(),
                           mask.defined() ? mask.data_ptr<float>() : nullptr,
                           row_lse.data_ptr<float>(),
                           upc.data_ptr<float>(),
                           (scalar_t*)dlogits.data_ptr(), (int)(rows / R),
                           V);
    });
    return dlogits;
}
