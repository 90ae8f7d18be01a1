// Batched masked cross-entropy (K10 of SURVEY.md §2b; reference:
// src/models/resnet.py:152-157 masked_fill(label_mask==0, 0) then CE).
//
// scores (N, R, C) fp32, labels (N, R) int64, mask (R, C) {0,1} or absent.
// Forward returns per-client mean NLL (R,) and also accumulates device-side
// training metrics (loss*count / correct / count) so the hipGraph-captured
// step needs no host work.  Backward: d_score = up_r/N * (softmax - onehot),
// zeroed where masked (masked_fill's gradient).
#include "common.h"

__global__ void __launch_bounds__(64)
masked_ce_fwd_kernel(const float* __restrict__ scores,
                     const long* __restrict__ labels,
                     const float* __restrict__ mask,
                     float* __restrict__ losses, float* __restrict__ metrics,
                     int N, int R, int C) {
    const int r = blockIdx.x;
    const int lane = threadIdx.x;
    float loss = 0.f;
    int correct = 0;
    // one lane per sample row (N <= 64 for HeteroFL batch shapes; loop else)
    for (int n = lane; n < N; n += blockDim.x) {
        const float* s = scores + ((long)n * R + r) * C;
        const long y = labels[(long)n * R + r];
        float mx = -1e30f;
        int arg = 0;
        for (int c = 0; c < C; ++c) {
            float v = s[c];
            if (mask) v = mask[r * C + c] != 0.f ? v : 0.f;
            if (v > mx) { mx = v; arg = c; }
        }
        float lse = 0.f;
        for (int c = 0; c < C; ++c) {
            float v = s[c];
            if (mask) v = mask[r * C + c] != 0.f ? v : 0.f;
            lse += __expf(v - mx);
        }
        lse = mx + __logf(lse);
        float vy = s[y];
        if (mask) vy = mask[r * C + y] != 0.f ? vy : 0.f;
        loss += lse - vy;
        correct += (arg == y);
    }
    // wave reduce
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        loss += __shfl_down(loss, off, WAVE);
        correct += __shfl_down(correct, off, WAVE);
    }
    if (lane == 0) {
        losses[r] = loss / N;
        if (metrics) {
            metrics[r * 3 + 0] += loss;         // sum NLL (= mean*count)
            metrics[r * 3 + 1] += correct;
            metrics[r * 3 + 2] += N;
        }
    }
}

__global__ void __launch_bounds__(256)
masked_ce_bwd_kernel(const float* __restrict__ scores,
                     const long* __restrict__ labels,
                     const float* __restrict__ mask,
                     const float* __restrict__ up,  // upstream d(losses), (R,)
                     float* __restrict__ dscores, int N, int R, int C) {
    const long i = blockIdx.x * blockDim.x + threadIdx.x;  // over N*R rows
    if (i >= (long)N * R) return;
    const int r = i % R;
    const float* s = scores + i * C;
    float* ds = dscores + i * C;
    const long y = labels[i];
    float mx = -1e30f;
    for (int c = 0; c < C; ++c) {
        float v = s[c];
        if (mask) v = mask[r * C + c] != 0.f ? v : 0.f;
        mx = fmaxf(mx, v);
    }
    float denom = 0.f;
    for (int c = 0; c < C; ++c) {
        float v = s[c];
        if (mask) v = mask[r * C + c] != 0.f ? v : 0.f;
        denom += __expf(v - mx);
    }
    const float scale = up[r] / N;
    for (int c = 0; c < C; ++c) {
        float v = s[c];
        bool live = true;
        if (mask) live = mask[r * C + c] != 0.f;
        v = live ? v : 0.f;
        float g = __expf(v - mx) / denom - (c == y ? 1.f : 0.f);
        ds[c] = live ? scale * g : 0.f;
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

std::vector<at::Tensor> masked_ce_fwd(at::Tensor scores, at::Tensor labels,
                                      at::Tensor mask, at::Tensor metrics) {
    TORCH_CHECK(scores.is_cuda() && scores.is_contiguous());
    TORCH_CHECK(scores.scalar_type() == at::kFloat);
    const int N = scores.size(0), R = scores.size(1), C = scores.size(2);
    auto losses = at::empty({R}, scores.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(masked_ce_fwd_kernel, dim3(R), dim3(64), 0, stream,
                       scores.data_ptr<float>(), labels.data_ptr<long>(),
                       mask.defined() ? mask.data_ptr<float>() : nullptr,
                       losses.data_ptr<float>(),
                       metrics.defined() ? metrics.data_ptr<float>() : nullptr,
                       N, R, C);
    return {losses};
}

at::Tensor masked_ce_bwd(at::Tensor scores, at::Tensor labels, at::Tensor mask,
                         at::Tensor up) {
    const int N = scores.size(0), R = scores.size(1), C = scores.size(2);
    auto dscores = at::empty_like(scores);
    const long rows = (long)N * R;
    const int threads = 256;
    const int blocks = (rows + threads - 1) / threads;
    auto upc = up.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(masked_ce_bwd_kernel, dim3(blocks), dim3(threads), 0,
                       stream, scores.data_ptr<float>(),
                       labels.data_ptr<long>(),
                       mask.defined() ? mask.data_ptr<float>() : nullptr,
                       upc.data_ptr<float>(), dscores.data_ptr<float>(), N, R,
                       C);
    return dscores;
}
