// Fused per-client grad-clip + momentum-SGD (K12 of SURVEY.md §2b;
// reference: src/train_classifier_fed.py:205-206 clip_grad_norm_(params, 1)
// then SGD(momentum=0.9, weight_decay=5e-4); src/utils.py:261-263).
//
// The batched engine trains R clients in one grouped model; every parameter
// tensor is an R-stack along dim 0 (or dim 0 = R for BLinear), so client r
// owns the contiguous slice [r*per, (r+1)*per) of each flat tensor.  Two
// kernels over a precomputed chunk table (built once per captured graph —
// pointers are stable under hipGraph replay):
//   1. sqnorm: per-chunk fp32 square-sum -> atomicAdd into normsq[R]
//   2. step:   scale = min(1, max_norm/(sqrt(normsq)+1e-6));
//              g = grad*scale + wd*param; buf = mom*buf + g;
//              param -= lr*buf     (torch SGD semantics, dampening 0)
#include "common.h"

struct Chunk {             // one contiguous piece of one client's slice
    const float* grad;
    float* param;
    float* buf;
    __hip_bfloat16* shadow;  // optional bf16 mirror of param (LM forward
                             // reads it; updating it here removes every
                             // per-step weight-cast launch from the path)
    int len;
    int client;
};

// Per-chunk partial square-sums (no atomics); sqnorm_reduce_kernel sums
// them per client in fixed chunk order so replays are bit-identical.
__global__ void __launch_bounds__(256)
sqnorm_kernel(const Chunk* __restrict__ chunks,
              float* __restrict__ partials) {
    const Chunk ck = chunks[blockIdx.x];
    float s = 0.f;
    for (int i = threadIdx.x; i < ck.len; i += blockDim.x) {
        const float g = ck.grad[i];
        s += g * g;
    }
    __shared__ float scratch[256 / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1) s += __shfl_down(s, off, WAVE);
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.f;
        for (int w = 0; w < blockDim.x / WAVE; ++w) t += scratch[w];
        partials[blockIdx.x] = t;
    }
}

__global__ void __launch_bounds__(64)
sqnorm_reduce_kernel(const float* __restrict__ partials,
                     const int* __restrict__ chunk_client, int n_chunks,
                     float* __restrict__ normsq) {
    const int r = blockIdx.x;
    float s = 0.f;
    // fixed serial order within each lane's stripe, then one wave reduce:
    // the stripe pattern is deterministic for a fixed chunk table
    for (int i = threadIdx.x; i < n_chunks; i += 64)
        if (chunk_client[i] == r) s += partials[i];
    for (int off = WAVE / 2; off > 0; off >>= 1) s += __shfl_down(s, off, WAVE);
    if (threadIdx.x == 0) normsq[r] = s;
}

__global__ void __launch_bounds__(256)
clip_sgd_step_kernel(const Chunk* __restrict__ chunks,
                     const float* __restrict__ normsq, float max_norm,
                     float lr, float momentum, float weight_decay) {
    const Chunk ck = chunks[blockIdx.x];
    const float norm = sqrtf(normsq[ck.client]);
    const float scale = fminf(max_norm / (norm + 1e-6f), 1.f);
    for (int i = threadIdx.x; i < ck.len; i += blockDim.x) {
        const float p = ck.param[i];
        float g = ck.grad[i] * scale + weight_decay * p;
        const float b = momentum * ck.buf[i] + g;
        ck.buf[i] = b;
        const float pn = p - lr * b;
        ck.param[i] = pn;
        if (ck.shadow) ck.shadow[i] = __float2bfloat16(pn);
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// Build the device chunk table from (grads, params, bufs) lists.  Returns
// (table_blob_u8, n_chunks).  Call once per captured graph; tensors must
// stay alive (the Python side keeps them on the graph object).
std::vector<at::Tensor> build_chunk_table(std::vector<at::Tensor> grads,
                                          std::vector<at::Tensor> params,
                                          std::vector<at::Tensor> bufs,
                                          int64_t R, int64_t chunk_elems,
                                          std::vector<at::Tensor> shadows) {
    std::vector<Chunk> host;
    for (size_t t = 0; t < grads.size(); ++t) {
        TORCH_CHECK(grads[t].is_contiguous() && params[t].is_contiguous() &&
                    bufs[t].is_contiguous());
        TORCH_CHECK(grads[t].scalar_type() == at::kFloat,
                    "clip_sgd expects fp32 master grads/params");
        const long numel = grads[t].numel();
        TORCH_CHECK(numel % R == 0);
        const long per = numel / R;
        const float* g = grads[t].data_ptr<float>();
        float* p = params[t].data_ptr<float>();
        float* b = bufs[t].data_ptr<float>();
        __hip_bfloat16* sh = nullptr;
        if (t < shadows.size() && shadows[t].defined() &&
            shadows[t].numel() > 0) {
            TORCH_CHECK(shadows[t].is_contiguous() &&
                        shadows[t].scalar_type() == at::kBFloat16 &&
                        shadows[t].numel() == numel,
                        "shadow must be a contiguous bf16 mirror");
            sh = (__hip_bfloat16*)shadows[t].data_ptr();
        }
        for (long r = 0; r < R; ++r) {
            for (long off = 0; off < per; off += chunk_elems) {
                Chunk ck;
                ck.grad = g + r * per + off;
                ck.param = p + r * per + off;
                ck.buf = b + r * per + off;
                ck.shadow = sh ? sh + r * per + off : nullptr;
                ck.len = (int)std::min((long)chunk_elems, per - off);
                ck.client = (int)r;
                host.push_back(ck);
            }
        }
    }
    auto blob = at::from_blob(host.data(),
                              {(long)(host.size() * sizeof(Chunk))},
                              at::TensorOptions().dtype(at::kByte))
                    .clone()
                    .to(grads[0].device(), /*non_blocking=*/false);
    std::vector<int> clients(host.size());
    for (size_t i = 0; i < host.size(); ++i) clients[i] = host[i].client;
    auto cl = at::from_blob(clients.data(), {(long)clients.size()},
                            at::TensorOptions().dtype(at::kInt))
                  .clone()
                  .to(grads[0].device(), /*non_blocking=*/false);
    auto n = at::scalar_tensor((long)host.size(), at::kLong);
    return {blob, n, cl};
}

// Fill an already-allocated device chunk table from (grads, params, bufs)
// pointer lists.  Used by the hipGraph path: the kernels are RECORDED
// against the blob's device address during capture, and the contents are
// written here once afterwards (grad pool addresses are stable per graph).
void fill_chunk_table(at::Tensor blob, std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> params,
                      std::vector<at::Tensor> bufs, int64_t R,
                      int64_t chunk_elems, std::vector<at::Tensor> shadows) {
    std::vector<Chunk> host;
    for (size_t t = 0; t < grads.size(); ++t) {
        const long numel = grads[t].numel();
        const long per = numel / R;
        const float* g = grads[t].data_ptr<float>();
        float* p = params[t].data_ptr<float>();
        float* b = bufs[t].data_ptr<float>();
        __hip_bfloat16* sh = nullptr;
        if (t < shadows.size() && shadows[t].defined() &&
            shadows[t].numel() > 0)
            sh = (__hip_bfloat16*)shadows[t].data_ptr();
        for (long r = 0; r < R; ++r)
            for (long off = 0; off < per; off += chunk_elems) {
                Chunk ck;
                ck.grad = g + r * per + off;
                ck.param = p + r * per + off;
                ck.buf = b + r * per + off;
                ck.shadow = sh ? sh + r * per + off : nullptr;
                ck.len = (int)std::min((long)chunk_elems, per - off);
                ck.client = (int)r;
                host.push_back(ck);
            }
    }
    TORCH_CHECK((long)(host.size() * sizeof(Chunk)) == blob.numel(),
                "chunk table size changed");
    auto src = at::from_blob(host.data(), {(long)(host.size() * sizeof(Chunk))},
                             at::TensorOptions().dtype(at::kByte)).clone();
    blob.copy_(src, /*non_blocking=*/false);
}

void clip_sgd_step(at::Tensor table_blob, int64_t n_chunks,
                   at::Tensor chunk_client, at::Tensor partials,
                   at::Tensor normsq, double max_norm, double lr,
                   double momentum, double weight_decay) {
    auto stream = at::hip::getCurrentHIPStream();
    const Chunk* chunks = (const Chunk*)table_blob.data_ptr();
    const int R = (int)normsq.numel();
    hipLaunchKernelGGL(sqnorm_kernel, dim3((int)n_chunks), dim3(256), 0,
                       stream, chunks, partials.data_ptr<float>());
    hipLaunchKernelGGL(sqnorm_reduce_kernel, dim3(R), dim3(64), 0, stream,
                       partials.data_ptr<float>(),
                       chunk_client.data_ptr<int>(), (int)n_chunks,
                       normsq.data_ptr<float>());
    hipLaunchKernelGGL(clip_sgd_step_kernel, dim3((int)n_chunks), dim3(256), 0,
                       stream, chunks, normsq.data_ptr<float>(),
                       (float)max_norm, (float)lr, (float)momentum,
                       (float)weight_decay);
}
