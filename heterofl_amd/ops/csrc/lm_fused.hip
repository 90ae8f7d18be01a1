// Fused LM glue kernels (the LM step carried ~300 small
// torch elementwise launches — scaler divisions, GELU, dropout, residual
// adds).  Two elementwise fusions cover the encoder layer's non-GEMM work
// (reference semantics: src/models/transformer.py:112-119):
//   gelu_drop:  y = dropout(gelu(x / rate), p)          [Scaler+GELU+drop]
//   res_drop:   t = src + dropout(h / rate, p)          [Scaler+drop+residual]
// The dropout mask is drawn by a counter-hash RNG keyed on (device u64
// seed cell, per-call-site salt, element index).  The seed cell advances
// ONCE per training step (rng_bump, launched by the step driver) so
// hipGraph replays draw fresh masks; the per-site salt decorrelates the
// many dropout sites within a step without per-call bump launches.
// Masks are SAVED as uint8 for the exact one-kernel backward.
#include "common.h"

__device__ __forceinline__ uint32_t mix32(uint32_t h) {
    h ^= h >> 16;
    h *= 0x85EBCA6Bu;
    h ^= h >> 13;
    h *= 0xC2B2AE35u;
    h ^= h >> 16;
    return h;
}

__device__ __forceinline__ uint32_t rng_at(unsigned long long seed,
                                           uint32_t salt, long i) {
    uint32_t h = (uint32_t)(seed & 0xFFFFFFFFull) * 0x9E3779B1u
                 + (uint32_t)(seed >> 32) * 0x85EBCA77u
                 + salt * 0x27D4EB2Fu
                 + (uint32_t)i * 0xC2B2AE3Du + (uint32_t)(i >> 32);
    return mix32(h);
}

__global__ void rng_bump_kernel(unsigned long long* seed) {
    *seed = *seed * 6364136223846793005ull + 1442695040888963407ull;
}

// GELU (erf form, torch default) and its derivative
__device__ __forceinline__ float gelu_f(float v) {
    return 0.5f * v * (1.f + erff(v * 0.70710678f));
}
__device__ __forceinline__ float dgelu_f(float v) {
    return 0.5f * (1.f + erff(v * 0.70710678f))
           + v * 0.39894228f * __expf(-0.5f * v * v);
}

template <typename T>
__global__ void __launch_bounds__(256)
gelu_drop_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                     unsigned char* __restrict__ mask,
                     const unsigned long long* __restrict__ seed,
                     unsigned salt, long n,
                     float inv_rate, float p, float inv_keep) {
    const unsigned long long sd = *seed;
    const uint32_t thresh = (uint32_t)(p * 4294967296.0);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float g = gelu_f(ld_f32(x + i) * inv_rate);
        unsigned char keep = 1;
        if (thresh) keep = rng_at(sd, salt, i) >= thresh;
        if (mask) mask[i] = keep;
        st_f32(y + i, keep ? g * inv_keep : 0.f);
    }
}

template <typename T>
__global__ void __launch_bounds__(256)
gelu_drop_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                     const unsigned char* __restrict__ mask,
                     T* __restrict__ dx, long n, float inv_rate,
                     float inv_keep) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float keep = mask ? (float)mask[i] : 1.f;
        const float v = ld_f32(x + i) * inv_rate;
        st_f32(dx + i,
               ld_f32(dy + i) * keep * inv_keep * dgelu_f(v) * inv_rate);
    }
}

template <typename T>
__global__ void __launch_bounds__(256)
res_drop_fwd_kernel(const T* __restrict__ src, const T* __restrict__ h,
                    T* __restrict__ t, unsigned char* __restrict__ mask,
                    const unsigned long long* __restrict__ seed,
                    unsigned salt, long n,
                    float inv_rate, float p, float inv_keep) {
    const unsigned long long sd = *seed;
    const uint32_t thresh = (uint32_t)(p * 4294967296.0);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        unsigned char keep = 1;
        if (thresh) keep = rng_at(sd, salt, i) >= thresh;
        if (mask) mask[i] = keep;
        const float hv = keep ? ld_f32(h + i) * inv_rate * inv_keep : 0.f;
        st_f32(t + i, ld_f32(src + i) + hv);
    }
}

template <typename T>
__global__ void __launch_bounds__(256)
drop_scale_bwd_kernel(const T* __restrict__ dt,
                      const unsigned char* __restrict__ mask,
                      T* __restrict__ dh, long n, float inv_rate,
                      float inv_keep) {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float keep = mask ? (float)mask[i] : 1.f;
        st_f32(dh + i, ld_f32(dt + i) * keep * inv_keep * inv_rate);
    }
}

// ------------------------------------------------------------ host layer
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_LMF(t, ...)                                                  \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

static dim3 lmf_grid(long n) {
    long blocks = (n + 256 * 8 - 1) / (256 * 8);
    if (blocks < 1) blocks = 1;
    if (blocks > 4096) blocks = 4096;
    return dim3((unsigned)blocks);
}

void rng_bump(at::Tensor seed) {
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(rng_bump_kernel, dim3(1), dim3(1), 0, stream,
                       (unsigned long long*)seed.data_ptr());
}

std::vector<at::Tensor> gelu_drop_fwd(at::Tensor x, at::Tensor seed,
                                      int64_t salt, double rate, double p) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const long n = x.numel();
    auto y = at::empty_like(x);
    const bool dropping = p > 0.0;
    auto mask = at::empty({dropping ? n : 0}, x.options().dtype(at::kByte));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_LMF(x.scalar_type(), {
        hipLaunchKernelGGL(gelu_drop_fwd_kernel<scalar_t>, lmf_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)x.data_ptr(),
                           (scalar_t*)y.data_ptr(),
                           dropping ? mask.data_ptr<unsigned char>() : nullptr,
                           (const unsigned long long*)seed.data_ptr(),
                           (unsigned)salt, n,
                           (float)(1.0 / rate), (float)p,
                           (float)(1.0 / (1.0 - p)));
    });
    return {y, mask};
}

at::Tensor gelu_drop_bwd(at::Tensor dy, at::Tensor x, at::Tensor mask,
                         double rate, double p) {
    const long n = x.numel();
    auto dyc = dy.contiguous();
    auto dx = at::empty_like(x);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_LMF(x.scalar_type(), {
        hipLaunchKernelGGL(gelu_drop_bwd_kernel<scalar_t>, lmf_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)dyc.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           mask.numel() ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                           (scalar_t*)dx.data_ptr(), n, (float)(1.0 / rate),
                           (float)(1.0 / (1.0 - p)));
    });
    return dx;
}

std::vector<at::Tensor> res_drop_fwd(at::Tensor src, at::Tensor h,
                                     at::Tensor seed, int64_t salt,
                                     double rate, double p) {
    TORCH_CHECK(src.is_cuda() && src.is_contiguous() && h.is_contiguous());
    TORCH_CHECK(src.numel() == h.numel());
    const long n = src.numel();
    auto t = at::empty_like(src);
    const bool dropping = p > 0.0;
    auto mask = at::empty({dropping ? n : 0},
                          src.options().dtype(at::kByte));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_LMF(src.scalar_type(), {
        hipLaunchKernelGGL(res_drop_fwd_kernel<scalar_t>, lmf_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)src.data_ptr(),
                           (const scalar_t*)h.data_ptr(),
                           (scalar_t*)t.data_ptr(),
                           dropping ? mask.data_ptr<unsigned char>() : nullptr,
                           (const unsigned long long*)seed.data_ptr(),
                           (unsigned)salt, n,
                           (float)(1.0 / rate), (float)p,
                           (float)(1.0 / (1.0 - p)));
    });
    return {t, mask};
}

at::Tensor drop_scale_bwd(at::Tensor dt, at::Tensor mask, double rate,
                          double p) {
    const long n = dt.numel();
    auto dtc = dt.contiguous();
    auto dh = at::empty_like(dtc);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_LMF(dt.scalar_type(), {
        hipLaunchKernelGGL(drop_scale_bwd_kernel<scalar_t>, lmf_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)dtc.data_ptr(),
                           mask.numel() ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                           (scalar_t*)dh.data_ptr(), n, (float)(1.0 / rate),
                           (float)(1.0 / (1.0 - p)));
    });
    return dh;
}

// ------------------------- K11: in-kernel Bernoulli token masking --------
// reference: src/models/transformer.py:149-151 — Bernoulli(mask_rate) mask
// -> fill with the <mask> token id.  One kernel instead of bernoulli +
// masked_fill (+ fill); same (seed, salt, index) RNG as the dropouts, so
// hipGraph replays draw fresh masks.
__global__ void __launch_bounds__(256)
token_mask_kernel(const long* __restrict__ in, long* __restrict__ out,
                  const unsigned long long* __restrict__ seed, unsigned salt,
                  long n, float rate, long mask_id) {
    const unsigned long long sd = *seed;
    const uint32_t thresh = (uint32_t)(rate * 4294967296.0);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x)
        out[i] = rng_at(sd, salt, i) < thresh ? mask_id : in[i];
}

at::Tensor token_mask(at::Tensor tokens, at::Tensor seed, int64_t salt,
                      double rate, int64_t mask_id) {
    TORCH_CHECK(tokens.is_cuda() && tokens.is_contiguous() &&
                tokens.scalar_type() == at::kLong);
    const long n = tokens.numel();
    auto out = at::empty_like(tokens);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(token_mask_kernel, lmf_grid(n), dim3(256), 0, stream,
                       tokens.data_ptr<long>(), out.data_ptr<long>(),
                       (const unsigned long long*)seed.data_ptr(),
                       (unsigned)salt, n, (float)rate, (long)mask_id);
    return out;
}

// ----------------- K9: fused embedding + positional + Scaler gather ------
// reference: src/models/transformer.py:29-37 — scaler(tok_emb(src)) +
// scaler(pos_emb(pos)); one gather-add-scale kernel forward, and a
// DETERMINISTIC backward (per-(r,vocab-row) reduction in fixed (b,s) scan
// order — no atomics, so hipGraph replays stay bit-identical) that writes
// fp32 grads straight for the masters (no shadow-upcast kernels).
template <typename TT>
__global__ void __launch_bounds__(256)
embed_pos_fwd_kernel(const long* __restrict__ ids,
                     const TT* __restrict__ table,
                     const TT* __restrict__ pos, TT* __restrict__ out,
                     int R, int B, int S, int E, int V, int P,
                     float inv_rate) {
    const long rows = (long)R * B * S;
    for (long i = (long)blockIdx.x * (blockDim.x / WAVE)
                  + threadIdx.x / WAVE;
         i < rows; i += (long)gridDim.x * (blockDim.x / WAVE)) {
        const int l = threadIdx.x & (WAVE - 1);
        const int r = (int)(i / ((long)B * S));
        const int s = (int)(i % S);
        const long id = ids[i];
        const TT* src = table + ((long)r * V + id) * E;
        const TT* pp = pos + ((long)r * P + s) * E;
        TT* o = out + i * E;
        for (int e = l; e < E; e += WAVE)
            o[e] = (TT)((ld_f32(src + e) + ld_f32(pp + e)) * inv_rate);
    }
}

// dtable[r, v, :] = inv_rate * sum over (b,s) with ids==v of dy rows.
// One wave owns one (r, v) pair; scans the B*S ids in fixed order.
template <typename TT>
__global__ void __launch_bounds__(256)
embed_bwd_table_kernel(const long* __restrict__ ids,
                       const TT* __restrict__ dy,
                       float* __restrict__ dtable, int R, int BS, int E,
                       int V, float inv_rate) {
    const long pairs = (long)R * V;
    const int l = threadIdx.x & (WAVE - 1);
    for (long p = (long)blockIdx.x * (blockDim.x / WAVE)
                  + threadIdx.x / WAVE;
         p < pairs; p += (long)gridDim.x * (blockDim.x / WAVE)) {
        const int r = (int)(p / V);
        const long v = p % V;
        float acc[16];
        const int ne = (E + WAVE - 1) / WAVE;
#pragma unroll
        for (int j = 0; j < 16; ++j) acc[j] = 0.f;
        bool any = false;
        const long* idr = ids + (long)r * BS;
        for (int t = 0; t < BS; ++t) {
            if (idr[t] == v) {
                any = true;
                const TT* dyr = dy + ((long)r * BS + t) * E;
                for (int j = 0; j < ne; ++j) {
                    const int e = l + j * WAVE;
                    if (e < E) acc[j] += ld_f32(dyr + e);
                }
            }
        }
        if (any)
            for (int j = 0; j < ne; ++j) {
                const int e = l + j * WAVE;
                if (e < E)
                    dtable[((long)r * V + v) * E + e] = acc[j] * inv_rate;
            }
    }
}

// dpos[r, s, :] = inv_rate * sum over b of dy[r, b, s, :]
template <typename TT>
__global__ void __launch_bounds__(256)
embed_bwd_pos_kernel(const TT* __restrict__ dy, float* __restrict__ dpos,
                     int R, int B, int S, int E, int P, float inv_rate) {
    const long rows = (long)R * S;
    const int l = threadIdx.x & (WAVE - 1);
    for (long p = (long)blockIdx.x * (blockDim.x / WAVE)
                  + threadIdx.x / WAVE;
         p < rows; p += (long)gridDim.x * (blockDim.x / WAVE)) {
        const int r = (int)(p / S);
        const int s = (int)(p % S);
        for (int e = l; e < E; e += WAVE) {
            float acc = 0.f;
            for (int b = 0; b < B; ++b)
                acc += ld_f32(dy + (((long)r * B + b) * S + s) * E + e);
            dpos[((long)r * P + s) * E + e] = acc * inv_rate;
        }
    }
}

#define DISPATCH_EMB(t, ...)                                                  \
    if ((t) == at::kFloat) { using emb_t = float; __VA_ARGS__; }              \
    else if ((t) == at::kBFloat16) { using emb_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

at::Tensor embed_pos_fwd(at::Tensor ids, at::Tensor table, at::Tensor pos,
                         double rate) {
    TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() &&
                table.is_contiguous() && pos.is_contiguous());
    const int R = table.size(0), V = table.size(1), E = table.size(2);
    const int P = pos.size(1);
    const int B = ids.size(1), S = ids.size(2);
    TORCH_CHECK(E <= 16 * WAVE, "embedding dim too large for bwd acc");
    auto out = at::empty({(long)R, B, S, E}, table.options());
    auto stream = at::hip::getCurrentHIPStream();
    const long rows = (long)R * B * S;
    long blocks = (rows + 3) / 4;
    if (blocks > 4096) blocks = 4096;
    DISPATCH_EMB(table.scalar_type(), {
        hipLaunchKernelGGL(embed_pos_fwd_kernel<emb_t>,
                           dim3((unsigned)blocks), dim3(256), 0, stream,
                           ids.data_ptr<long>(),
                           (const emb_t*)table.data_ptr(),
                           (const emb_t*)pos.data_ptr(),
                           (emb_t*)out.data_ptr(), R, B, S, E, V, P,
                           (float)(1.0 / rate));
    });
    return out;
}

std::vector<at::Tensor> embed_pos_bwd(at::Tensor dy, at::Tensor ids,
                                      int64_t V, int64_t P, double rate) {
    const int R = ids.size(0), B = ids.size(1), S = ids.size(2);
    const int E = dy.size(3);
    auto dyc = dy.contiguous();
    auto opts = dy.options().dtype(at::kFloat);
    auto dtable = at::zeros({(long)R, V, (long)E}, opts);
    auto dpos = at::zeros({(long)R, P, (long)E}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    long pb = ((long)R * V + 3) / 4;
    if (pb > 4096) pb = 4096;
    long sb = ((long)R * S + 3) / 4;
    if (sb > 4096) sb = 4096;
    DISPATCH_EMB(dy.scalar_type(), {
        hipLaunchKernelGGL(embed_bwd_table_kernel<emb_t>,
                           dim3((unsigned)pb), dim3(256), 0, stream,
                           ids.data_ptr<long>(),
                           (const emb_t*)dyc.data_ptr(),
                           dtable.data_ptr<float>(), R, B * S, E, (int)V,
                           (float)(1.0 / rate));
        hipLaunchKernelGGL(embed_bwd_pos_kernel<emb_t>,
                           dim3((unsigned)sb), dim3(256), 0, stream,
                           (const emb_t*)dyc.data_ptr(),
                           dpos.data_ptr<float>(), R, B, S, E, (int)P,
                           (float)(1.0 / rate));
    });
    return {dtable, dpos};
}
