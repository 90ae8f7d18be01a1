// Fused Scaler + static-BatchNorm + ReLU, forward and backward (K3+K5+K6 of
// SURVEY.md §2b; reference semantics: src/modules/modules.py:9-11 Scaler,
// src/models/resnet.py:16-17 sBN momentum=None/track=False,
// src/models/resnet.py:44-50 block order scaler->norm->relu).
//
// Train-mode BN normalizes with the live batch stats, so the Scaler's x/rate
// cancels exactly (BN(x/r) == BN(x)); the fused op therefore omits the
// division (the standalone Scaler path is used with norm='none').  One
// workgroup owns one channel (BN) or one (sample, group) (GroupNorm); the
// reduction set N*H*W is at most ~10k elements for HeteroFL's CIFAR shapes,
// so a single 256-thread block two-pass over L2-resident data is the right
// shape — the win vs eager is fusing 3 kernels (+3 backward) into 1 (+1) and
// halving HBM round-trips in the launch-bound tiny-batch regime.
#include "common.h"

// ------------------------------------------------------------------ forward
// y = relu((x - mean_c) * invstd_c * g_c + b_c), per channel c over (N,H,W).
// One workgroup per channel; the channel's N*HW elements are staged in LDS
// during the stats pass (HeteroFL shapes fit: <= 10240 elems), so the
// normalize pass reads LDS instead of a second global pass.
template <typename T, bool STAGE>
__global__ void __launch_bounds__(256)
bn_relu_fwd_kernel(const T* __restrict__ x, const float* __restrict__ gamma,
                   const float* __restrict__ beta, T* __restrict__ y,
                   float* __restrict__ mean_out, float* __restrict__ invstd_out,
                   int N, int C, int HW, float eps) {
    const int c = blockIdx.x;
    __shared__ float scratch[2 * 256 / WAVE];
    extern __shared__ __attribute__((aligned(16))) char smem[];
    T* stage = (T*)smem;
    float s1 = 0.f, s2 = 0.f;
    const long chan_off = (long)c * HW;
    const long samp_stride = (long)C * HW;
    for (int i = threadIdx.x; i < N * HW; i += blockDim.x) {
        const int n = i / HW, hw = i - n * HW;
        const T raw = x[n * samp_stride + chan_off + hw];
        if (STAGE) stage[i] = raw;
        const float v = (float)raw;
        s1 += v;
        s2 += v * v;
    }
    block_reduce2(s1, s2, scratch);
    const float inv_m = 1.f / (N * HW);
    const float mean = s1 * inv_m;
    const float var = fmaxf(s2 * inv_m - mean * mean, 0.f);
    const float invstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
        mean_out[c] = mean;
        invstd_out[c] = invstd;
    }
    const float g = gamma ? gamma[c] : 1.f;
    const float b = beta ? beta[c] : 0.f;
    const float scale = invstd * g;
    const float shift = b - mean * scale;
    for (int i = threadIdx.x; i < N * HW; i += blockDim.x) {
        const int n = i / HW, hw = i - n * HW;
        const long off = n * samp_stride + chan_off + hw;
        const float v = STAGE ? (float)stage[i] : ld_f32(x + off);
        st_f32(y + off, fmaxf(v * scale + shift, 0.f));
    }
}

// --------------------------------------------- forward, large-batch (3-kernel)
// The one-workgroup-per-channel kernel serializes when N*HW is large (the
// sBN statistics / eval pass runs N=2500): with C blocks only, 3.3 ms per
// call was the stats pass's real bound (kstats_r02).  Large batches use a
// grid-parallel 3-kernel path instead: per-(channel, sample-chunk) partial
// sums -> per-channel finalize -> grid-parallel normalize+ReLU.
template <typename T>
__global__ void __launch_bounds__(256)
bn_sums_kernel(const T* __restrict__ x, float* __restrict__ partials, int N,
               int C, int HW, int S) {
    const int c = blockIdx.x;
    const int s = blockIdx.y;
    const int chunk = (N + S - 1) / S;
    const int n1 = min(N, (s + 1) * chunk);
    const long chan_off = (long)c * HW;
    const long samp_stride = (long)C * HW;
    float s1 = 0.f, s2 = 0.f;
    for (int n = s * chunk; n < n1; ++n) {
        const T* base = x + n * samp_stride + chan_off;
        for (int hw = threadIdx.x; hw < HW; hw += blockDim.x) {
            const float v = ld_f32(base + hw);
            s1 += v;
            s2 += v * v;
        }
    }
    __shared__ float scratch[2 * 256 / WAVE];
    block_reduce2(s1, s2, scratch);
    if (threadIdx.x == 0) {
        partials[((long)c * S + s) * 2] = s1;
        partials[((long)c * S + s) * 2 + 1] = s2;
    }
}

__global__ void __launch_bounds__(256)
bn_finalize_kernel(const float* __restrict__ partials,
                   const float* __restrict__ gamma,
                   const float* __restrict__ beta,
                   float* __restrict__ mean_out,
                   float* __restrict__ invstd_out,
                   float* __restrict__ scale_shift, int C, int S,
                   float inv_m, float eps) {
    const int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float s1 = 0.f, s2 = 0.f;
    for (int s = 0; s < S; ++s) {
        s1 += partials[((long)c * S + s) * 2];
        s2 += partials[((long)c * S + s) * 2 + 1];
    }
    const float mean = s1 * inv_m;
    const float var = fmaxf(s2 * inv_m - mean * mean, 0.f);
    const float invstd = rsqrtf(var + eps);
    mean_out[c] = mean;
    invstd_out[c] = invstd;
    const float g = gamma ? gamma[c] : 1.f;
    const float b = beta ? beta[c] : 0.f;
    const float scale = invstd * g;
    scale_shift[2 * c] = scale;
    scale_shift[2 * c + 1] = b - mean * scale;
}

template <typename T>
__global__ void __launch_bounds__(256)
bn_apply_relu_kernel(const T* __restrict__ x,
                     const float* __restrict__ scale_shift,
                     T* __restrict__ y, int N, int C, int HW, int S) {
    const int c = blockIdx.x;
    const int s = blockIdx.y;
    const int chunk = (N + S - 1) / S;
    const int n1 = min(N, (s + 1) * chunk);
    const long chan_off = (long)c * HW;
    const long samp_stride = (long)C * HW;
    const float scale = scale_shift[2 * c];
    const float shift = scale_shift[2 * c + 1];
    for (int n = s * chunk; n < n1; ++n) {
        const long off = n * samp_stride + chan_off;
        for (int hw = threadIdx.x; hw < HW; hw += blockDim.x)
            st_f32(y + off + hw, fmaxf(ld_f32(x + off + hw) * scale + shift,
                                       0.f));
    }
}

// ----------------------------------------------------------------- backward
// relu mask from pre = xhat*g+b; dx = invstd*g*(dym - s1/M - xhat*s2/M).
// STAGE: the channel's x and (relu-masked) dy are kept in LDS from the
// reduction pass so the dx pass reads LDS, not global.
template <typename T, bool STAGE>
__global__ void __launch_bounds__(256)
bn_relu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                   const float* __restrict__ gamma, const float* __restrict__ beta,
                   const float* __restrict__ mean, const float* __restrict__ invstd,
                   T* __restrict__ dx, float* __restrict__ dgamma,
                   float* __restrict__ dbeta, int N, int C, int HW) {
    const int c = blockIdx.x;
    __shared__ float scratch[2 * 256 / WAVE];
    extern __shared__ __attribute__((aligned(16))) char smem[];
    T* sx = (T*)smem;
    T* sd = sx + (STAGE ? N * HW : 0);
    const long chan_off = (long)c * HW;
    const long samp_stride = (long)C * HW;
    const float m = mean[c], is = invstd[c];
    const float g = gamma ? gamma[c] : 1.f;
    const float b = beta ? beta[c] : 0.f;
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < N * HW; i += blockDim.x) {
        const int n = i / HW, hw = i - n * HW;
        const long off = n * samp_stride + chan_off + hw;
        const T xr = x[off];
        const float xh = ((float)xr - m) * is;
        const float pre = xh * g + b;
        const float d = pre > 0.f ? ld_f32(dy + off) : 0.f;
        if (STAGE) {
            sx[i] = xr;
            sd[i] = (T)d;
        }
        s1 += d;
        s2 += d * xh;
    }
    block_reduce2(s1, s2, scratch);
    if (threadIdx.x == 0) {
        dbeta[c] = s1;
        dgamma[c] = s2;
    }
    const float inv_M = 1.f / (N * HW);
    const float k1 = s1 * inv_M, k2 = s2 * inv_M;
    const float gis = g * is;
    for (int i = threadIdx.x; i < N * HW; i += blockDim.x) {
        const int n = i / HW, hw = i - n * HW;
        const long off = n * samp_stride + chan_off + hw;
        float xh, d;
        if (STAGE) {
            xh = ((float)sx[i] - m) * is;
            d = (float)sd[i];
        } else {
            xh = (ld_f32(x + off) - m) * is;
            const float pre = xh * g + b;
            d = pre > 0.f ? ld_f32(dy + off) : 0.f;
        }
        st_f32(dx + off, gis * (d - k1 - xh * k2));
    }
}

// ---------------------------------------------------------------- GroupNorm
// Stats per (sample n, group j) over the group's channels x HW
// (covers gn=4 / ln=1 / in=C groups, reference: src/models/resnet.py:19-26).
template <typename T, bool STAGE>
__global__ void __launch_bounds__(256)
gn_relu_fwd_kernel(const T* __restrict__ x, const float* __restrict__ gamma,
                   const float* __restrict__ beta, T* __restrict__ y,
                   float* __restrict__ mean_out, float* __restrict__ invstd_out,
                   int N, int C, int HW, int G, float eps) {
    const int ng = blockIdx.x;           // n*G + j
    const int n = ng / G, j = ng - n * G;
    const int cpg = C / G;
    __shared__ float scratch[2 * 256 / WAVE];
    extern __shared__ __attribute__((aligned(16))) char smem[];
    T* stage = (T*)smem;
    const long base = (long)n * C * HW + (long)j * cpg * HW;
    const int M = cpg * HW;
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < M; i += blockDim.x) {
        const T raw = x[base + i];
        if (STAGE) stage[i] = raw;
        const float v = (float)raw;
        s1 += v;
        s2 += v * v;
    }
    block_reduce2(s1, s2, scratch);
    const float inv_m = 1.f / M;
    const float mean = s1 * inv_m;
    const float var = fmaxf(s2 * inv_m - mean * mean, 0.f);
    const float invstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
        mean_out[ng] = mean;
        invstd_out[ng] = invstd;
    }
    for (int i = threadIdx.x; i < M; i += blockDim.x) {
        const int c = j * cpg + i / HW;
        const float g = gamma ? gamma[c] : 1.f;
        const float b = beta ? beta[c] : 0.f;
        const float v = STAGE ? (float)stage[i] : ld_f32(x + base + i);
        st_f32(y + base + i, fmaxf((v - mean) * invstd * g + b, 0.f));
    }
}

// dgamma/dbeta go to per-(n,g) partials (pgamma/pbeta, shape (N, C)),
// reduced over n in fixed order by gn_grad_reduce_kernel — deterministic
// under hipGraph replay (no atomics).
template <typename T>
__global__ void __launch_bounds__(256)
gn_relu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                   const float* __restrict__ gamma, const float* __restrict__ beta,
                   const float* __restrict__ mean, const float* __restrict__ invstd,
                   T* __restrict__ dx, float* __restrict__ pgamma,
                   float* __restrict__ pbeta, int N, int C, int HW, int G) {
    const int ng = blockIdx.x;
    const int n = ng / G, j = ng - n * G;
    const int cpg = C / G;
    __shared__ float scratch[2 * 256 / WAVE];
    const long base = (long)n * C * HW + (long)j * cpg * HW;
    const int M = cpg * HW;
    const float m = mean[ng], is = invstd[ng];
    // per-channel partials: one thread per channel, serial over HW
    if (pgamma) {
        for (int cc = threadIdx.x; cc < cpg; cc += blockDim.x) {
            const int c = j * cpg + cc;
            const float g = gamma ? gamma[c] : 1.f;
            const float b = beta ? beta[c] : 0.f;
            float sg = 0.f, sb = 0.f;
            for (int hw = 0; hw < HW; ++hw) {
                const long off = base + (long)cc * HW + hw;
                const float xh = (ld_f32(x + off) - m) * is;
                const float pre = xh * g + b;
                const float d = pre > 0.f ? ld_f32(dy + off) : 0.f;
                sg += d * xh;
                sb += d;
            }
            pgamma[(long)n * C + c] = sg;
            pbeta[(long)n * C + c] = sb;
        }
    }
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < M; i += blockDim.x) {
        const int c = j * cpg + i / HW;
        const float g = gamma ? gamma[c] : 1.f;
        const float b = beta ? beta[c] : 0.f;
        const float xh = (ld_f32(x + base + i) - m) * is;
        const float pre = xh * g + b;
        const float d = pre > 0.f ? ld_f32(dy + base + i) : 0.f;
        s1 += d * g;
        s2 += d * g * xh;
    }
    block_reduce2(s1, s2, scratch);
    const float inv_M = 1.f / M;
    const float k1 = s1 * inv_M, k2 = s2 * inv_M;
    for (int i = threadIdx.x; i < M; i += blockDim.x) {
        const int c = j * cpg + i / HW;
        const float g = gamma ? gamma[c] : 1.f;
        const float b = beta ? beta[c] : 0.f;
        const float xh = (ld_f32(x + base + i) - m) * is;
        const float pre = xh * g + b;
        const float d = pre > 0.f ? ld_f32(dy + base + i) : 0.f;
        st_f32(dx + base + i, is * (d * g - k1 - xh * k2));
    }
}

// ------------------------------------------------------------ host wrappers
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_FT(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> bn_relu_fwd(at::Tensor x, at::Tensor gamma,
                                    at::Tensor beta, double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const int N = x.size(0), C = x.size(1);
    const int HW = x.numel() / (N * C);
    auto y = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto mean = at::empty({C}, opts);
    auto invstd = at::empty({C}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    const long stage_bytes = (long)N * HW * x.element_size();
    const long M = (long)N * HW;
    if (M > 64 * 1024) {
        // large batch (stats/eval pass): grid-parallel 3-kernel path
        const int S = std::max(1, std::min(N, (4096 + C - 1) / C));
        auto partials = at::empty({(long)C * S * 2}, opts);
        auto scale_shift = at::empty({2L * C}, opts);
        DISPATCH_FT(x.scalar_type(), {
            hipLaunchKernelGGL(bn_sums_kernel<scalar_t>, dim3(C, S),
                               dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               partials.data_ptr<float>(), N, C, HW, S);
            hipLaunchKernelGGL(bn_finalize_kernel,
                               dim3((C + 255) / 256), dim3(256), 0, stream,
                               partials.data_ptr<float>(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               mean.data_ptr<float>(),
                               invstd.data_ptr<float>(),
                               scale_shift.data_ptr<float>(), C, S,
                               (float)(1.0 / M), (float)eps);
            hipLaunchKernelGGL(bn_apply_relu_kernel<scalar_t>, dim3(C, S),
                               dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               scale_shift.data_ptr<float>(),
                               (scalar_t*)y.data_ptr(), N, C, HW, S);
        });
        return {y, mean, invstd};
    }
    DISPATCH_FT(x.scalar_type(), {
        if (stage_bytes <= 64 * 1024)
            hipLaunchKernelGGL((bn_relu_fwd_kernel<scalar_t, true>), dim3(C),
                               dim3(256), (int)stage_bytes, stream,
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               (scalar_t*)y.data_ptr(),
                               mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), N, C, HW,
                               (float)eps);
        else
            hipLaunchKernelGGL((bn_relu_fwd_kernel<scalar_t, false>), dim3(C),
                               dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               (scalar_t*)y.data_ptr(),
                               mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), N, C, HW,
                               (float)eps);
    });
    return {y, mean, invstd};
}

std::vector<at::Tensor> bn_relu_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor gamma, at::Tensor beta,
                                    at::Tensor mean, at::Tensor invstd) {
    const int N = x.size(0), C = x.size(1);
    const int HW = x.numel() / (N * C);
    auto dx = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto dgamma = at::empty({C}, opts);
    auto dbeta = at::empty({C}, opts);
    auto dyc = dy.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    const long stage_bytes = 2L * N * (x.numel() / (N * C)) * x.element_size();
    DISPATCH_FT(x.scalar_type(), {
        if (stage_bytes <= 64 * 1024)
            hipLaunchKernelGGL((bn_relu_bwd_kernel<scalar_t, true>), dim3(C),
                               dim3(256), (int)stage_bytes, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               mean.data_ptr<float>(),
                               invstd.data_ptr<float>(),
                               (scalar_t*)dx.data_ptr(),
                               dgamma.data_ptr<float>(),
                               dbeta.data_ptr<float>(), N, C, HW);
        else
            hipLaunchKernelGGL((bn_relu_bwd_kernel<scalar_t, false>), dim3(C),
                               dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               mean.data_ptr<float>(),
                               invstd.data_ptr<float>(),
                               (scalar_t*)dx.data_ptr(),
                               dgamma.data_ptr<float>(),
                               dbeta.data_ptr<float>(), N, C, HW);
    });
    return {dx, dgamma, dbeta};
}

std::vector<at::Tensor> gn_relu_fwd(at::Tensor x, at::Tensor gamma,
                                    at::Tensor beta, int64_t G, double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const int N = x.size(0), C = x.size(1);
    const int HW = x.numel() / (N * C);
    TORCH_CHECK(C % G == 0, "channels not divisible by groups");
    auto y = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto mean = at::empty({N * G}, opts);
    auto invstd = at::empty({N * G}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    const long gsb = ((long)C / G) * (x.numel() / (N * C)) * x.element_size();
    DISPATCH_FT(x.scalar_type(), {
        if (gsb <= 64 * 1024)
            hipLaunchKernelGGL((gn_relu_fwd_kernel<scalar_t, true>),
                               dim3(N * G), dim3(256), (int)gsb, stream,
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), N, C, HW, (int)G,
                               (float)eps);
        else
            hipLaunchKernelGGL((gn_relu_fwd_kernel<scalar_t, false>),
                               dim3(N * G), dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               gamma.defined() ? gamma.data_ptr<float>()
                                               : nullptr,
                               beta.defined() ? beta.data_ptr<float>()
                                              : nullptr,
                               (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), N, C, HW, (int)G,
                               (float)eps);
    });
    return {y, mean, invstd};
}

__global__ void __launch_bounds__(256)
gn_grad_reduce_kernel(const float* __restrict__ pgamma,
                      const float* __restrict__ pbeta,
                      float* __restrict__ dgamma, float* __restrict__ dbeta,
                      int N, int C) {
    const int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float sg = 0.f, sb = 0.f;
    for (int n = 0; n < N; ++n) {
        sg += pgamma[(long)n * C + c];
        sb += pbeta[(long)n * C + c];
    }
    dgamma[c] = sg;
    dbeta[c] = sb;
}

std::vector<at::Tensor> gn_relu_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor gamma, at::Tensor beta,
                                    at::Tensor mean, at::Tensor invstd,
                                    int64_t G) {
    const int N = x.size(0), C = x.size(1);
    const int HW = x.numel() / (N * C);
    auto dx = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto pgamma = at::empty({N, C}, opts);
    auto pbeta = at::empty({N, C}, opts);
    auto dgamma = at::empty({C}, opts);
    auto dbeta = at::empty({C}, opts);
    auto dyc = dy.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_FT(x.scalar_type(), {
        hipLaunchKernelGGL(gn_relu_bwd_kernel<scalar_t>, dim3(N * G), dim3(256),
                           0, stream,
                           (const scalar_t*)dyc.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           gamma.defined() ? gamma.data_ptr<float>() : nullptr,
                           beta.defined() ? beta.data_ptr<float>() : nullptr,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           (scalar_t*)dx.data_ptr(), pgamma.data_ptr<float>(),
                           pbeta.data_ptr<float>(), N, C, HW, (int)G);
    });
    hipLaunchKernelGGL(gn_grad_reduce_kernel, dim3((C + 255) / 256),
                       dim3(256), 0, stream, pgamma.data_ptr<float>(),
                       pbeta.data_ptr<float>(), dgamma.data_ptr<float>(),
                       dbeta.data_ptr<float>(), N, C);
    return {dx, dgamma, dbeta};
}
