// Fused per-client LayerNorm for the batched transformer (K9 of SURVEY.md
// §2b; reference: src/models/transformer.py:37,97-102 — LN over the
// embedding dim).  x (R, B, S, E) with per-client affine (R, E); one wave
// per row, 4 rows per 256-thread block.  Replaces the ~12 torch elementwise/
// reduce launches per LN call with 1 forward and 2 backward kernels
// (per-block dgamma/dbeta partials reduced in fixed order — deterministic
// under hipGraph replay).
#include "common.h"

// rows = R*B*S, each of length E (E <= 1024); 4 waves per block
template <typename T>
__global__ void __launch_bounds__(256)
ln_fwd_kernel(const T* __restrict__ x, const float* __restrict__ gamma,
              const float* __restrict__ beta, T* __restrict__ y,
              float* __restrict__ mean_out, float* __restrict__ invstd_out,
              int rows, int rows_per_client, int E, float eps) {
    const int row = blockIdx.x * 4 + (threadIdx.x / WAVE);
    if (row >= rows) return;
    const int l = threadIdx.x & (WAVE - 1);
    const int r = row / rows_per_client;
    const T* xr = x + (long)row * E;
    float s1 = 0.f, s2 = 0.f;
    for (int e = l; e < E; e += WAVE) {
        const float v = ld_f32(xr + e);
        s1 += v;
        s2 += v * v;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        s1 += __shfl_down(s1, off, WAVE);
        s2 += __shfl_down(s2, off, WAVE);
    }
    s1 = __shfl(s1, 0, WAVE);
    s2 = __shfl(s2, 0, WAVE);
    const float inv_e = 1.f / E;
    const float mu = s1 * inv_e;
    const float var = fmaxf(s2 * inv_e - mu * mu, 0.f);
    const float is = rsqrtf(var + eps);
    if (l == 0) {
        mean_out[row] = mu;
        invstd_out[row] = is;
    }
    T* yr = y + (long)row * E;
    const float* g = gamma + (long)r * E;
    const float* b = beta + (long)r * E;
    for (int e = l; e < E; e += WAVE)
        st_f32(yr + e, (ld_f32(xr + e) - mu) * is * g[e] + b[e]);
}

// dx pass; per-row (dgamma, dbeta) contributions go to a (rows, 2, E)
// partials buffer reduced per client in fixed row order (deterministic).
template <typename T>
__global__ void __launch_bounds__(256)
ln_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
              const float* __restrict__ gamma, const float* __restrict__ mean,
              const float* __restrict__ invstd, T* __restrict__ dx,
              float* __restrict__ partials, int rows, int rows_per_client,
              int E) {
    const int wave = threadIdx.x / WAVE;
    const int row = blockIdx.x * 4 + wave;
    const int l = threadIdx.x & (WAVE - 1);
    if (row < rows) {
        const int r = row / rows_per_client;
        const T* xr = x + (long)row * E;
        const T* dyr = dy + (long)row * E;
        const float mu = mean[row], is = invstd[row];
        const float* g = gamma + (long)r * E;
        float s1 = 0.f, s2 = 0.f;
        for (int e = l; e < E; e += WAVE) {
            const float xh = (ld_f32(xr + e) - mu) * is;
            const float d = ld_f32(dyr + e);
            const float dg = d * g[e];
            s1 += dg;
            s2 += dg * xh;
        }
        for (int off = WAVE / 2; off > 0; off >>= 1) {
            s1 += __shfl_down(s1, off, WAVE);
            s2 += __shfl_down(s2, off, WAVE);
        }
        s1 = __shfl(s1, 0, WAVE);
        s2 = __shfl(s2, 0, WAVE);
        const float inv_e = 1.f / E;
        const float k1 = s1 * inv_e, k2 = s2 * inv_e;
        T* dxr = dx + (long)row * E;
        // partials (rows, 2, E): [row][0][e] = dy*xhat, [row][1][e] = dy
        float* pg = partials + ((long)row * 2) * E;
        float* pb = pg + E;
        for (int e = l; e < E; e += WAVE) {
            const float xh = (ld_f32(xr + e) - mu) * is;
            const float d = ld_f32(dyr + e);
            st_f32(dxr + e, is * (d * g[e] - k1 - xh * k2));
            pg[e] = d * xh;
            pb[e] = d;
        }
    }
}

// reduce per-row partials (rows, 2, E) into (R, E) dgamma/dbeta in fixed
// row order (deterministic)
__global__ void __launch_bounds__(256)
ln_grad_reduce_kernel(const float* __restrict__ partials,
                      float* __restrict__ dgamma, float* __restrict__ dbeta,
                      int rows_per_client, int E) {
    const int r = blockIdx.x;
    for (int e = threadIdx.x; e < E; e += blockDim.x) {
        float sg = 0.f, sb = 0.f;
        const float* base = partials + (long)r * rows_per_client * 2 * E;
        for (int i = 0; i < rows_per_client; ++i) {
            sg += base[(long)i * 2 * E + e];
            sb += base[(long)i * 2 * E + E + e];
        }
        dgamma[(long)r * E + e] = sg;
        dbeta[(long)r * E + e] = sb;
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_LT(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> ln_fwd(at::Tensor x, at::Tensor gamma,
                               at::Tensor beta, int64_t R, double eps) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous());
    const int E = x.size(-1);
    const long rows = x.numel() / E;
    TORCH_CHECK(rows % R == 0 && E <= 1024);
    auto y = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto mean = at::empty({rows}, opts);
    auto invstd = at::empty({rows}, opts);
    auto stream = at::hip::getCurrentHIPStream();
    const int blocks = (int)((rows + 3) / 4);
    DISPATCH_LT(x.scalar_type(), {
        hipLaunchKernelGGL(ln_fwd_kernel<scalar_t>, dim3(blocks), dim3(256),
                           0, stream, (const scalar_t*)x.data_ptr(),
                           gamma.data_ptr<float>(), beta.data_ptr<float>(),
                           (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), (int)rows,
                           (int)(rows / R), E, (float)eps);
    });
    return {y, mean, invstd};
}

std::vector<at::Tensor> ln_bwd(at::Tensor dy, at::Tensor x, at::Tensor gamma,
                               at::Tensor mean, at::Tensor invstd,
                               int64_t R) {
    const int E = x.size(-1);
    const long rows = x.numel() / E;
    auto dx = at::empty_like(x);
    auto opts = x.options().dtype(at::kFloat);
    auto partials = at::empty({rows, 2, (long)E}, opts);
    auto dgamma = at::empty({R, (long)E}, opts);
    auto dbeta = at::empty({R, (long)E}, opts);
    auto dyc = dy.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    const int blocks = (int)((rows + 3) / 4);
    DISPATCH_LT(x.scalar_type(), {
        hipLaunchKernelGGL(ln_bwd_kernel<scalar_t>, dim3(blocks), dim3(256),
                           0, stream, (const scalar_t*)dyc.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           gamma.data_ptr<float>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(),
                           (scalar_t*)dx.data_ptr(),
                           partials.data_ptr<float>(), (int)rows,
                           (int)(rows / R), E);
    });
    hipLaunchKernelGGL(ln_grad_reduce_kernel, dim3((int)R), dim3(256), 0,
                       stream, partials.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       (int)(rows / R), E);
    return {dx, dgamma, dbeta};
}
