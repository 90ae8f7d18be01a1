// K6 of SURVEY §2b: 2x2/stride-2 MaxPool fwd/bwd for the batched conv
// model (reference: src/models/conv.py:33 nn.MaxPool2d(2)).  Forward saves
// the winning index per output element; backward scatters dy to it — one
// kernel each way, deterministic (each input cell has at most one pool
// window owner at stride 2).
#include "common.h"

template <typename T>
__global__ void __launch_bounds__(256)
maxpool2_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                    unsigned char* __restrict__ arg, int NC, int H, int W,
                    int OH, int OW) {
    const long n = (long)NC * OH * OW;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const int ow = i % OW;
        const int oh = (i / OW) % OH;
        const long nc = i / ((long)OH * OW);
        const long base = nc * H * W + (long)(2 * oh) * W + 2 * ow;
        float best = ld_f32(x + base);
        unsigned char a = 0;
        const bool okw = 2 * ow + 1 < W, okh = 2 * oh + 1 < H;
        if (okw) {
            const float v = ld_f32(x + base + 1);
            if (v > best) { best = v; a = 1; }
        }
        if (okh) {
            const float v = ld_f32(x + base + W);
            if (v > best) { best = v; a = 2; }
        }
        if (okw && okh) {
            const float v = ld_f32(x + base + W + 1);
            if (v > best) { best = v; a = 3; }
        }
        y[i] = (T)best;
        arg[i] = a;
    }
}

template <typename T>
__global__ void __launch_bounds__(256)
maxpool2_bwd_kernel(const T* __restrict__ dy,
                    const unsigned char* __restrict__ arg,
                    T* __restrict__ dx, int NC, int H, int W, int OH,
                    int OW) {
    const long n = (long)NC * OH * OW;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const int ow = i % OW;
        const int oh = (i / OW) % OH;
        const long nc = i / ((long)OH * OW);
        const unsigned char a = arg[i];
        const long base = nc * H * W + (long)(2 * oh) * W + 2 * ow;
        const long off = base + (a & 1) + (a >> 1) * W;
        st_f32(dx + off, ld_f32(dy + i));
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

static dim3 pool_grid(long n) {
    long blocks = (n + 256 * 4 - 1) / (256 * 4);
    if (blocks < 1) blocks = 1;
    if (blocks > 4096) blocks = 4096;
    return dim3((unsigned)blocks);
}

#define DISPATCH_PL(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> maxpool2_fwd(at::Tensor x) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
    const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    const int OH = H / 2, OW = W / 2;
    auto y = at::empty({N, C, OH, OW}, x.options());
    auto arg = at::empty({(long)N * C * OH * OW},
                         x.options().dtype(at::kByte));
    auto stream = at::hip::getCurrentHIPStream();
    const long n = (long)N * C * OH * OW;
    DISPATCH_PL(x.scalar_type(), {
        hipLaunchKernelGGL(maxpool2_fwd_kernel<scalar_t>, pool_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)x.data_ptr(),
                           (scalar_t*)y.data_ptr(),
                           arg.data_ptr<unsigned char>(), N * C, H, W, OH,
                           OW);
    });
    return {y, arg};
}

at::Tensor maxpool2_bwd(at::Tensor dy, at::Tensor arg, int64_t H,
                        int64_t W) {
    const int N = dy.size(0), C = dy.size(1), OH = dy.size(2),
              OW = dy.size(3);
    auto dyc = dy.contiguous();
    auto dx = at::zeros({N, C, H, W}, dy.options());
    auto stream = at::hip::getCurrentHIPStream();
    const long n = (long)N * C * OH * OW;
    DISPATCH_PL(dy.scalar_type(), {
        hipLaunchKernelGGL(maxpool2_bwd_kernel<scalar_t>, pool_grid(n),
                           dim3(256), 0, stream,
                           (const scalar_t*)dyc.data_ptr(),
                           arg.data_ptr<unsigned char>(),
                           (scalar_t*)dx.data_ptr(), N * C, (int)H, (int)W,
                           OH, OW);
    });
    return dx;
}
