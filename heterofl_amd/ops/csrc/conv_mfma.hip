// Hand-written gfx950 MFMA implicit-GEMM grouped convolution (K1/K2 of
// SURVEY.md §2b): forward, backward-data and backward-weight for the NCHW
// 3x3/1x1 convs of HeteroFL's batched client models (reference conv sites:
// src/models/resnet.py:33-42, src/models/conv.py:29).
//
// Why hand-written: the round's convs are tiny (batch 10, 32x32 .. 4x4
// spatial, grouped over R clients) — MIOpen/CK fall back to naive or
// heavyweight xdl kernels at 50-225us per call plus im2col/transpose glue
// (profiles/kstats_r01_fused_v1.txt).  One direct implicit-GEMM kernel per
// direction, bf16 inputs with fp32 MFMA accumulate (or exact-f32 MFMA for
// the fp32 path), no im2col materialization, fp32 weight gradients.
//
// GEMM view (per group g):
//   fwd:  y[m,p]  = sum_k  w[m,k] * patch[k,p]      M=Cout, K=Cin*kh*kw,
//                                                   P=N*OH*OW
//   bwdW: dw[m,k] = sum_p  dy[m,p] * patch[k,p]
//   bwdD: dx[c,q] = sum_j  w'[c,j] * dyp[j,q]       j=Cout*kh*kw, q=N*H*W
//
// Tiling: 256-thread workgroups (4 waves), 64x64 output tile, BK=32,
// mfma_f32_16x16x32_bf16 (or 8x mfma_f32_16x16x4_f32), LDS staged with
// +8-element row padding against ds_read_b128 bank conflicts.
// A-fragment: lane l holds A[row=l%16][k=(l/16)*8+j]; B-fragment:
// B[k=(l/16)*8+j][col=l%16]; C/D: row=(l/16)*4+reg, col=l%16
// (cdna_hip_programming.md §3 fragment layout).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int BM = 64;
constexpr int BP = 64;
constexpr int BK = 32;
constexpr int LDK = BK + 8;  // padded k-stride for [.][k]-contiguous tiles

__device__ __forceinline__ float bf_to_f(__hip_bfloat16 v) {
    return __bfloat162float(v);
}

// one 16x16 MFMA accumulation over a BK=32 K-slab, element type T
template <typename T>
__device__ __forceinline__ f32x4 mfma_tile(const T* a_row, const T* b_col,
                                           f32x4 acc);

// bf16: a_row points at A_lds[row][0] (LDK stride), b_col at B_lds[col][0]
template <>
__device__ __forceinline__ f32x4 mfma_tile<__hip_bfloat16>(
        const __hip_bfloat16* a_row, const __hip_bfloat16* b_col, f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int kb = (l >> 4) * 8;
    bf16x8 a = *(const bf16x8*)(a_row + kb);
    bf16x8 b = *(const bf16x8*)(b_col + kb);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
}

template <>
__device__ __forceinline__ f32x4 mfma_tile<float>(const float* a_row,
                                                  const float* b_col,
                                                  f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int ko = l >> 4;
#pragma unroll
    for (int q = 0; q < 8; ++q) {
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a_row[q * 4 + ko],
                                                   b_col[q * 4 + ko], acc, 0,
                                                   0, 0);
    }
    return acc;
}

// ---------------------------------------------------------------- geometry
struct ConvGeom {
    int G;        // groups
    int N, H, W;  // input spatial
    int Cin, Cout;
    int OH, OW;
    int khw;      // kernel size (3 or 1)
    int stride, pad;
};

// -------------------------------------------------------------- fwd kernel
// grid: (ceil(M/BM), ceil(P/BP), G); x (N, G*Cin, H, W) T; w (G*Cout,
// Cin, khw, khw) float (converted on load); y (N, G*Cout, OH, OW) T.
template <typename T>
__global__ void __launch_bounds__(256)
conv_fwd_kernel(const T* __restrict__ x, const float* __restrict__ w,
                const float* __restrict__ bias, T* __restrict__ y,
                ConvGeom gm) {
    __shared__ T a_lds[BM][LDK];
    __shared__ T b_lds[BP][LDK];
    const int g = blockIdx.z;
    const int m0 = blockIdx.x * BM;
    const int p0 = blockIdx.y * BP;
    const int K = gm.Cin * gm.khw * gm.khw;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int GC = gm.G * gm.Cin;
    const int GM = gm.G * gm.Cout;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int l = tid & (WAVE - 1);
    const int wm = (wave >> 1) * 32;  // wave's 32x32 quadrant
    const int wp = (wave & 1) * 32;
    f32x4 acc[2][2] = {};
    const int kk2 = gm.khw * gm.khw;

    for (int k0 = 0; k0 < K; k0 += BK) {
        // A tile: w[g*Cout + m0+mm][k0+kk], fp32 -> T
        for (int e = tid; e < BM * BK; e += 256) {
            const int kk = e & (BK - 1), mm = e >> 5;
            const int m = m0 + mm, k = k0 + kk;
            float v = 0.f;
            if (m < M && k < K) v = w[(long)(g * gm.Cout + m) * K + k];
            a_lds[mm][kk] = (T)v;
        }
        // B tile: patch[k0+kk][p0+pp] stored [pp][kk]
        for (int e = tid; e < BK * BP; e += 256) {
            const int pp = e & (BP - 1), kk = e >> 6;
            const int k = k0 + kk, p = p0 + pp;
            float v = 0.f;
            if (k < K && p < P) {
                const int cin = k / kk2, r = k - cin * kk2;
                const int kh = r / gm.khw, kw = r - kh * gm.khw;
                const int n = p / OHW, hw = p - n * OHW;
                const int oh = hw / gm.OW, ow = hw - oh * gm.OW;
                const int ih = oh * gm.stride + kh - gm.pad;
                const int iw = ow * gm.stride + kw - gm.pad;
                if (ih >= 0 && ih < gm.H && iw >= 0 && iw < gm.W)
                    v = ld_f32(x + ((long)(n * GC + g * gm.Cin + cin) * gm.H
                                    + ih) * gm.W + iw);
            }
            b_lds[pp][kk] = (T)v;
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile<T>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    // epilogue: D[row][col] with row->m, col->p
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                const int p = p0 + wp + fp * 16 + (l & 15);
                if (m < M && p < P) {
                    const int n = p / OHW, hw = p - n * OHW;
                    const float b = bias ? bias[g * gm.Cout + m] : 0.f;
                    st_f32(y + (long)(n * GM + g * gm.Cout + m) * OHW + hw,
                           acc[fm][fp][r] + b);
                }
            }
}

// -------------------------------------------------------- bwd-data kernel
// dx[c,q]: A'[c][j] = w[g*Cout + cout][c*khw2 + kh*khw + kw] with
// j = cout*khw2 + kh*khw + kw; B'[j][q] = dy at (oh,ow) if stride divides.
template <typename T>
__global__ void __launch_bounds__(256)
conv_bwd_data_kernel(const T* __restrict__ dy, const float* __restrict__ w,
                     T* __restrict__ dx, ConvGeom gm) {
    __shared__ T a_lds[BM][LDK];
    __shared__ T b_lds[BP][LDK];
    const int g = blockIdx.z;
    const int c0 = blockIdx.x * BM;   // over Cin
    const int q0 = blockIdx.y * BP;   // over N*H*W
    const int kk2 = gm.khw * gm.khw;
    const int J = gm.Cout * kk2;
    const int HW = gm.H * gm.W;
    const int Q = gm.N * HW;
    const int OHW = gm.OH * gm.OW;
    const int GC = gm.G * gm.Cin;
    const int GM = gm.G * gm.Cout;
    const int K = gm.Cin * kk2;
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int l = tid & (WAVE - 1);
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    f32x4 acc[2][2] = {};

    for (int j0 = 0; j0 < J; j0 += BK) {
        for (int e = tid; e < BM * BK; e += 256) {
            const int jj = e & (BK - 1), cc = e >> 5;
            const int c = c0 + cc, j = j0 + jj;
            float v = 0.f;
            if (c < gm.Cin && j < J) {
                const int cout = j / kk2, r = j - cout * kk2;
                v = w[(long)(g * gm.Cout + cout) * K + c * kk2 + r];
            }
            a_lds[cc][jj] = (T)v;
        }
        for (int e = tid; e < BK * BP; e += 256) {
            const int qq = e & (BP - 1), jj = e >> 6;
            const int j = j0 + jj, q = q0 + qq;
            float v = 0.f;
            if (j < J && q < Q) {
                const int cout = j / kk2, r = j - cout * kk2;
                const int kh = r / gm.khw, kw = r - kh * gm.khw;
                const int n = q / HW, hw = q - n * HW;
                const int ih = hw / gm.W, iw = hw - ih * gm.W;
                const int ohs = ih + gm.pad - kh;
                const int ows = iw + gm.pad - kw;
                if (ohs >= 0 && ows >= 0 && ohs % gm.stride == 0 &&
                    ows % gm.stride == 0) {
                    const int oh = ohs / gm.stride, ow = ows / gm.stride;
                    if (oh < gm.OH && ow < gm.OW)
                        v = ld_f32(dy + ((long)(n * GM + g * gm.Cout + cout)
                                         * gm.OH + oh) * gm.OW + ow);
                }
            }
            b_lds[qq][jj] = (T)v;
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile<T>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int c = c0 + wm + fm * 16 + (l >> 4) * 4 + r;
                const int q = q0 + wp + fp * 16 + (l & 15);
                if (c < gm.Cin && q < Q) {
                    const int n = q / HW, hw = q - n * HW;
                    st_f32(dx + (long)(n * GC + g * gm.Cin + c) * HW + hw,
                           acc[fm][fp][r]);
                }
            }
}

// ------------------------------------------------------ bwd-weight kernel
// dw[m][k] (fp32 out) = sum_p dy[m][p] * patch[k][p]; grid z = G * SPLITP,
// split-P partials accumulated with atomicAdd (dw zeroed by caller).
template <typename T>
__global__ void __launch_bounds__(256)
conv_bwd_weight_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                       float* __restrict__ dw, ConvGeom gm, int splitp) {
    __shared__ T a_lds[BM][LDK];
    __shared__ T b_lds[BP][LDK];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int k0 = blockIdx.y * BP;   // over K = Cin*khw2
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int GC = gm.G * gm.Cin;
    const int GM = gm.G * gm.Cout;
    const int pchunk = (P + splitp - 1) / splitp;
    const int pstart = sp * pchunk;
    const int pend = min(P, pstart + pchunk);
    const int tid = threadIdx.x;
    const int wave = tid / WAVE;
    const int l = tid & (WAVE - 1);
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    f32x4 acc[2][2] = {};

    for (int p0 = pstart; p0 < pend; p0 += BK) {
        // A tile: dy[m][p0+pp] stored [mm][pp]
        for (int e = tid; e < BM * BK; e += 256) {
            const int pp = e & (BK - 1), mm = e >> 5;
            const int m = m0 + mm, p = p0 + pp;
            float v = 0.f;
            if (m < M && p < pend) {
                const int n = p / OHW, hw = p - n * OHW;
                v = ld_f32(dy + (long)(n * GM + g * gm.Cout + m) * OHW + hw);
            }
            a_lds[mm][pp] = (T)v;
        }
        // B tile: patch[k0+kk][p0+pp] stored [kk][pp]
        for (int e = tid; e < BK * BP; e += 256) {
            const int pp = e & (BK - 1), kk = e >> 5;
            const int k = k0 + kk, p = p0 + pp;
            float v = 0.f;
            if (k < K && p < pend) {
                const int cin = k / kk2, r = k - cin * kk2;
                const int kh = r / gm.khw, kw = r - kh * gm.khw;
                const int n = p / OHW, hw = p - n * OHW;
                const int oh = hw / gm.OW, ow = hw - oh * gm.OW;
                const int ih = oh * gm.stride + kh - gm.pad;
                const int iw = ow * gm.stride + kw - gm.pad;
                if (ih >= 0 && ih < gm.H && iw >= 0 && iw < gm.W)
                    v = ld_f32(x + ((long)(n * GC + g * gm.Cin + cin) * gm.H
                                    + ih) * gm.W + iw);
            }
            b_lds[kk][pp] = (T)v;
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile<T>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                const int k = k0 + wp + fp * 16 + (l & 15);
                if (m < M && k < K) {
                    float* out = dw + (long)(g * gm.Cout + m) * K + k;
                    if (splitp > 1)
                        atomicAdd(out, acc[fm][fp][r]);
                    else
                        *out = acc[fm][fp][r];
                }
            }
}

// ------------------------------------------------------------ host layer
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_CONV_FT(t, ...)                                              \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

static ConvGeom make_geom(const at::Tensor& x, const at::Tensor& w,
                          int64_t groups, int64_t stride, int64_t pad) {
    ConvGeom gm;
    gm.G = (int)groups;
    gm.N = x.size(0);
    gm.H = x.size(2);
    gm.W = x.size(3);
    gm.Cin = (int)w.size(1);
    gm.Cout = (int)(w.size(0) / groups);
    gm.khw = (int)w.size(2);
    gm.stride = (int)stride;
    gm.pad = (int)pad;
    gm.OH = (gm.H + 2 * gm.pad - gm.khw) / gm.stride + 1;
    gm.OW = (gm.W + 2 * gm.pad - gm.khw) / gm.stride + 1;
    return gm;
}

at::Tensor conv_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                    int64_t groups, int64_t stride, int64_t pad) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
    TORCH_CHECK(w.scalar_type() == at::kFloat, "weights must be fp32 master");
    auto gm = make_geom(x, w, groups, stride, pad);
    auto y = at::empty({gm.N, gm.G * gm.Cout, gm.OH, gm.OW}, x.options());
    const int M = gm.Cout, P = gm.N * gm.OH * gm.OW;
    dim3 grid((M + BM - 1) / BM, (P + BP - 1) / BP, gm.G);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_CONV_FT(x.scalar_type(), {
        hipLaunchKernelGGL(conv_fwd_kernel<scalar_t>, grid, dim3(256), 0,
                           stream, (const scalar_t*)x.data_ptr(),
                           w.data_ptr<float>(),
                           bias.defined() ? bias.data_ptr<float>() : nullptr,
                           (scalar_t*)y.data_ptr(), gm);
    });
    return y;
}

at::Tensor conv_bwd_data(at::Tensor dy, at::Tensor w, int64_t groups,
                         int64_t stride, int64_t pad, int64_t H, int64_t W) {
    TORCH_CHECK(dy.is_cuda() && w.is_contiguous());
    auto dyc = dy.contiguous();
    ConvGeom gm;
    gm.G = (int)groups;
    gm.N = dy.size(0);
    gm.H = (int)H;
    gm.W = (int)W;
    gm.Cin = (int)w.size(1);
    gm.Cout = (int)(w.size(0) / groups);
    gm.khw = (int)w.size(2);
    gm.stride = (int)stride;
    gm.pad = (int)pad;
    gm.OH = dy.size(2);
    gm.OW = dy.size(3);
    auto dx = at::empty({gm.N, gm.G * gm.Cin, gm.H, gm.W}, dy.options());
    const int Q = gm.N * gm.H * gm.W;
    dim3 grid((gm.Cin + BM - 1) / BM, (Q + BP - 1) / BP, gm.G);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_CONV_FT(dy.scalar_type(), {
        hipLaunchKernelGGL(conv_bwd_data_kernel<scalar_t>, grid, dim3(256), 0,
                           stream, (const scalar_t*)dyc.data_ptr(),
                           w.data_ptr<float>(), (scalar_t*)dx.data_ptr(), gm);
    });
    return dx;
}

at::Tensor conv_bwd_weight(at::Tensor dy, at::Tensor x, int64_t groups,
                           int64_t stride, int64_t pad, int64_t khw) {
    auto dyc = dy.contiguous();
    ConvGeom gm;
    gm.G = (int)groups;
    gm.N = x.size(0);
    gm.H = x.size(2);
    gm.W = x.size(3);
    gm.Cin = (int)(x.size(1) / groups);
    gm.Cout = (int)(dy.size(1) / groups);
    gm.khw = (int)khw;
    gm.stride = (int)stride;
    gm.pad = (int)pad;
    gm.OH = dy.size(2);
    gm.OW = dy.size(3);
    const int K = gm.Cin * gm.khw * gm.khw;
    const int P = gm.N * gm.OH * gm.OW;
    // split P so the grid fills the 256-CU chip when M,K tiles are few
    const int mk_tiles = ((gm.Cout + BM - 1) / BM) * ((K + BP - 1) / BP) * gm.G;
    int splitp = 1;
    while (mk_tiles * splitp < 256 && splitp * BK * 4 < P) splitp *= 2;
    auto dw = at::empty({(long)gm.G * gm.Cout, gm.Cin, gm.khw, gm.khw},
                        x.options().dtype(at::kFloat));
    if (splitp > 1) dw.zero_();
    dim3 grid((gm.Cout + BM - 1) / BM, (K + BP - 1) / BP, gm.G * splitp);
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_CONV_FT(x.scalar_type(), {
        hipLaunchKernelGGL(conv_bwd_weight_kernel<scalar_t>, grid, dim3(256),
                           0, stream, (const scalar_t*)dyc.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           dw.data_ptr<float>(), gm, splitp);
    });
    return dw;
}

// --------------------------------------------------- MFMA layout self-test
// One 16x16x32 bf16 tile with the fragment layout assumed above; the GPU
// test compares against torch matmul to pin the layout empirically.
__global__ void mfma_probe_kernel(const float* __restrict__ A,
                                  const float* __restrict__ B,
                                  float* __restrict__ D) {
    const int l = threadIdx.x;
    const int kb = (l >> 4) * 8;
    bf16x8 a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        a[j] = (__bf16)A[(l & 15) * 32 + kb + j];   // A[row][k]
        b[j] = (__bf16)B[(kb + j) * 16 + (l & 15)]; // B[k][col]
    }
    f32x4 acc = {};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r)
        D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
}

at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
    TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) &&
                B.sizes() == at::IntArrayRef({32, 16}));
    auto Ac = A.contiguous().to(at::kFloat).cuda();
    auto Bc = B.contiguous().to(at::kFloat).cuda();
    auto D = at::empty({16, 16}, Ac.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                       Ac.data_ptr<float>(), Bc.data_ptr<float>(),
                       D.data_ptr<float>());
    return D;
}
