// Hand-written gfx950 MFMA implicit-GEMM grouped convolution (K1/K2 of
// SURVEY.md §2b): forward (+fused residual add), backward-data and
// deterministic split-P backward-weight for the NCHW 3x3/1x1 convs of
// HeteroFL's batched client models (reference conv sites:
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
//   bwdW: dw[m,k] = sum_p  dy[m,p] * patch[k,p]     (split-P partials +
//                                                    deterministic reduce)
//   bwdD: dx[c,q] = sum_j  w'[c,j] * dyp[j,q]       j=Cout*kh*kw, q=N*H*W
//
// Tiling: 256-thread workgroups (4 waves), 64x64 output tile, BK=32,
// mfma_f32_16x16x32_bf16 (or 8x mfma_f32_16x16x4_f32), LDS tiles stored
// [row][k] with +8-element padding; per-block index-decomposition tables
// (pixel -> (n, ihbase, iwbase)) remove the per-element div/mod chains from
// the gather loaders.  Fragment layout (verified on hardware by
// tests/test_gpu.py::test_mfma_probe_layout): A: lane l holds
// A[row=l%16][k=(l/16)*8+j]; B: B[k=(l/16)*8+j][col=l%16]; C/D:
// row=(l/16)*4+reg, col=l%16.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int BM = 64;
constexpr int BP = 64;
constexpr int BK = 32;
constexpr int LDK = BK + 8;  // padded k-stride for [.][k]-contiguous tiles

// one 16x16 MFMA accumulation over a BK=32 K-slab, element type T
template <typename T>
__device__ __forceinline__ f32x4 mfma_tile(const T* a_row, const T* b_col,
                                           f32x4 acc);

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


// store 8 fp32 values as one contiguous 16B (bf16) / 32B (f32) LDS write
__device__ __forceinline__ void st8_lds(__hip_bfloat16* dst, const float* v) {
    bf16x8 t;
#pragma unroll
    for (int j = 0; j < 8; ++j) t[j] = (__bf16)v[j];
    *(bf16x8*)dst = t;
}
__device__ __forceinline__ void st8_lds(float* dst, const float* v) {
    typedef __attribute__((ext_vector_type(4))) float f4;
    *(f4*)dst = *(const f4*)v;
    *(f4*)(dst + 4) = *(const f4*)(v + 4);
}

// fp8 LDS tile element tags (OCP formats; gfx950 fp8 MFMA is e4m3/e5m2 —
// cdna_hip_programming.md §4).  e4m3 carries weights/activations, e5m2
// gradients; the mixed mfma_f32_16x16x32_{fp8,bf8}_{fp8,bf8} forms multiply
// them directly (CDNA4 fp8 MFMA, BASELINE config 5).
#include <hip/hip_fp8.h>
struct e4m3 { unsigned char x; };
struct e5m2 { unsigned char x; };
typedef long long i64;

__device__ __forceinline__ void st8_lds(e4m3* dst, const float* v) {
    unsigned char t[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
        t[j] = __hip_cvt_float_to_fp8(v[j], __HIP_SATFINITE, __HIP_E4M3);
    *(i64*)dst = *(const i64*)t;
}
__device__ __forceinline__ void st8_lds(e5m2* dst, const float* v) {
    unsigned char t[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
        t[j] = __hip_cvt_float_to_fp8(v[j], __HIP_SATFINITE, __HIP_E5M2);
    *(i64*)dst = *(const i64*)t;
}

// mixed-type MFMA: a_row/b_col point at the lane's LDS rows; the k-offset
// (l/16)*8 selects the lane's 8-element fragment (layout as above).
template <typename TA, typename TB>
__device__ __forceinline__ f32x4 mfma_tile2(const TA* a_row, const TB* b_col,
                                            f32x4 acc) {
    return mfma_tile<TA>(a_row, b_col, acc);  // TA == TB fast path
}
template <>
__device__ __forceinline__ f32x4 mfma_tile2<e4m3, e4m3>(const e4m3* a_row,
                                                        const e4m3* b_col,
                                                        f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int kb = (l >> 4) * 8;
    i64 a = *(const i64*)(a_row + kb);
    i64 b = *(const i64*)(b_col + kb);
    return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc, 0, 0, 0);
}
template <>
__device__ __forceinline__ f32x4 mfma_tile2<e4m3, e5m2>(const e4m3* a_row,
                                                        const e5m2* b_col,
                                                        f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int kb = (l >> 4) * 8;
    i64 a = *(const i64*)(a_row + kb);
    i64 b = *(const i64*)(b_col + kb);
    return __builtin_amdgcn_mfma_f32_16x16x32_fp8_bf8(a, b, acc, 0, 0, 0);
}
template <>
__device__ __forceinline__ f32x4 mfma_tile2<e5m2, e4m3>(const e5m2* a_row,
                                                        const e4m3* b_col,
                                                        f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int kb = (l >> 4) * 8;
    i64 a = *(const i64*)(a_row + kb);
    i64 b = *(const i64*)(b_col + kb);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf8_fp8(a, b, acc, 0, 0, 0);
}

struct ConvGeom {
    int G;
    int N, H, W;
    int Cin, Cout;
    int OH, OW;
    int khw;
    int stride, pad;
};

// -------------------------------------------------------------- fwd kernel
// grid: (ceil(M/BM), ceil(P/BP), G).  Optional residual is added in the
// epilogue (the ResNet block's `out += shortcut`, src/models/resnet.py:49).
// splitk > 1: each split writes fp32 partial slabs (sp, N*G*Cout*OHW);
// conv_out_reduce_kernel sums them in fixed order and applies bias/residual.
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_fwd_kernel(const T* __restrict__ x, const float* __restrict__ w,
                const float* __restrict__ bias, const T* __restrict__ residual,
                T* __restrict__ y, float* __restrict__ partial, ConvGeom gm,
                int splitk) {
    __shared__ TA a_lds[2][BM][LDK];
    __shared__ TB b_lds[2][BP][LDK];
    __shared__ int t_ihb[BP], t_iwb[BP];
    __shared__ long t_xbase[BP], t_ybase[BP];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int p0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int HW = gm.H * gm.W;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int nkc = ((K + BK - 1) / BK + splitk - 1) / splitk;  // BK-chunks
    const int ks = sp * nkc * BK;
    const int ke = min(K, ks + nkc * BK);

    // per-block pixel decomposition tables
    if (tid < BP) {
        const int p = p0 + tid;
        if (p < P) {
            const int n = p / OHW, hw = p - n * OHW;
            const int oh = hw / gm.OW, ow = hw - oh * gm.OW;
            t_ihb[tid] = oh * gm.stride - gm.pad;
            t_iwb[tid] = ow * gm.stride - gm.pad;
            t_xbase[tid] = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW;
            t_ybase[tid] = ((long)n * gm.G * gm.Cout + (long)g * gm.Cout) * OHW
                           + hw;
        } else {
            t_ihb[tid] = 1 << 28;  // forces out-of-bounds -> zeros
            t_iwb[tid] = 1 << 28;
            t_xbase[tid] = 0;
            t_ybase[tid] = -1;
        }
    }
    __syncthreads();

    f32x4 acc[2][2] = {};
    const int mm_a = tid >> 2, kkb = (tid & 3) * 8;
    const int pp_b = tid >> 2;
    float va[8], vb[8];
    // loader for one K-step into registers (2-phase: issue next tile's
    // loads before this tile's MFMA so HBM/L2 latency hides under it)
    auto load_regs = [&](int k0) {
        const int m = m0 + mm_a;
        const float* wrow = w + (long)(g * gm.Cout + m) * K + k0 + kkb;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int k = k0 + kkb + j;
            va[j] = (m < M && k < K) ? wrow[j] : 0.f;
        }
        // incremental k -> (cin, kh, kw): one divide per 8 contiguous k
        // (runtime integer divides cost ~60 VALU cycles each)
        int k = k0 + kkb;
        int cin = k / kk2, r = k - cin * kk2;
        int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            vb[j] = 0.f;
            if (k + j < K) {
                const int ih = t_ihb[pp_b] + kh, iw = t_iwb[pp_b] + kw;
                if (ih >= 0 && ih < gm.H && iw >= 0 && iw < gm.W)
                    vb[j] = ld_f32(x + t_xbase[pp_b] + (long)cin * HW
                                   + ih * gm.W + iw);
            }
            if (++kw == gm.khw) {
                kw = 0;
                if (++kh == gm.khw) {
                    kh = 0;
                    ++cin;
                }
            }
        }
    };
    // double-buffered k-loop: ONE barrier per iteration; tile t+1's loads
    // stay in flight across tile t's whole MFMA phase
    load_regs(ks);
    st8_lds(&a_lds[0][mm_a][kkb], va);
    st8_lds(&b_lds[0][pp_b][kkb], vb);
    if (ks + BK < ke) load_regs(ks + BK);
    __syncthreads();
    int cur = 0;
    for (int k0 = ks; k0 < ke; k0 += BK) {
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[cur][wm + fm * 16 + (l & 15)][0],
                    &b_lds[cur][wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        if (k0 + BK < ke) {
            st8_lds(&a_lds[cur ^ 1][mm_a][kkb], va);
            st8_lds(&b_lds[cur ^ 1][pp_b][kkb], vb);
            if (k0 + 2 * BK < ke) load_regs(k0 + 2 * BK);
        }
        __syncthreads();
        cur ^= 1;
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cout * OHW;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int pp = wp + fp * 16 + (l & 15);
            const long yb = t_ybase[pp];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (m < M && yb >= 0) {
                    const long off = yb + (long)m * OHW;
                    float v = acc[fm][fp][r];
                    if (splitk == 1) {
                        if (bias) v += bias[g * gm.Cout + m];
                        if (residual) v += ld_f32(residual + off);
                        st_f32(y + off, v);
                    } else {
                        partial[slab + off] = v;
                    }
                }
            }
        }
}

// sum split-K partials in fixed order; add bias/residual; cast to T
template <typename T>
__global__ void __launch_bounds__(256)
conv_out_reduce_kernel(const float* __restrict__ partial,
                       const float* __restrict__ bias,
                       const T* __restrict__ residual, T* __restrict__ out,
                       long total, long chanstride, int nchan, int splitk) {
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= total) return;
    float v = 0.f;
    for (int sp = 0; sp < splitk; ++sp) v += partial[(long)sp * total + i];
    if (bias) v += bias[(i / chanstride) % nchan];
    if (residual) v += ld_f32(residual + i);
    st_f32(out + i, v);
}

// -------------------------------------------------------- bwd-data kernel
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_data_kernel(const T* __restrict__ dy, const float* __restrict__ w,
                     T* __restrict__ dx, float* __restrict__ partial,
                     ConvGeom gm, int splitk) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ int t_oh[BP], t_ow[BP];  // ih+pad, iw+pad (pre-division)
    __shared__ long t_dybase[BP], t_xbase[BP];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int c0 = blockIdx.x * BM;
    const int q0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int J = gm.Cout * kk2;
    const int K = gm.Cin * kk2;
    const int HW = gm.H * gm.W;
    const int Q = gm.N * HW;
    const int OHW = gm.OH * gm.OW;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;

    if (tid < BP) {
        const int q = q0 + tid;
        if (q < Q) {
            const int n = q / HW, hw = q - n * HW;
            const int ih = hw / gm.W, iw = hw - ih * gm.W;
            t_oh[tid] = ih + gm.pad;
            t_ow[tid] = iw + gm.pad;
            t_dybase[tid] = ((long)n * gm.G * gm.Cout + (long)g * gm.Cout)
                            * OHW;
            t_xbase[tid] = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW
                           + hw;
        } else {
            t_oh[tid] = -(1 << 28);
            t_ow[tid] = -(1 << 28);
            t_dybase[tid] = 0;
            t_xbase[tid] = -1;
        }
    }
    __syncthreads();

    const int njc = ((J + BK - 1) / BK + splitk - 1) / splitk;
    const int js = sp * njc * BK;
    const int je = min(J, js + njc * BK);
    f32x4 acc[2][2] = {};
    const int cc_a = tid >> 2, jjb = (tid & 3) * 8;
    float va[8], vb[8];
    auto load_regs = [&](int j0) {
        const int c = c0 + cc_a;
#pragma unroll
        for (int j8 = 0; j8 < 8; ++j8) {
            const int j = j0 + jjb + j8;
            va[j8] = 0.f;
            if (c < gm.Cin && j < J) {
                const int cout = j / kk2, r = j - cout * kk2;
                va[j8] = w[(long)(g * gm.Cout + cout) * K + c * kk2 + r];
            }
        }
#pragma unroll
        for (int j8 = 0; j8 < 8; ++j8) {
            const int j = j0 + jjb + j8;
            vb[j8] = 0.f;
            if (j < J) {
                const int cout = j / kk2, r = j - cout * kk2;
                const int kh = r / gm.khw, kw = r - kh * gm.khw;
                const int ohs = t_oh[cc_a] - kh, ows = t_ow[cc_a] - kw;
                if (ohs >= 0 && ows >= 0) {
                    if (gm.stride == 1) {
                        if (ohs < gm.OH && ows < gm.OW)
                            vb[j8] = ld_f32(dy + t_dybase[cc_a]
                                            + (long)cout * OHW
                                            + ohs * gm.OW + ows);
                    } else if ((ohs & 1) == 0 && (ows & 1) == 0) {
                        const int oh = ohs >> 1, ow = ows >> 1;
                        if (oh < gm.OH && ow < gm.OW)
                            vb[j8] = ld_f32(dy + t_dybase[cc_a]
                                            + (long)cout * OHW
                                            + oh * gm.OW + ow);
                    }
                }
            }
        }
    };
    if (js < je) load_regs(js);
    for (int j0 = js; j0 < je; j0 += BK) {
        st8_lds(&a_lds[cc_a][jjb], va);
        st8_lds(&b_lds[cc_a][jjb], vb);
        __syncthreads();
        if (j0 + BK < je) load_regs(j0 + BK);
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cin * HW;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int qq = wp + fp * 16 + (l & 15);
            const long xb = t_xbase[qq];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int c = c0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (c < gm.Cin && xb >= 0) {
                    if (splitk == 1)
                        st_f32(dx + xb + (long)c * HW, acc[fm][fp][r]);
                    else
                        partial[slab + xb + (long)c * HW] = acc[fm][fp][r];
                }
            }
        }
}


// ------------------------------------------- stride-2 bwd-data (parity)
// For stride 2 only 1/4 of the (kh, kw) taps can contribute to a given
// input pixel (ohs/ows parity).  Pixels are tiled by parity class
// (ph, pw): each class is a dense (H/2, W/2) sub-image with a compressed
// tap list, so no MFMA work is spent on dead taps.  grid.y = 4 * tiles.
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_data_s2_kernel(const T* __restrict__ dy, const float* __restrict__ w,
                        T* __restrict__ dx, float* __restrict__ partial,
                        ConvGeom gm, int splitk, int nyp) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ int t_oh2[BP], t_ow2[BP];
    __shared__ long t_dybase[BP], t_xbase[BP];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int c0 = blockIdx.x * BM;
    const int cls = blockIdx.y / nyp;
    const int q0 = (blockIdx.y - cls * nyp) * BP;
    const int ph = cls >> 1, pw = cls & 1;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int H2 = gm.H >> 1, W2 = gm.W >> 1;
    const int HW2 = H2 * W2;
    const int Qp = gm.N * HW2;
    const int HW = gm.H * gm.W;
    const int OHW = gm.OH * gm.OW;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    // compressed tap lists for this parity class (khw <= 3)
    int lkh[3], lkw[3], skh[3], skw[3];
    int nkh = 0, nkw = 0;
    for (int kh = 0; kh < gm.khw; ++kh)
        if (((ph + gm.pad - kh) & 1) == 0) {
            lkh[nkh] = kh;
            skh[nkh] = (ph + gm.pad - kh) >> 1;
            ++nkh;
        }
    for (int kw = 0; kw < gm.khw; ++kw)
        if (((pw + gm.pad - kw) & 1) == 0) {
            lkw[nkw] = kw;
            skw[nkw] = (pw + gm.pad - kw) >> 1;
            ++nkw;
        }
    const int tap2 = nkh * nkw;
    const int J = gm.Cout * tap2;

    if (tid < BP) {
        const int q = q0 + tid;
        if (q < Qp) {
            const int n = q / HW2, hw2 = q - n * HW2;
            const int ih2 = hw2 / W2, iw2 = hw2 - ih2 * W2;
            t_oh2[tid] = ih2;
            t_ow2[tid] = iw2;
            t_dybase[tid] = ((long)n * gm.G * gm.Cout + (long)g * gm.Cout)
                            * OHW;
            t_xbase[tid] = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW
                           + (2 * ih2 + ph) * gm.W + (2 * iw2 + pw);
        } else {
            t_oh2[tid] = -(1 << 28);
            t_ow2[tid] = -(1 << 28);
            t_dybase[tid] = 0;
            t_xbase[tid] = -1;
        }
    }
    __syncthreads();

    const int njc = ((J + BK - 1) / BK + splitk - 1) / splitk;
    const int js = sp * njc * BK;
    const int je = min(J, js + njc * BK);
    f32x4 acc[2][2] = {};
    const int cc_a = tid >> 2, jjb = (tid & 3) * 8;
    float va[8], vb[8];
    auto load_regs = [&](int j0) {
        const int c = c0 + cc_a;
#pragma unroll
        for (int j8 = 0; j8 < 8; ++j8) {
            const int j = j0 + jjb + j8;
            va[j8] = 0.f;
            if (c < gm.Cin && j < J) {
                const int cout = j / tap2, rr = j - cout * tap2;
                const int kh = lkh[rr / nkw], kw = lkw[rr - (rr / nkw) * nkw];
                va[j8] = w[(long)(g * gm.Cout + cout) * K + c * kk2
                           + kh * gm.khw + kw];
            }
        }
#pragma unroll
        for (int j8 = 0; j8 < 8; ++j8) {
            const int j = j0 + jjb + j8;
            vb[j8] = 0.f;
            if (j < J) {
                const int cout = j / tap2, rr = j - cout * tap2;
                const int a = rr / nkw, b = rr - a * nkw;
                const int oh = t_oh2[cc_a] + skh[a];
                const int ow = t_ow2[cc_a] + skw[b];
                if (oh >= 0 && oh < gm.OH && ow >= 0 && ow < gm.OW)
                    vb[j8] = ld_f32(dy + t_dybase[cc_a] + (long)cout * OHW
                                    + oh * gm.OW + ow);
            }
        }
    };
    if (js < je) load_regs(js);
    for (int j0 = js; j0 < je; j0 += BK) {
        st8_lds(&a_lds[cc_a][jjb], va);
        st8_lds(&b_lds[cc_a][jjb], vb);
        __syncthreads();
        if (j0 + BK < je) load_regs(j0 + BK);
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cin * HW;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int qq = wp + fp * 16 + (l & 15);
            const long xb = t_xbase[qq];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int c = c0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (c < gm.Cin && xb >= 0) {
                    if (splitk == 1)
                        st_f32(dx + xb + (long)c * HW, acc[fm][fp][r]);
                    else
                        partial[slab + xb + (long)c * HW] = acc[fm][fp][r];
                }
            }
        }
}

// ------------------------------------------------------ bwd-weight kernel
// Writes per-split partials (splitp, G*Cout*K) when splitp > 1; a
// deterministic reduce kernel sums them in fixed order (no atomics, so the
// hipGraph-captured step replays bit-identically run to run).
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_weight_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                       float* __restrict__ out, ConvGeom gm, int splitp) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ int t_kcin[BP], t_kkh[BP], t_kkw[BP];   // per-block k tables
    __shared__ int t_ihb[BK], t_iwb[BK];               // per-step p tables
    __shared__ long t_xb[BK], t_dyb[BK];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int k0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int HW = gm.H * gm.W;
    const long GK = (long)gm.G * gm.Cout * K;  // one split's slab
    const int pchunk = (P + splitp - 1) / splitp;
    const int pstart = sp * pchunk;
    const int pend = min(P, pstart + pchunk);
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    f32x4 acc[2][2] = {};

    if (tid < BP) {
        const int k = k0 + tid;
        if (k < K) {
            const int cin = k / kk2, r = k - cin * kk2;
            t_kcin[tid] = cin;
            t_kkh[tid] = r / gm.khw;
            t_kkw[tid] = r - (r / gm.khw) * gm.khw;
        } else {
            t_kcin[tid] = 0;
            t_kkh[tid] = 1 << 28;
            t_kkw[tid] = 1 << 28;
        }
    }

    for (int p0 = pstart; p0 < pend; p0 += BK) {
        __syncthreads();
        if (tid < BK) {
            const int p = p0 + tid;
            if (p < pend) {
                const int n = p / OHW, hw = p - n * OHW;
                const int oh = hw / gm.OW, ow = hw - oh * gm.OW;
                t_ihb[tid] = oh * gm.stride - gm.pad;
                t_iwb[tid] = ow * gm.stride - gm.pad;
                t_xb[tid] = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW;
                t_dyb[tid] = ((long)n * gm.G * gm.Cout
                              + (long)g * gm.Cout) * OHW + hw;
            } else {
                t_ihb[tid] = 1 << 28;
                t_iwb[tid] = 1 << 28;
                t_xb[tid] = 0;
                t_dyb[tid] = -1;
            }
        }
        __syncthreads();
        {   // a tile rows: 64 rows x 32 p; thread owns 8 contiguous p
            const int mm = (tid >> 2) & 63, ppb = (tid & 3) * 8;
            const int m = m0 + mm;
            float v[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int pp = ppb + j;
                v[j] = (m < M && t_dyb[pp] >= 0)
                           ? ld_f32(dy + t_dyb[pp] + (long)m * OHW) : 0.f;
            }
            st8_lds(&a_lds[mm][ppb], v);
        }
        {
            const int kk = (tid >> 2) & 63, ppb = (tid & 3) * 8;
            const int k = k0 + kk;
            float v[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int pp = ppb + j;
                v[j] = 0.f;
                if (k < K) {
                    const int ih = t_ihb[pp] + t_kkh[kk];
                    const int iw = t_iwb[pp] + t_kkw[kk];
                    if (ih >= 0 && ih < gm.H && iw >= 0 && iw < gm.W)
                        v[j] = ld_f32(x + t_xb[pp] + (long)t_kcin[kk] * HW
                                      + ih * gm.W + iw);
                }
            }
            st8_lds(&b_lds[kk][ppb], v);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    float* slab = out + (long)sp * GK;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                const int k = k0 + wp + fp * 16 + (l & 15);
                if (m < M && k < K)
                    slab[(long)(g * gm.Cout + m) * K + k] = acc[fm][fp][r];
            }
}

__global__ void __launch_bounds__(256)
splitp_reduce_kernel(const float* __restrict__ partials,
                     float* __restrict__ dw, long GK, int splitp) {
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= GK) return;
    float s = 0.f;
    for (int sp = 0; sp < splitp; ++sp) s += partials[(long)sp * GK + i];
    dw[i] = s;
}


// ---------------------------------------------- staged fwd (im2col in LDS)
// When a BP-pixel tile lies inside one sample and spans whole output rows
// (OHW %% BP == 0 and BP %% OW == 0), the tile's input window is a dense
// (channels x rows x width) slab: stage it into LDS with coalesced loads
// once per k-step and build the implicit-GEMM B tile from LDS.  This
// replaces 16 scattered 2-byte global loads per thread per k-step (the
// measured bound: 65%% wave-wait, profiles/pmc_conv_fwd_r01.txt) with ~90
// coalesced transactions per block.
constexpr int WIN_CH = BK / 9 + 2;   // channels a BK k-slab can touch (3x3)
constexpr int WIN_ROWS = 16;         // (BP/OW - 1)*stride + khw  (<= 16)
constexpr int WIN_W = 40;            // padded input width (W + 2*pad <= 34)

template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_fwd_staged_kernel(const T* __restrict__ x, const float* __restrict__ w,
                       const float* __restrict__ bias,
                       const T* __restrict__ residual, T* __restrict__ y,
                       float* __restrict__ partial, ConvGeom gm, int splitk) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ float win[WIN_CH][WIN_ROWS][WIN_W];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int p0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int HW = gm.H * gm.W;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int nkc = ((K + BK - 1) / BK + splitk - 1) / splitk;
    const int ks = sp * nkc * BK;
    const int ke = min(K, ks + nkc * BK);
    // tile geometry: one sample, whole output rows
    const int n = p0 / OHW;
    const int oh0 = (p0 - n * OHW) / gm.OW;
    const int nrow_out = BP / gm.OW;
    const int rows_in = (nrow_out - 1) * gm.stride + gm.khw;
    const int ih0 = oh0 * gm.stride - gm.pad;   // window top (may be < 0)
    const long xn = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW;
    const int tile_p = min(BP, P - p0);
    // per-thread fragment ownership (as the gather kernel)
    const int mm_a = tid >> 2, kkb = (tid & 3) * 8;
    const int pp_b = tid >> 2;
    const int ow_b = (p0 + pp_b - n * OHW) - (oh0 + pp_b / gm.OW) * gm.OW;
    const int ohl_b = (pp_b / gm.OW) * gm.stride;   // window-row base
    const int iwl_b = ow_b * gm.stride;             // window-col base (pre-pad)
    float va[8];

    f32x4 acc[2][2] = {};
    for (int k0 = ks; k0 < ke; k0 += BK) {
        const int cin0 = k0 / kk2;
        const int cin1 = min(gm.Cin - 1, (k0 + BK - 1) / kk2);
        const int nch = cin1 - cin0 + 1;
        // stage the window slab (coalesced along the input width)
        const int wtot = nch * rows_in * gm.W;
        for (int e = tid; e < wtot; e += 256) {
            const int ww = e % gm.W;
            const int rr = (e / gm.W) % rows_in;
            const int cc = e / (gm.W * rows_in);
            const int ih = ih0 + rr;
            win[cc][rr][ww + gm.pad] =
                (ih >= 0 && ih < gm.H)
                    ? ld_f32(x + xn + (long)(cin0 + cc) * HW + ih * gm.W + ww)
                    : 0.f;
        }
        // zero the horizontal pad columns
        for (int e = tid; e < nch * rows_in * gm.pad * 2; e += 256) {
            const int side = e & 1;
            const int pe = e >> 1;
            const int pcol = pe % gm.pad;
            const int rr = (pe / gm.pad) % rows_in;
            const int cc = pe / (gm.pad * rows_in);
            win[cc][rr][side ? gm.pad + gm.W + pcol : pcol] = 0.f;
        }
        // A tile (weights, coalesced fp32)
        {
            const int m = m0 + mm_a;
            const float* wrow = w + (long)(g * gm.Cout + m) * K + k0 + kkb;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int k = k0 + kkb + j;
                va[j] = (m < M && k < K) ? wrow[j] : 0.f;
            }
            st8_lds(&a_lds[mm_a][kkb], va);
        }
        __syncthreads();
        // B tile built from the LDS window
        {
            float vb[8];
            int k = k0 + kkb;
            int cin = k / kk2, r = k - cin * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                vb[j] = (k + j < K && pp_b < tile_p)
                            ? win[cin - cin0][ohl_b + kh][iwl_b + kw]
                            : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cin;
                    }
                }
            }
            st8_lds(&b_lds[pp_b][kkb], vb);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cout * OHW;
    const long yb0 = ((long)n * gm.G * gm.Cout + (long)g * gm.Cout) * OHW
                     + (p0 - n * OHW);
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int pp = wp + fp * 16 + (l & 15);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (m < M && pp < tile_p) {
                    const long off = yb0 + pp + (long)m * OHW;
                    float v = acc[fm][fp][r];
                    if (splitk == 1) {
                        if (bias) v += bias[g * gm.Cout + m];
                        if (residual) v += ld_f32(residual + off);
                        st_f32(y + off, v);
                    } else {
                        partial[slab + off] = v;
                    }
                }
            }
        }
}


// -------------------------- staged fwd, multi-sample (small output planes)
// L4-family shapes have OH*OW < BP (e.g. 4x4 = 16 pixels), so a BP tile
// spans ns = BP/OHW WHOLE samples: stage each sample's full (padded) input
// plane for the k-slab's channel span and build the B tile from LDS — the
// same im2col-in-LDS idea as conv_fwd_staged_kernel, which requires
// OHW %% BP == 0 and therefore left these shapes on the scattered-gather
// kernel (1.9s of kstats_r02).
constexpr int MS_ROWS = 40;          // ns * ((OH-1)*stride + khw)

template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_fwd_staged_ms_kernel(const T* __restrict__ x,
                          const float* __restrict__ w,
                          const float* __restrict__ bias,
                          const T* __restrict__ residual, T* __restrict__ y,
                          float* __restrict__ partial, ConvGeom gm,
                          int splitk) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ float win[WIN_CH][MS_ROWS][WIN_W];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int p0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int HW = gm.H * gm.W;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int nkc = ((K + BK - 1) / BK + splitk - 1) / splitk;
    const int ks = sp * nkc * BK;
    const int ke = min(K, ks + nkc * BK);
    // tile geometry: ns whole samples, full output planes
    const int ns = BP / OHW;
    const int n0 = p0 / OHW;
    const int rows_in = (gm.OH - 1) * gm.stride + gm.khw;
    const long gch = ((long)g * gm.Cin) * HW;
    const long sampstride = (long)gm.G * gm.Cin * HW;
    const int tile_p = min(BP, P - p0);
    const int mm_a = tid >> 2, kkb = (tid & 3) * 8;
    const int pp_b = tid >> 2;
    const int ls_b = pp_b / OHW;
    const int pix_b = pp_b - ls_b * OHW;
    const int ohl_b = (pix_b / gm.OW) * gm.stride;
    const int iwl_b = (pix_b - (pix_b / gm.OW) * gm.OW) * gm.stride;
    const int rowb_b = ls_b * rows_in;
    float va[8];

    f32x4 acc[2][2] = {};
    for (int k0 = ks; k0 < ke; k0 += BK) {
        const int cin0 = k0 / kk2;
        const int cin1 = min(gm.Cin - 1, (k0 + BK - 1) / kk2);
        const int nch = cin1 - cin0 + 1;
        // stage ns sample planes (coalesced along the input width)
        const int wtot = nch * ns * rows_in * gm.W;
        for (int e = tid; e < wtot; e += 256) {
            const int ww = e % gm.W;
            const int rr = (e / gm.W) % rows_in;
            const int lsl = (e / (gm.W * rows_in)) % ns;
            const int cc = e / (gm.W * rows_in * ns);
            const int ih = rr - gm.pad;
            const int n = n0 + lsl;
            win[cc][lsl * rows_in + rr][ww + gm.pad] =
                (ih >= 0 && ih < gm.H && n < gm.N)
                    ? ld_f32(x + (long)n * sampstride + gch
                             + (long)(cin0 + cc) * HW + ih * gm.W + ww)
                    : 0.f;
        }
        for (int e = tid; e < nch * ns * rows_in * gm.pad * 2; e += 256) {
            const int side = e & 1;
            const int pe = e >> 1;
            const int pcol = pe % gm.pad;
            const int rr = (pe / gm.pad) % (ns * rows_in);
            const int cc = pe / (gm.pad * ns * rows_in);
            win[cc][rr][side ? gm.pad + gm.W + pcol : pcol] = 0.f;
        }
        {   // A tile (weights, coalesced fp32)
            const int m = m0 + mm_a;
            const float* wrow = w + (long)(g * gm.Cout + m) * K + k0 + kkb;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int k = k0 + kkb + j;
                va[j] = (m < M && k < K) ? wrow[j] : 0.f;
            }
            st8_lds(&a_lds[mm_a][kkb], va);
        }
        __syncthreads();
        {   // B tile from the LDS windows
            float vb[8];
            int k = k0 + kkb;
            int cin = k / kk2, r = k - cin * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                vb[j] = (k + j < K && pp_b < tile_p)
                            ? win[cin - cin0][rowb_b + ohl_b + kh]
                                 [iwl_b + kw]
                            : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cin;
                    }
                }
            }
            st8_lds(&b_lds[pp_b][kkb], vb);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cout * OHW;
    const long ystride = (long)gm.G * gm.Cout * OHW;
    const long ygch = (long)g * gm.Cout * OHW;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int pp = wp + fp * 16 + (l & 15);
            const int lsl = pp / OHW;
            const int pix = pp - lsl * OHW;
            const long yb = (long)(n0 + lsl) * ystride + ygch + pix;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (m < M && pp < tile_p) {
                    const long off = yb + (long)m * OHW;
                    float v = acc[fm][fp][r];
                    if (splitk == 1) {
                        if (bias) v += bias[g * gm.Cout + m];
                        if (residual) v += ld_f32(residual + off);
                        st_f32(y + off, v);
                    } else {
                        partial[slab + off] = v;
                    }
                }
            }
        }
}

// ----------------------------------- staged stride-1 bwd-data (LDS window)
// Same idea as conv_fwd_staged_kernel: a BP-pixel dx tile within one
// sample spanning whole input rows has a dense dy window; stage it
// coalesced and build the transposed-conv patch matrix from LDS.
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_data_staged_kernel(const T* __restrict__ dy,
                            const float* __restrict__ w, T* __restrict__ dx,
                            float* __restrict__ partial, ConvGeom gm,
                            int splitk) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ float win[WIN_CH][WIN_ROWS][WIN_W];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int c0 = blockIdx.x * BM;
    const int q0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int J = gm.Cout * kk2;
    const int K = gm.Cin * kk2;
    const int HW = gm.H * gm.W;
    const int OHW = gm.OH * gm.OW;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int njc = ((J + BK - 1) / BK + splitk - 1) / splitk;
    const int js = sp * njc * BK;
    const int je = min(J, js + njc * BK);
    // tile geometry (stride 1: OH=H, OW=W)
    const int n = q0 / HW;
    const int ih0 = (q0 - n * HW) / gm.W;
    const int nrow_in = BP / gm.W;
    const int rows_w = nrow_in + gm.khw - 1;
    const int ohs_min = ih0 + gm.pad - (gm.khw - 1);
    const long dyn = ((long)n * gm.G * gm.Cout + (long)g * gm.Cout) * OHW;
    const int cc_a = tid >> 2, jjb = (tid & 3) * 8;
    // this thread's pixel for the B build
    const int iwl = (q0 + cc_a - n * HW) - (ih0 + cc_a / gm.W) * gm.W;
    const int rowl = (cc_a / gm.W) + (gm.khw - 1);   // ih - ih0 + (khw-1)
    float va[8];

    f32x4 acc[2][2] = {};
    for (int j0 = js; j0 < je; j0 += BK) {
        const int co0 = j0 / kk2;
        const int co1 = min(gm.Cout - 1, (j0 + BK - 1) / kk2);
        const int nch = co1 - co0 + 1;
        const int wtot = nch * rows_w * gm.OW;
        for (int e = tid; e < wtot; e += 256) {
            const int ww = e % gm.OW;
            const int rr = (e / gm.OW) % rows_w;
            const int cc = e / (gm.OW * rows_w);
            const int ohs = ohs_min + rr;
            win[cc][rr][ww + gm.pad] =
                (ohs >= 0 && ohs < gm.OH)
                    ? ld_f32(dy + dyn + (long)(co0 + cc) * OHW
                             + ohs * gm.OW + ww)
                    : 0.f;
        }
        for (int e = tid; e < nch * rows_w * gm.pad * 2; e += 256) {
            const int side = e & 1;
            const int pe = e >> 1;
            const int pcol = pe % gm.pad;
            const int rr = (pe / gm.pad) % rows_w;
            const int cc = pe / (gm.pad * rows_w);
            win[cc][rr][side ? gm.pad + gm.OW + pcol : pcol] = 0.f;
        }
        {   // A tile: w[g*Cout + cout][c*kk2 + r] rows=c
            const int c = c0 + cc_a;
            int j = j0 + jjb;
            int cout = j / kk2, r = j - cout * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j8 = 0; j8 < 8; ++j8) {
                va[j8] = (c < gm.Cin && j + j8 < J)
                             ? w[(long)(g * gm.Cout + cout) * K + c * kk2
                                 + kh * gm.khw + kw]
                             : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cout;
                    }
                }
            }
            st8_lds(&a_lds[cc_a][jjb], va);
        }
        __syncthreads();
        {   // B tile from window: row = rowl - kh, col = iwl + 2*pad? no:
            // needed dy[ohs = ih + pad - kh][ows = iw + pad - kw];
            // win row = ohs - ohs_min = rowl - kh; win col = ows + pad
            //         = iwl + 2*pad - kw... stored col = ows + pad where
            // ows = iw + pad - kw -> col = iw + 2*pad - kw
            float vb[8];
            int j = j0 + jjb;
            int cout = j / kk2, r = j - cout * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j8 = 0; j8 < 8; ++j8) {
                vb[j8] = (j + j8 < J)
                             ? win[cout - co0][rowl - kh]
                                  [iwl + 2 * gm.pad - kw]
                             : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cout;
                    }
                }
            }
            st8_lds(&b_lds[cc_a][jjb], vb);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cin * HW;
    const long xb0 = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW
                     + (q0 - n * HW);
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int qq = wp + fp * 16 + (l & 15);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int c = c0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (c < gm.Cin) {
                    const long off = xb0 + qq + (long)c * HW;
                    if (splitk == 1)
                        st_f32(dx + off, acc[fm][fp][r]);
                    else
                        partial[slab + off] = acc[fm][fp][r];
                }
            }
        }
}


// ------------------- staged stride-1 bwd-data, multi-sample (small planes)
// Same extension as conv_fwd_staged_ms_kernel: a BP-pixel dx tile spanning
// BP/HW whole samples stages each sample's full padded dy plane per J-step.
template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_data_staged_ms_kernel(const T* __restrict__ dy,
                               const float* __restrict__ w,
                               T* __restrict__ dx,
                               float* __restrict__ partial, ConvGeom gm,
                               int splitk) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ float win[WIN_CH][MS_ROWS][WIN_W];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int c0 = blockIdx.x * BM;
    const int q0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int J = gm.Cout * kk2;
    const int K = gm.Cin * kk2;
    const int HW = gm.H * gm.W;
    const int OHW = gm.OH * gm.OW;
    const int Q = gm.N * HW;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int njc = ((J + BK - 1) / BK + splitk - 1) / splitk;
    const int js = sp * njc * BK;
    const int je = min(J, js + njc * BK);
    const int ns = BP / HW;
    const int n0 = q0 / HW;
    // dy rows needed for a full dx plane: ohs = ih + pad - kh over ih in
    // [0, H) -> rows [pad - (khw-1), H - 1 + pad]; stage rows_w rows from
    // ohs_min with zero-fill outside [0, OH)
    const int rows_w = gm.H + gm.khw - 1;
    const int ohs_min = gm.pad - (gm.khw - 1);
    const long dgch = (long)g * gm.Cout * OHW;
    const long dystride = (long)gm.G * gm.Cout * OHW;
    const int tile_q = min(BP, Q - q0);
    const int cc_a = tid >> 2, jjb = (tid & 3) * 8;
    const int ls_b = cc_a / HW;
    const int pix_b = cc_a - ls_b * HW;
    const int iwl = pix_b - (pix_b / gm.W) * gm.W;
    const int rowl = (pix_b / gm.W) + (gm.khw - 1);
    const int rowb = ls_b * rows_w;
    float va[8];

    f32x4 acc[2][2] = {};
    for (int j0 = js; j0 < je; j0 += BK) {
        const int co0 = j0 / kk2;
        const int co1 = min(gm.Cout - 1, (j0 + BK - 1) / kk2);
        const int nch = co1 - co0 + 1;
        const int wtot = nch * ns * rows_w * gm.OW;
        for (int e = tid; e < wtot; e += 256) {
            const int ww = e % gm.OW;
            const int rr = (e / gm.OW) % rows_w;
            const int lsl = (e / (gm.OW * rows_w)) % ns;
            const int cc = e / (gm.OW * rows_w * ns);
            const int ohs = ohs_min + rr;
            const int n = n0 + lsl;
            win[cc][lsl * rows_w + rr][ww + gm.pad] =
                (ohs >= 0 && ohs < gm.OH && n < gm.N)
                    ? ld_f32(dy + (long)n * dystride + dgch
                             + (long)(co0 + cc) * OHW + ohs * gm.OW + ww)
                    : 0.f;
        }
        for (int e = tid; e < nch * ns * rows_w * gm.pad * 2; e += 256) {
            const int side = e & 1;
            const int pe = e >> 1;
            const int pcol = pe % gm.pad;
            const int rr = (pe / gm.pad) % (ns * rows_w);
            const int cc = pe / (gm.pad * ns * rows_w);
            win[cc][rr][side ? gm.pad + gm.OW + pcol : pcol] = 0.f;
        }
        {   // A tile: w rows = input channel
            const int c = c0 + cc_a;
            int j = j0 + jjb;
            int cout = j / kk2, r = j - cout * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j8 = 0; j8 < 8; ++j8) {
                va[j8] = (c < gm.Cin && j + j8 < J)
                             ? w[(long)(g * gm.Cout + cout) * K + c * kk2
                                 + kh * gm.khw + kw]
                             : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cout;
                    }
                }
            }
            st8_lds(&a_lds[cc_a][jjb], va);
        }
        __syncthreads();
        {   // B tile from the windows
            float vb[8];
            int j = j0 + jjb;
            int cout = j / kk2, r = j - cout * kk2;
            int kh = r / gm.khw, kw = r - kh * gm.khw;
#pragma unroll
            for (int j8 = 0; j8 < 8; ++j8) {
                vb[j8] = (j + j8 < J && cc_a < tile_q)
                             ? win[cout - co0][rowb + rowl - kh]
                                  [iwl + 2 * gm.pad - kw]
                             : 0.f;
                if (++kw == gm.khw) {
                    kw = 0;
                    if (++kh == gm.khw) {
                        kh = 0;
                        ++cout;
                    }
                }
            }
            st8_lds(&b_lds[cc_a][jjb], vb);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
        __syncthreads();
    }
    const long slab = (long)sp * gm.N * gm.G * gm.Cin * HW;
    const long xstride = (long)gm.G * gm.Cin * HW;
    const long xgch = (long)g * gm.Cin * HW;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp) {
            const int qq = wp + fp * 16 + (l & 15);
            const int lsl = qq / HW;
            const int pix = qq - lsl * HW;
            const long xb = (long)(n0 + lsl) * xstride + xgch + pix;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int c = c0 + wm + fm * 16 + (l >> 4) * 4 + r;
                if (c < gm.Cin && qq < tile_q) {
                    const long off = xb + (long)c * HW;
                    if (splitk == 1)
                        st_f32(dx + off, acc[fm][fp][r]);
                    else
                        partial[slab + off] = acc[fm][fp][r];
                }
            }
        }
}

// -------------------------------------- staged bwd-weight (x window in LDS)
// The k-tile (hence the input-channel span) is FIXED per block, so the x
// window only advances rows as the P loop walks output rows: stage the
// dense (channels x rows x width) slab per p-step and build the patch
// tile from LDS.  Channels per 64-wide k-tile: ceil(64/9)+1 <= 9.
constexpr int WWIN_CH = 9;

template <typename T, typename TA, typename TB>
__global__ void __launch_bounds__(256)
conv_bwd_weight_staged_kernel(const T* __restrict__ dy,
                              const T* __restrict__ x,
                              float* __restrict__ out, ConvGeom gm,
                              int splitp) {
    __shared__ TA a_lds[BM][LDK];
    __shared__ TB b_lds[BP][LDK];
    __shared__ float win[WWIN_CH][WIN_ROWS][WIN_W];
    __shared__ long t_dyb[BK];
    const int g = blockIdx.z % gm.G;
    const int sp = blockIdx.z / gm.G;
    const int m0 = blockIdx.x * BM;
    const int k0 = blockIdx.y * BP;
    const int kk2 = gm.khw * gm.khw;
    const int K = gm.Cin * kk2;
    const int M = gm.Cout;
    const int OHW = gm.OH * gm.OW;
    const int P = gm.N * OHW;
    const int HW = gm.H * gm.W;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    const int pchunk0 = (P + splitp - 1) / splitp;
    const int pchunk = ((pchunk0 + BK - 1) / BK) * BK;  // BK-aligned chunks
    const int pstart = sp * pchunk;
    const int pend = min(P, pstart + pchunk);
    const int cin0 = k0 / kk2;
    const int nch = min(gm.Cin - 1, (k0 + BP - 1) / kk2) - cin0 + 1;
    const int nrow_out = BK / gm.OW;          // 32 pixels per p-step
    const int rows_in = (nrow_out - 1) * gm.stride + gm.khw;
    // per-thread B ownership: fixed k, 8 consecutive pixels
    const int kk_b = (tid >> 2) & 63, ppb = (tid & 3) * 8;
    const int k_b = k0 + kk_b;
    int cin_b = 0, kh_b = 0, kw_b = 0;
    if (k_b < K) {
        cin_b = k_b / kk2;
        const int r = k_b - cin_b * kk2;
        kh_b = r / gm.khw;
        kw_b = r - kh_b * gm.khw;
    }
    const int mm_a = tid >> 2;   // A rows (dy), 8 consecutive p via t_dyb

    f32x4 acc[2][2] = {};
    for (int p0 = pstart; p0 < pend; p0 += BK) {
        __syncthreads();
        const int n = p0 / OHW;
        const int oh0 = (p0 - n * OHW) / gm.OW;
        const int ih0 = oh0 * gm.stride - gm.pad;
        const long xn = ((long)n * gm.G * gm.Cin + (long)g * gm.Cin) * HW;
        if (tid < BK) {
            const int p = p0 + tid;
            t_dyb[tid] = (p < pend && p < P)
                             ? ((long)n * gm.G * gm.Cout
                                + (long)g * gm.Cout) * OHW + (p - n * OHW)
                             : -1;
        }
        const int wtot = nch * rows_in * gm.W;
        for (int e = tid; e < wtot; e += 256) {
            const int ww = e % gm.W;
            const int rr = (e / gm.W) % rows_in;
            const int cc = e / (gm.W * rows_in);
            const int ih = ih0 + rr;
            win[cc][rr][ww + gm.pad] =
                (ih >= 0 && ih < gm.H)
                    ? ld_f32(x + xn + (long)(cin0 + cc) * HW + ih * gm.W + ww)
                    : 0.f;
        }
        for (int e = tid; e < nch * rows_in * gm.pad * 2; e += 256) {
            const int side = e & 1;
            const int pe = e >> 1;
            const int pcol = pe % gm.pad;
            const int rr = (pe / gm.pad) % rows_in;
            const int cc = pe / (gm.pad * rows_in);
            win[cc][rr][side ? gm.pad + gm.W + pcol : pcol] = 0.f;
        }
        __syncthreads();
        {   // A tile: dy rows (coalesced via t_dyb)
            const int m = m0 + mm_a;
            float v[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int pp = ppb + j;
                v[j] = (m < M && t_dyb[pp] >= 0)
                           ? ld_f32(dy + t_dyb[pp] + (long)m * OHW) : 0.f;
            }
            st8_lds(&a_lds[mm_a][ppb], v);
        }
        {   // B tile from the window: 8 consecutive pixels of one k
            float v[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int pp = ppb + j;
                v[j] = 0.f;
                if (k_b < K && p0 + pp < pend) {
                    const int ohl = (pp / gm.OW) * gm.stride;
                    const int owp = pp - (pp / gm.OW) * gm.OW;
                    v[j] = win[cin_b - cin0][ohl + kh_b]
                              [owp * gm.stride + kw_b];
                }
            }
            st8_lds(&b_lds[kk_b][ppb], v);
        }
        __syncthreads();
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_tile2<TA, TB>(
                    &a_lds[wm + fm * 16 + (l & 15)][0],
                    &b_lds[wp + fp * 16 + (l & 15)][0], acc[fm][fp]);
    }
    __syncthreads();
    const long GK = (long)gm.G * gm.Cout * K;
    float* slab = out + (long)sp * GK;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = m0 + wm + fm * 16 + (l >> 4) * 4 + r;
                const int k = k0 + wp + fp * 16 + (l & 15);
                if (m < M && k < K)
                    slab[(long)(g * gm.Cout + m) * K + k] = acc[fm][fp][r];
            }
}

// ------------------------------------------------------------ host layer
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <cstdlib>

// target concurrent-block count for the split-K/split-P decompositions
// (tunable: HETEROFL_CONV_SPLIT_TARGET, default 1024 = 4 blocks/CU)
static int split_target() {
    static int t = [] {
        const char* e = std::getenv("HETEROFL_CONV_SPLIT_TARGET");
        return e ? std::atoi(e) : 1024;
    }();
    return t;
}

#define DISPATCH_CONV_FT(t, ...)                                              \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

at::Tensor conv_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                    at::Tensor residual, int64_t groups, int64_t stride,
                    int64_t pad, int64_t fp8) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
    TORCH_CHECK(w.scalar_type() == at::kFloat, "weights must be fp32 master");
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
    auto y = at::empty({gm.N, gm.G * gm.Cout, gm.OH, gm.OW}, x.options());
    const int M = gm.Cout, P = gm.N * gm.OH * gm.OW;
    const int K = gm.Cin * gm.khw * gm.khw;
    const int kchunks = (K + BK - 1) / BK;
    const int tiles = ((M + BM - 1) / BM) * ((P + BP - 1) / BP) * gm.G;
    int splitk = 1;
    // same rule as bwdW (measured, L3): short reductions with enough tiles
    // run best unsplit — the split's partial-write + reduce traffic beats
    // the extra parallelism (HETEROFL_SPLIT_NOSKIP=1 restores the old
    // always-split heuristic for A/B)
    static const bool noskip_f = [] {
        const char* e = std::getenv("HETEROFL_SPLIT_NOSKIP");
        return e && e[0] == '1';
    }();
    if (noskip_f || K > 1024 || tiles < 512)
        while (tiles * splitk < split_target() && splitk * 2 <= kchunks / 2)
            splitk *= 2;
    dim3 grid((M + BM - 1) / BM, (P + BP - 1) / BP, gm.G * splitk);
    auto stream = at::hip::getCurrentHIPStream();
    at::Tensor partial;
    if (splitk > 1)
        partial = at::empty({(long)splitk * y.numel()},
                            x.options().dtype(at::kFloat));
    const int OHW = gm.OH * gm.OW;
    const bool staged = (P % BP == 0 || gm.N * gm.OH * gm.OW >= BP)
                        && (gm.OH * gm.OW) % BP == 0 && BP % gm.OW == 0
                        && (gm.W + 2 * gm.pad) <= WIN_W - 2
                        && ((BP / gm.OW - 1) * gm.stride + gm.khw) <= WIN_ROWS
                        && (BK / (gm.khw * gm.khw) + 2) <= WIN_CH
                        && std::getenv("HETEROFL_CONV_NO_STAGED") == nullptr;
    // small output planes: a BP tile spans BP/OHW whole samples.
    // MEASURED NEGATIVE at the training shapes (convbench r02c: L4 fwd
    // 107->124us, L4d 67->101us — the 32 KB window slab costs more
    // occupancy than the gather kernel loses to scattered loads); kept
    // opt-in for future large-batch study
    static const bool ms_on = [] {
        const char* e = std::getenv("HETEROFL_CONV_MS");
        return e && e[0] == '1';
    }();
    const bool staged_ms = ms_on && !staged && OHW < BP && BP % OHW == 0
                          && (gm.W + 2 * gm.pad) <= WIN_W - 2
                          && (BP / OHW) * ((gm.OH - 1) * gm.stride + gm.khw)
                                 <= MS_ROWS
                          && (BK / (gm.khw * gm.khw) + 2) <= WIN_CH
                          && std::getenv("HETEROFL_CONV_NO_STAGED") == nullptr;
    DISPATCH_CONV_FT(x.scalar_type(), {
        if (staged_ms && !fp8) {
            hipLaunchKernelGGL((conv_fwd_staged_ms_kernel<scalar_t, scalar_t,
                                                          scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               w.data_ptr<float>(),
                               bias.defined() ? bias.data_ptr<float>()
                                              : nullptr,
                               residual.defined()
                                   ? (const scalar_t*)residual.data_ptr()
                                   : nullptr,
                               (scalar_t*)y.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        } else if (staged && !fp8) {
            hipLaunchKernelGGL((conv_fwd_staged_kernel<scalar_t, scalar_t,
                                                       scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               w.data_ptr<float>(),
                               bias.defined() ? bias.data_ptr<float>()
                                              : nullptr,
                               residual.defined()
                                   ? (const scalar_t*)residual.data_ptr()
                                   : nullptr,
                               (scalar_t*)y.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        } else if (fp8 && x.scalar_type() == at::kBFloat16)
            hipLaunchKernelGGL((conv_fwd_kernel<scalar_t, e4m3, e4m3>), grid,
                               dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               w.data_ptr<float>(),
                               bias.defined() ? bias.data_ptr<float>()
                                              : nullptr,
                               residual.defined()
                                   ? (const scalar_t*)residual.data_ptr()
                                   : nullptr,
                               (scalar_t*)y.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        else
            hipLaunchKernelGGL((conv_fwd_kernel<scalar_t, scalar_t, scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)x.data_ptr(),
                               w.data_ptr<float>(),
                               bias.defined() ? bias.data_ptr<float>()
                                              : nullptr,
                               residual.defined()
                                   ? (const scalar_t*)residual.data_ptr()
                                   : nullptr,
                               (scalar_t*)y.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        if (splitk > 1) {
            const long total = y.numel();
            const int blocks = (int)((total + 255) / 256);
            hipLaunchKernelGGL(conv_out_reduce_kernel<scalar_t>, dim3(blocks),
                               dim3(256), 0, stream,
                               partial.data_ptr<float>(),
                               bias.defined() ? bias.data_ptr<float>()
                                              : nullptr,
                               residual.defined()
                                   ? (const scalar_t*)residual.data_ptr()
                                   : nullptr,
                               (scalar_t*)y.data_ptr(), total,
                               (long)gm.OH * gm.OW, gm.G * gm.Cout, splitk);
        }
    });
    return y;
}

at::Tensor conv_bwd_data(at::Tensor dy, at::Tensor w, int64_t groups,
                         int64_t stride, int64_t pad, int64_t H, int64_t W,
                         int64_t fp8) {
    TORCH_CHECK(dy.is_cuda() && w.is_contiguous());
    TORCH_CHECK(stride == 1 || stride == 2, "stride must be 1 or 2");
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
    const bool s2 = (gm.stride == 2 && gm.H % 2 == 0 && gm.W % 2 == 0);
    // parity path: 4 classes of Q/4 pixels, ~1/4 the taps each
    const int Jfull = gm.Cout * gm.khw * gm.khw;
    const int J = s2 ? (Jfull + 3) / 4 : Jfull;
    const int jchunks = (J + BK - 1) / BK;
    const int qtile = s2 ? Q / 4 : Q;
    const int nyp = (qtile + BP - 1) / BP;
    const int tiles = ((gm.Cin + BM - 1) / BM) * nyp * (s2 ? 4 : 1) * gm.G;
    int splitk = 1;
    static const bool noskip_d = [] {
        const char* e = std::getenv("HETEROFL_SPLIT_NOSKIP");
        return e && e[0] == '1';
    }();
    if (noskip_d || J > 1024 || tiles < 512)
        while (tiles * splitk < split_target() && splitk * 2 <= jchunks / 2)
            splitk *= 2;
    dim3 grid((gm.Cin + BM - 1) / BM, nyp * (s2 ? 4 : 1), gm.G * splitk);
    auto stream = at::hip::getCurrentHIPStream();
    at::Tensor partial;
    if (splitk > 1)
        partial = at::empty({(long)splitk * dx.numel()},
                            dy.options().dtype(at::kFloat));
    const bool stg = !s2 && gm.stride == 1
                     && (gm.H * gm.W) % BP == 0 && BP % gm.W == 0
                     && (gm.OW + 2 * gm.pad) <= WIN_W - 2
                     && (BP / gm.W + gm.khw - 1) <= WIN_ROWS
                     && (BK / (gm.khw * gm.khw) + 2) <= WIN_CH
                     && std::getenv("HETEROFL_CONV_NO_STAGED") == nullptr;
    const int HWd = gm.H * gm.W;
    static const bool ms_on_d = [] {
        const char* e = std::getenv("HETEROFL_CONV_MS");
        return e && e[0] == '1';
    }();
    const bool stg_ms = ms_on_d && !stg && !s2 && gm.stride == 1
                        && HWd < BP && BP % HWd == 0
                        && (gm.OW + 2 * gm.pad) <= WIN_W - 2
                        && (BP / HWd) * (gm.H + gm.khw - 1) <= MS_ROWS
                        && (BK / (gm.khw * gm.khw) + 2) <= WIN_CH
                        && std::getenv("HETEROFL_CONV_NO_STAGED") == nullptr;
    DISPATCH_CONV_FT(dy.scalar_type(), {
        const bool q = fp8 && dy.scalar_type() == at::kBFloat16;
        if (stg_ms && !q) {
            hipLaunchKernelGGL((conv_bwd_data_staged_ms_kernel<scalar_t,
                                                               scalar_t,
                                                               scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               w.data_ptr<float>(), (scalar_t*)dx.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        } else if (stg && !q) {
            hipLaunchKernelGGL((conv_bwd_data_staged_kernel<scalar_t,
                                                            scalar_t,
                                                            scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               w.data_ptr<float>(), (scalar_t*)dx.data_ptr(),
                               splitk > 1 ? partial.data_ptr<float>()
                                          : nullptr,
                               gm, splitk);
        } else if (s2) {
            if (q)
                hipLaunchKernelGGL((conv_bwd_data_s2_kernel<scalar_t, e4m3,
                                                            e5m2>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   w.data_ptr<float>(),
                                   (scalar_t*)dx.data_ptr(),
                                   splitk > 1 ? partial.data_ptr<float>()
                                              : nullptr,
                                   gm, splitk, nyp);
            else
                hipLaunchKernelGGL((conv_bwd_data_s2_kernel<scalar_t, scalar_t,
                                                            scalar_t>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   w.data_ptr<float>(),
                                   (scalar_t*)dx.data_ptr(),
                                   splitk > 1 ? partial.data_ptr<float>()
                                              : nullptr,
                                   gm, splitk, nyp);
        } else {
            if (q)
                hipLaunchKernelGGL((conv_bwd_data_kernel<scalar_t, e4m3,
                                                         e5m2>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   w.data_ptr<float>(),
                                   (scalar_t*)dx.data_ptr(),
                                   splitk > 1 ? partial.data_ptr<float>()
                                              : nullptr,
                                   gm, splitk);
            else
                hipLaunchKernelGGL((conv_bwd_data_kernel<scalar_t, scalar_t,
                                                         scalar_t>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   w.data_ptr<float>(),
                                   (scalar_t*)dx.data_ptr(),
                                   splitk > 1 ? partial.data_ptr<float>()
                                              : nullptr,
                                   gm, splitk);
        }
        if (splitk > 1) {
            const long total = dx.numel();
            const int blocks = (int)((total + 255) / 256);
            hipLaunchKernelGGL(conv_out_reduce_kernel<scalar_t>, dim3(blocks),
                               dim3(256), 0, stream,
                               partial.data_ptr<float>(), nullptr, nullptr,
                               (scalar_t*)dx.data_ptr(), total, 1L, 1,
                               splitk);
        }
    });
    return dx;
}

at::Tensor conv_bwd_weight(at::Tensor dy, at::Tensor x, int64_t groups,
                           int64_t stride, int64_t pad, int64_t khw,
                           int64_t fp8) {
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
    const long GK = (long)gm.G * gm.Cout * K;
    const int mk_tiles = ((gm.Cout + BM - 1) / BM) * ((K + BP - 1) / BP)
                         * gm.G;
    int splitp = 1;
    // measured (bwdw2 r2): small-P shapes that already have >=512 (m,k)
    // tiles need no split at all — L3 bwdW 89.4us (splitp 2) -> 74.5us
    // (splitp 1) = 1.00x MIOpen; splitting only pays when tiles are scarce
    // or P is long enough to amortize the partial reduce
    if (P > 1024 || mk_tiles < 512)
        while (mk_tiles * splitp < split_target() && splitp * BK * 4 < P)
            splitp *= 2;
    // measured NEGATIVE (bwdw_sweep r2): pushing split-P further on small-P
    // shapes (L3: splitp 8 -> 95.6us vs 89.4us at splitp 2) trades too much
    // partial-reduce traffic for the extra blocks; default off, kept gated
    // for future study (HETEROFL_BWDW_SMALLP=1)
    static const bool smallp_boost = [] {
        const char* e = std::getenv("HETEROFL_BWDW_SMALLP");
        return e && e[0] == '1';
    }();
    if (smallp_boost)
        while (P <= 2048 && mk_tiles * splitp < 4 * split_target()
               && splitp * BK < P)
            splitp *= 2;
    auto dw = at::empty({(long)gm.G * gm.Cout, gm.Cin, gm.khw, gm.khw},
                        x.options().dtype(at::kFloat));
    auto stream = at::hip::getCurrentHIPStream();
    dim3 grid((gm.Cout + BM - 1) / BM, (K + BP - 1) / BP, gm.G * splitp);
    const bool q = fp8 && x.scalar_type() == at::kBFloat16;
    // staged path: 3x3 kernels (a 64-wide k-tile then spans <= 9 input
    // channels = WWIN_CH), p-steps inside one sample spanning whole rows
    const bool stg = gm.khw == 3 && (gm.OH * gm.OW) % BK == 0
                     && BK % gm.OW == 0
                     && (gm.W + 2 * gm.pad) <= WIN_W - 2
                     && ((BK / gm.OW - 1) * gm.stride + gm.khw) <= WIN_ROWS
                     && std::getenv("HETEROFL_CONV_NO_STAGED") == nullptr;
    if (splitp == 1) {
        DISPATCH_CONV_FT(x.scalar_type(), {
            if (q)
                hipLaunchKernelGGL((conv_bwd_weight_kernel<scalar_t, e5m2,
                                                           e4m3>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   (const scalar_t*)x.data_ptr(),
                                   dw.data_ptr<float>(), gm, 1);
            else if (stg)
                hipLaunchKernelGGL((conv_bwd_weight_staged_kernel<scalar_t,
                                                                  scalar_t,
                                                                  scalar_t>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   (const scalar_t*)x.data_ptr(),
                                   dw.data_ptr<float>(), gm, 1);
            else
                hipLaunchKernelGGL((conv_bwd_weight_kernel<scalar_t, scalar_t,
                                                           scalar_t>),
                                   grid, dim3(256), 0, stream,
                                   (const scalar_t*)dyc.data_ptr(),
                                   (const scalar_t*)x.data_ptr(),
                                   dw.data_ptr<float>(), gm, 1);
        });
        return dw;
    }
    auto partials = at::empty({splitp, GK}, x.options().dtype(at::kFloat));
    DISPATCH_CONV_FT(x.scalar_type(), {
        if (q)
            hipLaunchKernelGGL((conv_bwd_weight_kernel<scalar_t, e5m2, e4m3>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               (const scalar_t*)x.data_ptr(),
                               partials.data_ptr<float>(), gm, splitp);
        else if (stg)
            hipLaunchKernelGGL((conv_bwd_weight_staged_kernel<scalar_t,
                                                              scalar_t,
                                                              scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               (const scalar_t*)x.data_ptr(),
                               partials.data_ptr<float>(), gm, splitp);
        else
            hipLaunchKernelGGL((conv_bwd_weight_kernel<scalar_t, scalar_t,
                                                       scalar_t>),
                               grid, dim3(256), 0, stream,
                               (const scalar_t*)dyc.data_ptr(),
                               (const scalar_t*)x.data_ptr(),
                               partials.data_ptr<float>(), gm, splitp);
    });
    const int threads = 256;
    const int blocks = (int)((GK + threads - 1) / threads);
    hipLaunchKernelGGL(splitp_reduce_kernel, dim3(blocks), dim3(threads), 0,
                       stream, partials.data_ptr<float>(),
                       dw.data_ptr<float>(), GK, splitp);
    return dw;
}

// --------------------------------------------------- MFMA layout self-test
__global__ void mfma_probe_kernel(const float* __restrict__ A,
                                  const float* __restrict__ B,
                                  float* __restrict__ D) {
    const int l = threadIdx.x;
    const int kb = (l >> 4) * 8;
    bf16x8 a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        a[j] = (__bf16)A[(l & 15) * 32 + kb + j];
        b[j] = (__bf16)B[(kb + j) * 16 + (l & 15)];
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
