// Fused multi-head attention for the HeteroFL masked-LM transformer (K8 of
// SURVEY.md §2b; reference: src/models/transformer.py:46-56 ScaledDotProduct
// — QK^T/temp -> softmax -> PV).  HeteroFL's sequence length is tiny
// (bptt = 64, head_dim <= 32, reference src/utils.py:201), so one workgroup
// computes a whole (batch*head) attention in one launch: Q,K staged in LDS,
// QK^T on MFMA (the 64x64x32 tile is exactly the conv tile shape), row
// softmax, P·V on MFMA.  Replaces ~8 rocBLAS/elementwise launches per
// attention with 1 forward + 1 backward kernel — the LM hot loop is
// launch-bound, not FLOP-bound.
//
// Fragment layout as in conv_mfma.hip (verified by mfma_probe).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int SMAX = 64;   // bptt
constexpr int DMAX = 32;   // head_dim
constexpr int ALDK = 40;   // padded k-stride (elems)

__device__ __forceinline__ f32x4 mfma_bf16(const __hip_bfloat16* a_row,
                                           const __hip_bfloat16* b_col,
                                           f32x4 acc) {
    const int l = threadIdx.x & (WAVE - 1);
    const int kb = (l >> 4) * 8;
    bf16x8 a = *(const bf16x8*)(a_row + kb);
    bf16x8 b = *(const bf16x8*)(b_col + kb);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
}

// C tile (64x64) from 4 waves of 2x2 16x16 frags -> LDS, rows padded
__device__ __forceinline__ void acc_to_lds(float dst[SMAX][SMAX + 1],
                                           f32x4 acc[2][2], int wm, int wp,
                                           int l) {
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fp = 0; fp < 2; ++fp)
#pragma unroll
            for (int r = 0; r < 4; ++r)
                dst[wm + fm * 16 + (l >> 4) * 4 + r]
                   [wp + fp * 16 + (l & 15)] = acc[fm][fp][r];
}

// q,k,v (B, S, d) in T; out (B, S, d) T; p_save (B, S, S) bf16
template <typename T>
__global__ void __launch_bounds__(256)
attn_fwd_kernel(const T* __restrict__ q, const T* __restrict__ k,
                const T* __restrict__ v, T* __restrict__ out,
                __hip_bfloat16* __restrict__ p_save, int S, int d,
                float inv_temp) {
    __shared__ __hip_bfloat16 a_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 b_lds[SMAX][ALDK];
    __shared__ float s_lds[SMAX][SMAX + 1];
    __shared__ __hip_bfloat16 p_lds[SMAX][SMAX + 8];
    const long base = (long)blockIdx.x * S * d;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;

    // stage Q and K (rows 0..S-1, cols 0..d-1; zero-pad to 64x32)
    {
        const int row = tid >> 2, cb = (tid & 3) * 8;
        float va[8], vb[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = cb + j;
            const bool ok = row < S && c < d;
            va[j] = ok ? (float)q[base + (long)row * d + c] : 0.f;
            vb[j] = ok ? (float)k[base + (long)row * d + c] : 0.f;
        }
        bf16x8 ta, tb;
#pragma unroll
        for (int j = 0; j < 8; ++j) { ta[j] = (__bf16)va[j]; tb[j] = (__bf16)vb[j]; }
        *(bf16x8*)&a_lds[row][cb] = ta;
        *(bf16x8*)&b_lds[row][cb] = tb;
    }
    __syncthreads();
    // scores = Q K^T / temp
    {
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_bf16(&a_lds[wm + fm * 16 + (l & 15)][0],
                                        &b_lds[wp + fp * 16 + (l & 15)][0],
                                        acc[fm][fp]);
        acc_to_lds(s_lds, acc, wm, wp, l);
    }
    __syncthreads();
    // row softmax over the first S columns (thread t < 64 owns row t)
    if (tid < SMAX) {
        const int i = tid;
        if (i < S) {
            float mx = -1e30f;
            for (int j = 0; j < S; ++j) {
                const float sc = s_lds[i][j] * inv_temp;
                s_lds[i][j] = sc;
                mx = fmaxf(mx, sc);
            }
            float sum = 0.f;
            for (int j = 0; j < S; ++j) {
                const float e = __expf(s_lds[i][j] - mx);
                s_lds[i][j] = e;
                sum += e;
            }
            const float inv = 1.f / sum;
            for (int j = 0; j < SMAX; ++j) {
                const float p = j < S ? s_lds[i][j] * inv : 0.f;
                p_lds[i][j] = (__hip_bfloat16)p;
            }
        } else {
            for (int j = 0; j < SMAX; ++j) p_lds[i][j] = (__hip_bfloat16)0.f;
        }
    }
    __syncthreads();
    // save P rows to global (each thread 16 contiguous cols of one row)
    {
        const int row = tid >> 2, cb = (tid & 3) * 16;
        if (row < S) {
            for (int j = 0; j < 16; ++j) {
                const int c = cb + j;
                if (c < S)
                    p_save[(long)blockIdx.x * S * S + (long)row * S + c] =
                        p_lds[row][c];
            }
        }
        // stage V into b_lds
        const int vb8 = (tid & 3) * 8;
        float vv[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = vb8 + j;
            vv[j] = (row < S && c < d) ? (float)v[base + (long)row * d + c]
                                       : 0.f;
        }
        bf16x8 tv;
#pragma unroll
        for (int j = 0; j < 8; ++j) tv[j] = (__bf16)vv[j];
        *(bf16x8*)&b_lds[row][vb8] = tv;
    }
    __syncthreads();
    // out = P V  (M=S rows, N=d<=32 cols, K=S): A = P [i][j],
    // B[k=j][col]=V[j][c].  d <= 32, so only the wp=0 waves have live
    // output columns (cols 0..31); wp=32 waves idle here (their gathers
    // would also run past the 40-element V row pitch).
    if (wp == 0) {
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int kslab = 0; kslab < 2; ++kslab) {
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp) {
                    const int kb = (l >> 4) * 8 + kslab * 32;
                    bf16x8 a = *(const bf16x8*)
                        &p_lds[wm + fm * 16 + (l & 15)][kb];
                    const int col = fp * 16 + (l & 15);  // 0..31 in-bounds
                    bf16x8 b;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        b[j] = (__bf16)(float)b_lds[kb + j][col];
                    acc[fm][fp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[fm][fp], 0, 0, 0);
                }
        }
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int i = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = fp * 16 + (l & 15);
                    if (i < S && c < d)
                        out[base + (long)i * d + c] = (T)acc[fm][fp][r];
                }
    }
}

// ===================== blockwise (flash-style) path, any S ================
// For S > 64 (SURVEY §5 long-context readiness): the whole-matrix kernel
// above cannot hold P, so the blockwise path tiles KV in 64-row blocks with
// an online softmax (running row max/sum, output rescale) and saves only
// the per-row LSE.  Backward is TWO deterministic passes (dQ over KV
// blocks; dK/dV over Q blocks) with P recomputed from LSE — no atomics, so
// hipGraph replays stay bit-identical.

// forward: grid (B, ceil(S/64)); out (B,S,d), lse (B,S)
template <typename T>
__global__ void __launch_bounds__(256)
attn_fwd_block_kernel(const T* __restrict__ q, const T* __restrict__ k,
                      const T* __restrict__ v, T* __restrict__ out,
                      float* __restrict__ lse, int S, int d, float inv_temp) {
    __shared__ __hip_bfloat16 q_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 k_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 v_lds[SMAX][ALDK];
    __shared__ float s_lds[SMAX][SMAX + 1];
    __shared__ __hip_bfloat16 p_lds[SMAX][SMAX + 8];
    __shared__ float m_sh[SMAX], l_sh[SMAX], scale_sh[SMAX];
    const int b = blockIdx.x;
    const int i0 = blockIdx.y * SMAX;
    const int rows_q = min(SMAX, S - i0);
    const long base = (long)b * S * d;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    {   // stage this block's Q rows
        const int row = tid >> 2, cb = (tid & 3) * 8;
        bf16x8 t1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = cb + j;
            t1[j] = (row < rows_q && c < d)
                        ? (__bf16)(float)q[base + (long)(i0 + row) * d + c]
                        : (__bf16)0.f;
        }
        *(bf16x8*)&q_lds[row][cb] = t1;
    }
    if (tid < SMAX) {
        m_sh[tid] = -1e30f;
        l_sh[tid] = 0.f;
    }
    f32x4 oacc[2][2] = {};
    for (int j0 = 0; j0 < S; j0 += SMAX) {
        const int rows_kv = min(SMAX, S - j0);
        __syncthreads();
        {   // stage K and V blocks
            const int row = tid >> 2, cb = (tid & 3) * 8;
            bf16x8 t1, t2;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = cb + j;
                const bool ok = row < rows_kv && c < d;
                t1[j] = ok ? (__bf16)(float)k[base + (long)(j0 + row) * d + c]
                           : (__bf16)0.f;
                t2[j] = ok ? (__bf16)(float)v[base + (long)(j0 + row) * d + c]
                           : (__bf16)0.f;
            }
            *(bf16x8*)&k_lds[row][cb] = t1;
            *(bf16x8*)&v_lds[row][cb] = t2;
        }
        __syncthreads();
        {   // S1 = Q K^T
            f32x4 acc[2][2] = {};
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
                    acc[fm][fp] =
                        mfma_bf16(&q_lds[wm + fm * 16 + (l & 15)][0],
                                  &k_lds[wp + fp * 16 + (l & 15)][0],
                                  acc[fm][fp]);
            acc_to_lds(s_lds, acc, wm, wp, l);
        }
        __syncthreads();
        if (tid < SMAX) {   // online-softmax row pass
            const int i = tid;
            float scale = 1.f;
            if (i < rows_q) {
                float mx = m_sh[i];
                for (int j = 0; j < rows_kv; ++j)
                    mx = fmaxf(mx, s_lds[i][j] * inv_temp);
                scale = __expf(m_sh[i] - mx);
                float sum = 0.f;
                for (int j = 0; j < SMAX; ++j) {
                    const float e = j < rows_kv
                        ? __expf(s_lds[i][j] * inv_temp - mx) : 0.f;
                    p_lds[i][j] = (__hip_bfloat16)e;
                    sum += e;
                }
                l_sh[i] = l_sh[i] * scale + sum;
                m_sh[i] = mx;
            } else {
                for (int j = 0; j < SMAX; ++j)
                    p_lds[i][j] = (__hip_bfloat16)0.f;
            }
            scale_sh[i] = scale;
        }
        __syncthreads();
        if (wp == 0) {   // rescale O and accumulate P~ V
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        oacc[fm][fp][r] *=
                            scale_sh[wm + fm * 16 + (l >> 4) * 4 + r];
#pragma unroll
            for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
                for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                    for (int fp = 0; fp < 2; ++fp) {
                        const int kb = (l >> 4) * 8 + kslab * 32;
                        bf16x8 a = *(const bf16x8*)
                            &p_lds[wm + fm * 16 + (l & 15)][kb];
                        const int col = fp * 16 + (l & 15);
                        bf16x8 bb;
#pragma unroll
                        for (int j = 0; j < 8; ++j)
                            bb[j] = (__bf16)(float)v_lds[kb + j][col];
                        oacc[fm][fp] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                a, bb, oacc[fm][fp], 0, 0, 0);
                    }
        }
    }
    __syncthreads();
    if (wp == 0) {
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int i = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = fp * 16 + (l & 15);
                    if (i < rows_q && c < d)
                        out[base + (long)(i0 + i) * d + c] =
                            (T)(oacc[fm][fp][r] / l_sh[i]);
                }
    }
    if (tid < SMAX && tid < rows_q)
        lse[(long)b * S + i0 + tid] = m_sh[tid] + __logf(l_sh[tid]);
}

// D[b,i] = rowsum(dO * O): grid (B, ceil(S/64)), 64 threads own rows
template <typename T>
__global__ void __launch_bounds__(64)
attn_bwd_d_kernel(const T* __restrict__ dout, const T* __restrict__ out,
                  float* __restrict__ dvec, int S, int d) {
    const int b = blockIdx.x;
    const int i = blockIdx.y * 64 + threadIdx.x;
    if (i >= S) return;
    const long off = (long)b * S * d + (long)i * d;
    float s = 0.f;
    for (int c = 0; c < d; ++c)
        s += ld_f32(dout + off + c) * ld_f32(out + off + c);
    dvec[(long)b * S + i] = s;
}

// pass A: dQ = sum over KV blocks of dS K;  grid (B, ceil(S/64))
template <typename T>
__global__ void __launch_bounds__(256)
attn_bwd_dq_kernel(const T* __restrict__ dout, const T* __restrict__ q,
                   const T* __restrict__ k, const T* __restrict__ v,
                   const float* __restrict__ lse,
                   const float* __restrict__ dvec, T* __restrict__ dq,
                   int S, int d, float inv_temp) {
    __shared__ __hip_bfloat16 q_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 do_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 k_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 v_lds[SMAX][ALDK];
    __shared__ float s_lds[SMAX][SMAX + 1];
    __shared__ __hip_bfloat16 ds_lds[SMAX][SMAX + 8];
    const int b = blockIdx.x;
    const int i0 = blockIdx.y * SMAX;
    const int rows_q = min(SMAX, S - i0);
    const long base = (long)b * S * d;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    {   // stage Q and dO rows of this block
        const int row = tid >> 2, cb = (tid & 3) * 8;
        bf16x8 t1, t2;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = cb + j;
            const bool ok = row < rows_q && c < d;
            t1[j] = ok ? (__bf16)(float)q[base + (long)(i0 + row) * d + c]
                       : (__bf16)0.f;
            t2[j] = ok ? (__bf16)(float)dout[base + (long)(i0 + row) * d + c]
                       : (__bf16)0.f;
        }
        *(bf16x8*)&q_lds[row][cb] = t1;
        *(bf16x8*)&do_lds[row][cb] = t2;
    }
    f32x4 dqacc[2][2] = {};
    for (int j0 = 0; j0 < S; j0 += SMAX) {
        const int rows_kv = min(SMAX, S - j0);
        __syncthreads();
        {   // stage K and V blocks
            const int row = tid >> 2, cb = (tid & 3) * 8;
            bf16x8 t1, t2;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = cb + j;
                const bool ok = row < rows_kv && c < d;
                t1[j] = ok ? (__bf16)(float)k[base + (long)(j0 + row) * d + c]
                           : (__bf16)0.f;
                t2[j] = ok ? (__bf16)(float)v[base + (long)(j0 + row) * d + c]
                           : (__bf16)0.f;
            }
            *(bf16x8*)&k_lds[row][cb] = t1;
            *(bf16x8*)&v_lds[row][cb] = t2;
        }
        __syncthreads();
        {   // S1 = Q K^T
            f32x4 acc[2][2] = {};
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
                    acc[fm][fp] =
                        mfma_bf16(&q_lds[wm + fm * 16 + (l & 15)][0],
                                  &k_lds[wp + fp * 16 + (l & 15)][0],
                                  acc[fm][fp]);
            acc_to_lds(s_lds, acc, wm, wp, l);
        }
        __syncthreads();
        if (tid < SMAX) {   // P from LSE (into ds_lds as staging)
            const int i = tid;
            const bool live = i < rows_q;
            const float ls = live ? lse[(long)b * S + i0 + i] : 0.f;
            for (int j = 0; j < SMAX; ++j) {
                const float p = (live && j < rows_kv)
                    ? __expf(s_lds[i][j] * inv_temp - ls) : 0.f;
                ds_lds[i][j] = (__hip_bfloat16)p;
            }
        }
        __syncthreads();
        {   // dP = dO V^T
            f32x4 acc[2][2] = {};
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
                    acc[fm][fp] =
                        mfma_bf16(&do_lds[wm + fm * 16 + (l & 15)][0],
                                  &v_lds[wp + fp * 16 + (l & 15)][0],
                                  acc[fm][fp]);
            acc_to_lds(s_lds, acc, wm, wp, l);
        }
        __syncthreads();
        if (tid < SMAX) {   // dS = P*(dP - D)*inv_temp in place
            const int i = tid;
            const bool live = i < rows_q;
            const float dv_i = live ? dvec[(long)b * S + i0 + i] : 0.f;
            for (int j = 0; j < SMAX; ++j) {
                const float p = (float)ds_lds[i][j];
                const float dsv = live
                    ? p * (s_lds[i][j] - dv_i) * inv_temp : 0.f;
                ds_lds[i][j] = (__hip_bfloat16)dsv;
            }
        }
        __syncthreads();
        if (wp == 0) {   // dQ += dS K
#pragma unroll
            for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
                for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                    for (int fp = 0; fp < 2; ++fp) {
                        const int kb = (l >> 4) * 8 + kslab * 32;
                        bf16x8 a = *(const bf16x8*)
                            &ds_lds[wm + fm * 16 + (l & 15)][kb];
                        bf16x8 bb;
#pragma unroll
                        for (int j = 0; j < 8; ++j)
                            bb[j] = k_lds[kb + j][fp * 16 + (l & 15)];
                        dqacc[fm][fp] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                a, bb, dqacc[fm][fp], 0, 0, 0);
                    }
        }
    }
    __syncthreads();
    if (wp == 0) {
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int i = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = fp * 16 + (l & 15);
                    if (i < rows_q && c < d)
                        dq[base + (long)(i0 + i) * d + c] =
                            (T)dqacc[fm][fp][r];
                }
    }
}

// pass B: dK = sum over Q blocks of dS^T Q; dV = sum of P~^T dO;
// grid (B, ceil(S/64))
template <typename T>
__global__ void __launch_bounds__(256)
attn_bwd_dkv_kernel(const T* __restrict__ dout, const T* __restrict__ q,
                    const T* __restrict__ k, const T* __restrict__ v,
                    const float* __restrict__ lse,
                    const float* __restrict__ dvec, T* __restrict__ dk,
                    T* __restrict__ dv, int S, int d, float inv_temp) {
    __shared__ __hip_bfloat16 q_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 do_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 k_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 v_lds[SMAX][ALDK];
    __shared__ float s_lds[SMAX][SMAX + 1];
    __shared__ __hip_bfloat16 p_lds[SMAX][SMAX + 8];
    __shared__ __hip_bfloat16 ds_lds[SMAX][SMAX + 8];
    const int b = blockIdx.x;
    const int j0 = blockIdx.y * SMAX;
    const int rows_kv = min(SMAX, S - j0);
    const long base = (long)b * S * d;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;
    {   // stage this block's K and V rows
        const int row = tid >> 2, cb = (tid & 3) * 8;
        bf16x8 t1, t2;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = cb + j;
            const bool ok = row < rows_kv && c < d;
            t1[j] = ok ? (__bf16)(float)k[base + (long)(j0 + row) * d + c]
                       : (__bf16)0.f;
            t2[j] = ok ? (__bf16)(float)v[base + (long)(j0 + row) * d + c]
                       : (__bf16)0.f;
        }
        *(bf16x8*)&k_lds[row][cb] = t1;
        *(bf16x8*)&v_lds[row][cb] = t2;
    }
    f32x4 dkacc[2][2] = {};
    f32x4 dvacc[2][2] = {};
    for (int i0 = 0; i0 < S; i0 += SMAX) {
        const int rows_q = min(SMAX, S - i0);
        __syncthreads();
        {   // stage Q and dO blocks
            const int row = tid >> 2, cb = (tid & 3) * 8;
            bf16x8 t1, t2;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = cb + j;
                const bool ok = row < rows_q && c < d;
                t1[j] = ok ? (__bf16)(float)q[base + (long)(i0 + row) * d + c]
                           : (__bf16)0.f;
                t2[j] = ok
                    ? (__bf16)(float)dout[base + (long)(i0 + row) * d + c]
                    : (__bf16)0.f;
            }
            *(bf16x8*)&q_lds[row][cb] = t1;
            *(bf16x8*)&do_lds[row][cb] = t2;
        }
        __syncthreads();
        {   // S1 = Q K^T
            f32x4 acc[2][2] = {};
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
                    acc[fm][fp] =
                        mfma_bf16(&q_lds[wm + fm * 16 + (l & 15)][0],
                                  &k_lds[wp + fp * 16 + (l & 15)][0],
                                  acc[fm][fp]);
            acc_to_lds(s_lds, acc, wm, wp, l);
        }
        __syncthreads();
        if (tid < SMAX) {   // P from LSE
            const int i = tid;
            const bool live = i < rows_q;
            const float ls = live ? lse[(long)b * S + i0 + i] : 0.f;
            for (int j = 0; j < SMAX; ++j) {
                const float p = (live && j < rows_kv)
                    ? __expf(s_lds[i][j] * inv_temp - ls) : 0.f;
                p_lds[i][j] = (__hip_bfloat16)p;
            }
        }
        __syncthreads();
        {   // dP = dO V^T
            f32x4 acc[2][2] = {};
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp)
                    acc[fm][fp] =
                        mfma_bf16(&do_lds[wm + fm * 16 + (l & 15)][0],
                                  &v_lds[wp + fp * 16 + (l & 15)][0],
                                  acc[fm][fp]);
            acc_to_lds(s_lds, acc, wm, wp, l);
        }
        __syncthreads();
        if (tid < SMAX) {   // dS
            const int i = tid;
            const bool live = i < rows_q;
            const float dv_i = live ? dvec[(long)b * S + i0 + i] : 0.f;
            for (int j = 0; j < SMAX; ++j) {
                const float p = (float)p_lds[i][j];
                const float dsv = live
                    ? p * (s_lds[i][j] - dv_i) * inv_temp : 0.f;
                ds_lds[i][j] = (__hip_bfloat16)dsv;
            }
        }
        __syncthreads();
        if (wp == 0) {   // dK += dS^T Q ; dV += P~^T dO
#pragma unroll
            for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
                for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                    for (int fp = 0; fp < 2; ++fp) {
                        const int kb = (l >> 4) * 8 + kslab * 32;
                        bf16x8 a1, b1, a2, b2;
#pragma unroll
                        for (int j = 0; j < 8; ++j) {
                            a1[j] = ds_lds[kb + j][wm + fm * 16 + (l & 15)];
                            b1[j] = q_lds[kb + j][fp * 16 + (l & 15)];
                            a2[j] = p_lds[kb + j][wm + fm * 16 + (l & 15)];
                            b2[j] = do_lds[kb + j][fp * 16 + (l & 15)];
                        }
                        dkacc[fm][fp] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                a1, b1, dkacc[fm][fp], 0, 0, 0);
                        dvacc[fm][fp] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                a2, b2, dvacc[fm][fp], 0, 0, 0);
                    }
        }
    }
    __syncthreads();
    if (wp == 0) {
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int j = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = fp * 16 + (l & 15);
                    if (j < rows_kv && c < d) {
                        dk[base + (long)(j0 + j) * d + c] =
                            (T)dkacc[fm][fp][r];
                        dv[base + (long)(j0 + j) * d + c] =
                            (T)dvacc[fm][fp][r];
                    }
                }
    }
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DISPATCH_AT(t, ...)                                                   \
    if ((t) == at::kFloat) { using scalar_t = float; __VA_ARGS__; }           \
    else if ((t) == at::kBFloat16) { using scalar_t = __hip_bfloat16; __VA_ARGS__; } \
    else { TORCH_CHECK(false, "unsupported dtype"); }

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double temperature) {
    TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                v.is_contiguous());
    const int B = q.size(0), S = q.size(1), d = q.size(2);
    TORCH_CHECK(S <= SMAX && d <= DMAX, "attn kernel supports S<=64, d<=32");
    auto out = at::empty_like(q);
    auto p = at::empty({B, S, S}, q.options().dtype(at::kBFloat16));
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_AT(q.scalar_type(), {
        hipLaunchKernelGGL(attn_fwd_kernel<scalar_t>, dim3(B), dim3(256), 0,
                           stream, (const scalar_t*)q.data_ptr(),
                           (const scalar_t*)k.data_ptr(),
                           (const scalar_t*)v.data_ptr(),
                           (scalar_t*)out.data_ptr(),
                           (__hip_bfloat16*)p.data_ptr(), S, d,
                           (float)(1.0 / temperature));
    });
    return {out, p};
}

// --------------------------------------------------------------- backward
// dV = P^T dO;  dP = dO V^T;  dS = P*(dP - rowsum(dP*P))*inv_temp;
// dQ = dS K;  dK = dS^T Q.   One workgroup per (batch*head).
template <typename T>
__global__ void __launch_bounds__(256)
attn_bwd_kernel(const T* __restrict__ dout, const T* __restrict__ q,
                const T* __restrict__ k, const T* __restrict__ v,
                const __hip_bfloat16* __restrict__ p_save,
                T* __restrict__ dq, T* __restrict__ dk, T* __restrict__ dv,
                int S, int d, float inv_temp) {
    __shared__ __hip_bfloat16 do_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 q_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 k_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 v_lds[SMAX][ALDK];
    __shared__ __hip_bfloat16 p_lds[SMAX][SMAX + 8];
    __shared__ __hip_bfloat16 ds_lds[SMAX][SMAX + 8];
    __shared__ float t_lds[SMAX][SMAX + 1];
    const long base = (long)blockIdx.x * S * d;
    const long pbase = (long)blockIdx.x * S * S;
    const int tid = threadIdx.x;
    const int l = tid & (WAVE - 1);
    const int wave = tid / WAVE;
    const int wm = (wave >> 1) * 32;
    const int wp = (wave & 1) * 32;

    {   // stage dO, Q, K, V (zero-padded) and P
        const int row = tid >> 2, cb = (tid & 3) * 8;
        bf16x8 t1, t2, t3, t4;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = cb + j;
            const bool ok = row < S && c < d;
            t1[j] = ok ? (__bf16)(float)dout[base + (long)row * d + c]
                       : (__bf16)0.f;
            t2[j] = ok ? (__bf16)(float)q[base + (long)row * d + c]
                       : (__bf16)0.f;
            t3[j] = ok ? (__bf16)(float)k[base + (long)row * d + c]
                       : (__bf16)0.f;
            t4[j] = ok ? (__bf16)(float)v[base + (long)row * d + c]
                       : (__bf16)0.f;
        }
        *(bf16x8*)&do_lds[row][cb] = t1;
        *(bf16x8*)&q_lds[row][cb] = t2;
        *(bf16x8*)&k_lds[row][cb] = t3;
        *(bf16x8*)&v_lds[row][cb] = t4;
        const int pcb = (tid & 3) * 16;
        for (int j = 0; j < 16; ++j) {
            const int c = pcb + j;
            p_lds[row][c] = (row < S && c < S)
                                ? p_save[pbase + (long)row * S + c]
                                : (__hip_bfloat16)0.f;
        }
    }
    __syncthreads();
    {   // dP = dO V^T  (A = dO[i][c], B[k=c][col=j] = V[j][c] -> v rows)
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
                acc[fm][fp] = mfma_bf16(&do_lds[wm + fm * 16 + (l & 15)][0],
                                        &v_lds[wp + fp * 16 + (l & 15)][0],
                                        acc[fm][fp]);
        acc_to_lds(t_lds, acc, wm, wp, l);
    }
    __syncthreads();
    // dS row pass (thread t < 64 owns row t)
    if (tid < SMAX) {
        const int i = tid;
        float rd = 0.f;
        for (int j = 0; j < S; ++j)
            rd += t_lds[i][j] * (float)p_lds[i][j];
        for (int j = 0; j < SMAX; ++j) {
            const float p = (float)p_lds[i][j];
            const float dsv = (i < S && j < S)
                                  ? p * (t_lds[i][j] - rd) * inv_temp : 0.f;
            ds_lds[i][j] = (__hip_bfloat16)dsv;
        }
    }
    __syncthreads();
    if (wp == 0) {  // dQ = dS K (d <= 32: only wp=0 cols live)
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp) {
                    const int kb = (l >> 4) * 8 + kslab * 32;
                    bf16x8 a = *(const bf16x8*)
                        &ds_lds[wm + fm * 16 + (l & 15)][kb];
                    bf16x8 b;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        b[j] = k_lds[kb + j][wp + fp * 16 + (l & 15)];
                    acc[fm][fp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[fm][fp], 0, 0, 0);
                }
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int i = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = wp + fp * 16 + (l & 15);
                    if (i < S && c < d)
                        dq[base + (long)i * d + c] = (T)acc[fm][fp][r];
                }
    }
    if (wp == 0) {  // dK = dS^T Q (d <= 32)
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp) {
                    const int kb = (l >> 4) * 8 + kslab * 32;
                    bf16x8 a, b;
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        a[j] = ds_lds[kb + j][wm + fm * 16 + (l & 15)];
                        b[j] = q_lds[kb + j][wp + fp * 16 + (l & 15)];
                    }
                    acc[fm][fp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[fm][fp], 0, 0, 0);
                }
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int j = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = wp + fp * 16 + (l & 15);
                    if (j < S && c < d)
                        dk[base + (long)j * d + c] = (T)acc[fm][fp][r];
                }
    }
    if (wp == 0) {  // dV = P^T dO (d <= 32)
        f32x4 acc[2][2] = {};
#pragma unroll
        for (int kslab = 0; kslab < 2; ++kslab)
#pragma unroll
            for (int fm = 0; fm < 2; ++fm)
#pragma unroll
                for (int fp = 0; fp < 2; ++fp) {
                    const int kb = (l >> 4) * 8 + kslab * 32;
                    bf16x8 a, b;
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        a[j] = p_lds[kb + j][wm + fm * 16 + (l & 15)];
                        b[j] = do_lds[kb + j][wp + fp * 16 + (l & 15)];
                    }
                    acc[fm][fp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[fm][fp], 0, 0, 0);
                }
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
            for (int fp = 0; fp < 2; ++fp)
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int j = wm + fm * 16 + (l >> 4) * 4 + r;
                    const int c = wp + fp * 16 + (l & 15);
                    if (j < S && c < d)
                        dv[base + (long)j * d + c] = (T)acc[fm][fp][r];
                }
    }
}

std::vector<at::Tensor> attn_fwd_block(at::Tensor q, at::Tensor k,
                                       at::Tensor v, double temperature) {
    TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                v.is_contiguous());
    const int B = q.size(0), S = q.size(1), d = q.size(2);
    TORCH_CHECK(d <= DMAX, "blockwise attn supports head_dim<=32");
    auto out = at::empty_like(q);
    auto lse = at::empty({B, S}, q.options().dtype(at::kFloat));
    auto stream = at::hip::getCurrentHIPStream();
    const int nqb = (S + SMAX - 1) / SMAX;
    DISPATCH_AT(q.scalar_type(), {
        hipLaunchKernelGGL(attn_fwd_block_kernel<scalar_t>, dim3(B, nqb),
                           dim3(256), 0, stream,
                           (const scalar_t*)q.data_ptr(),
                           (const scalar_t*)k.data_ptr(),
                           (const scalar_t*)v.data_ptr(),
                           (scalar_t*)out.data_ptr(),
                           lse.data_ptr<float>(), S, d,
                           (float)(1.0 / temperature));
    });
    return {out, lse};
}

std::vector<at::Tensor> attn_bwd_block(at::Tensor dout, at::Tensor q,
                                       at::Tensor k, at::Tensor v,
                                       at::Tensor out, at::Tensor lse,
                                       double temperature) {
    const int B = q.size(0), S = q.size(1), d = q.size(2);
    auto dq = at::empty_like(q);
    auto dk = at::empty_like(k);
    auto dv = at::empty_like(v);
    auto dvec = at::empty({B, (long)S}, q.options().dtype(at::kFloat));
    auto dc = dout.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    const int nb = (S + SMAX - 1) / SMAX;
    DISPATCH_AT(q.scalar_type(), {
        hipLaunchKernelGGL(attn_bwd_d_kernel<scalar_t>, dim3(B, nb),
                           dim3(64), 0, stream,
                           (const scalar_t*)dc.data_ptr(),
                           (const scalar_t*)out.data_ptr(),
                           dvec.data_ptr<float>(), S, d);
        hipLaunchKernelGGL(attn_bwd_dq_kernel<scalar_t>, dim3(B, nb),
                           dim3(256), 0, stream,
                           (const scalar_t*)dc.data_ptr(),
                           (const scalar_t*)q.data_ptr(),
                           (const scalar_t*)k.data_ptr(),
                           (const scalar_t*)v.data_ptr(),
                           lse.data_ptr<float>(), dvec.data_ptr<float>(),
                           (scalar_t*)dq.data_ptr(), S, d,
                           (float)(1.0 / temperature));
        hipLaunchKernelGGL(attn_bwd_dkv_kernel<scalar_t>, dim3(B, nb),
                           dim3(256), 0, stream,
                           (const scalar_t*)dc.data_ptr(),
                           (const scalar_t*)q.data_ptr(),
                           (const scalar_t*)k.data_ptr(),
                           (const scalar_t*)v.data_ptr(),
                           lse.data_ptr<float>(), dvec.data_ptr<float>(),
                           (scalar_t*)dk.data_ptr(),
                           (scalar_t*)dv.data_ptr(), S, d,
                           (float)(1.0 / temperature));
    });
    return {dq, dk, dv};
}

std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor p,
                                 double temperature) {
    const int B = q.size(0), S = q.size(1), d = q.size(2);
    auto dq = at::empty_like(q);
    auto dk = at::empty_like(k);
    auto dv = at::empty_like(v);
    auto dc = dout.contiguous();
    auto stream = at::hip::getCurrentHIPStream();
    DISPATCH_AT(q.scalar_type(), {
        hipLaunchKernelGGL(attn_bwd_kernel<scalar_t>, dim3(B), dim3(256), 0,
                           stream, (const scalar_t*)dc.data_ptr(),
                           (const scalar_t*)q.data_ptr(),
                           (const scalar_t*)k.data_ptr(),
                           (const scalar_t*)v.data_ptr(),
                           (const __hip_bfloat16*)p.data_ptr(),
                           (scalar_t*)dq.data_ptr(),
                           (scalar_t*)dk.data_ptr(),
                           (scalar_t*)dv.data_ptr(), S, d,
                           (float)(1.0 / temperature));
    });
    return {dq, dk, dv};
}
