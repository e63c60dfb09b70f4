// MFMA flash attention for the LARGE-context NP regime (K >= ~100:
// image self-attention over 300..4000 pixels, cross-attention to big
// context sets).  The thread-per-query VALU kernel in attn.hip stays the
// dispatch target for tiny K (1D models, K ~ 10..100) where MFMA tiles
// would be mostly padding; ext.cpp picks per call.
//
// Same op contract as attn.hip: out = softmax_k(scale*QK^T)V with
// row-logsumexp written for the backward; per-head D, Dv <= 32 (configs use
// 16), contraction padded to the MFMA K=32 slice.
//
// Layouts (v_mfma_f32_16x16x32_bf16; cdna_hip_programming.md):
//   A: lane row = lane&15, k = (lane>>4)*8..+8 | B: lane col = lane&15,
//   k likewise | C/D: lane col = lane&15, row = (lane>>4)*4 + reg.
//
// forward, per workgroup (4 waves, 64 queries):
//   Q frag per wave lives in registers (scale folded in).  Loop 32-key
//   tiles staged in LDS: S^T = mfma(K, Q^T) gives per lane 8 scores of ONE
//   query (col = q); the online-softmax max/sum for a query is 2
//   shfl_xor(16/32) lane reductions.  P^T goes through LDS (layout flip
//   C/D -> A), O accumulates via mfma(P, V^T) with per-row rescale factors
//   broadcast by shfl.
//
// backward (FA2 split): npf_attn_delta (tiny elementwise-reduce), then
//   dKV: workgroup owns 64 keys, loops 32-query tiles, P^T/dS^T recomputed
//        from lse, dK/dV accumulate in registers (no atomics);
//   dQ:  workgroup owns 64 queries, loops 32-key tiles.

#include "common.h"

#define AM_BLOCK 256        // 4 waves
#define AM_D 32             // padded head dim (contraction slice)
#define AM_TQ 64            // queries per workgroup (fwd / dq)
#define AM_TK 64            // keys per workgroup (dkv)
#define AM_KT 32            // keys per inner tile
#define AM_PAD 8
#define AM_STRIDE (AM_D + AM_PAD)       // K/Q/V row stride in LDS
#define AM_PSTRIDE (AM_KT + AM_PAD)     // P/dS row stride in LDS
#define NEG_INF (-1e30f)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ bf16x8 ldb8(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

// stage [rows, width<=AM_D] (global fp32/bf16) -> LDS bf16 [rows][AM_STRIDE],
// zero-padding width to AM_D; rows beyond `rows` zeroed up to rows_p
template <typename T>
__device__ __forceinline__ void stage_rows(__hip_bfloat16 (*dst)[AM_STRIDE],
                                           const T* __restrict__ src, int rows,
                                           int rows_p, int width, float mul) {
  for (int i = threadIdx.x; i < rows_p * AM_D; i += AM_BLOCK) {
    const int r = i / AM_D, c = i % AM_D;
    float v = 0.f;
    if (r < rows && c < width) v = ldf(src + (size_t)r * width + c) * mul;
    dst[r][c] = __float2bfloat16(v);
  }
}

// transposed: LDS [AM_D][cols] (row = feature, col = item)
template <typename T>
__device__ __forceinline__ void stage_rows_t(
    __hip_bfloat16 (*dst)[AM_KT + AM_PAD], const T* __restrict__ src, int rows,
    int rows_p, int width, float mul) {
  for (int i = threadIdx.x; i < rows_p * AM_D; i += AM_BLOCK) {
    const int r = i / AM_D, c = i % AM_D;
    float v = 0.f;
    if (r < rows && c < width) v = ldf(src + (size_t)r * width + c) * mul;
    dst[c][r] = __float2bfloat16(v);
  }
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(AM_BLOCK) npf_attn_mfma_fwd(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    T* __restrict__ out, float* __restrict__ lse, int N, int Q, int K, int D,
    int Dv, float scale) {
  __shared__ __align__(16) __hip_bfloat16 k_lds[AM_KT][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 vt_lds[AM_D][AM_KT + AM_PAD];
  __shared__ __align__(16) __hip_bfloat16 q_lds[AM_TQ][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 p_lds[AM_TQ][AM_PSTRIDE];

  const int n = blockIdx.x;
  const int q0 = blockIdx.y * AM_TQ;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row0 = wave * 16;  // this wave's 16 queries within the tile

  const int qt = min(AM_TQ, Q - q0);
  stage_rows(q_lds, q + ((size_t)n * Q + q0) * D, qt, AM_TQ, D, scale);
  __syncthreads();

  // Q B-frag: col = my query (lane&15), k = d
  const bf16x8 qfrag = ldb8(&q_lds[row0 + (lane & 15)][(lane >> 4) * 8]);

  float m_q = NEG_INF, l_q = 0.f;     // per-lane: this lane's query (col)
  f32x4 o_acc[AM_D / 16];
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) o_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += AM_KT) {
    const int kt = min(AM_KT, K - k0);
    stage_rows(k_lds, k + ((size_t)n * K + k0) * D, kt, AM_KT, D, 1.f);
    stage_rows_t(vt_lds, v + ((size_t)n * K + k0) * Dv, kt, AM_KT, Dv, 1.f);
    __syncthreads();

    // S^T tiles: m = key (two 16-blocks), n = my query
    f32x4 st[2];
    #pragma unroll
    for (int t = 0; t < 2; ++t) {
      const bf16x8 kf = ldb8(&k_lds[t * 16 + (lane & 15)][(lane >> 4) * 8]);
      st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          kf, qfrag, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
    }
    // my 8 scores: keys t*16 + (lane>>4)*4 + i, query = lane&15
    float s[8];
    #pragma unroll
    for (int t = 0; t < 2; ++t)
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int key = t * 16 + (lane >> 4) * 4 + i;
        s[t * 4 + i] = (key < kt) ? st[t][i] : NEG_INF;
      }
    // tile max for my query: reduce over the 4 lane-groups holding it
    float tm = s[0];
    #pragma unroll
    for (int i = 1; i < 8; ++i) tm = fmaxf(tm, s[i]);
    tm = fmaxf(tm, __shfl_xor(tm, 16, 64));
    tm = fmaxf(tm, __shfl_xor(tm, 32, 64));

    const float m_new = fmaxf(m_q, tm);
    const float r = __expf(m_q - m_new);
    float tl = 0.f;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      s[i] = (s[i] > NEG_INF * 0.5f) ? __expf(s[i] - m_new) : 0.f;
      tl += s[i];
    }
    tl += __shfl_xor(tl, 16, 64);
    tl += __shfl_xor(tl, 32, 64);
    l_q = l_q * r + tl;
    m_q = m_new;

    // write P transposed into [query][key] for the A-frag of P@V
    {
      const int myq = row0 + (lane & 15);
      #pragma unroll
      for (int t = 0; t < 2; ++t)
        #pragma unroll
        for (int i = 0; i < 4; ++i)
          p_lds[myq][t * 16 + (lane >> 4) * 4 + i] =
              __float2bfloat16(s[t * 4 + i]);
    }
    __syncthreads();

    // O rescale: factor of the query owning each C/D row
    const float r_row0 = __shfl(r, (lane >> 4) * 4 + 0, 64);
    const float r_row1 = __shfl(r, (lane >> 4) * 4 + 1, 64);
    const float r_row2 = __shfl(r, (lane >> 4) * 4 + 2, 64);
    const float r_row3 = __shfl(r, (lane >> 4) * 4 + 3, 64);
    const bf16x8 pf = ldb8(&p_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
    #pragma unroll
    for (int i = 0; i < AM_D / 16; ++i) {
      o_acc[i][0] *= r_row0;
      o_acc[i][1] *= r_row1;
      o_acc[i][2] *= r_row2;
      o_acc[i][3] *= r_row3;
      const bf16x8 vf = ldb8(&vt_lds[i * 16 + (lane & 15)][(lane >> 4) * 8]);
      o_acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, o_acc[i], 0,
                                                         0, 0);
    }
    __syncthreads();
  }

  // epilogue: divide by l (per C/D row's query), store out + lse
  const float il0 = 1.f / __shfl(l_q, (lane >> 4) * 4 + 0, 64);
  const float il1 = 1.f / __shfl(l_q, (lane >> 4) * 4 + 1, 64);
  const float il2 = 1.f / __shfl(l_q, (lane >> 4) * 4 + 2, 64);
  const float il3 = 1.f / __shfl(l_q, (lane >> 4) * 4 + 3, 64);
  const float il[4] = {il0, il1, il2, il3};
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) {
    const int dv = i * 16 + (lane & 15);
    if (dv >= Dv) continue;
    #pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int qq = q0 + row0 + (lane >> 4) * 4 + rr;
      if (qq < Q) stf(out + ((size_t)n * Q + qq) * Dv + dv, o_acc[i][rr] * il[rr]);
    }
  }
  const int myq = q0 + row0 + (lane & 15);
  if ((lane >> 4) == 0 && myq < Q)
    lse[(size_t)n * Q + myq] = m_q + __logf(l_q);
}

// ---------------------------------------------------------------------------
// delta_q = rowsum(dO * O)
// ---------------------------------------------------------------------------

template <typename T>
__global__ void npf_attn_delta(const T* __restrict__ dout,
                               const T* __restrict__ out,
                               float* __restrict__ delta, long rows, int Dv) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= rows) return;
  float acc = 0.f;
  for (int d = 0; d < Dv; ++d)
    acc += ldf(dout + i * Dv + d) * ldf(out + i * Dv + d);
  delta[i] = acc;
}

// ---------------------------------------------------------------------------
// backward dK/dV: workgroup owns AM_TK keys, loops 32-query tiles
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(AM_BLOCK) npf_attn_mfma_bwd_dkv(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const T* __restrict__ dout, T* __restrict__ dk, T* __restrict__ dv,
    int N, int Q, int K, int D, int Dv, float scale) {
  __shared__ __align__(16) __hip_bfloat16 k_lds[AM_TK][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 v_lds[AM_TK][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 q_lds[AM_KT][AM_STRIDE];   // scaled
  __shared__ __align__(16) __hip_bfloat16 qt_lds[AM_D][AM_KT + AM_PAD];  // unscaled
  __shared__ __align__(16) __hip_bfloat16 do_lds[AM_KT][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 ds_lds[AM_TK][AM_PSTRIDE];
  __shared__ __align__(16) __hip_bfloat16 pt_lds[AM_TK][AM_PSTRIDE];
  __shared__ float lse_lds[AM_KT];
  __shared__ float dl_lds[AM_KT];

  const int n = blockIdx.x;
  const int k0 = blockIdx.y * AM_TK;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row0 = wave * 16;  // this wave's 16 keys

  const int ktw = min(AM_TK, K - k0);
  stage_rows(k_lds, k + ((size_t)n * K + k0) * D, ktw, AM_TK, D, 1.f);
  stage_rows(v_lds, v + ((size_t)n * K + k0) * Dv, ktw, AM_TK, Dv, 1.f);
  __syncthreads();

  const bf16x8 kfrag = ldb8(&k_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
  const bf16x8 vfrag = ldb8(&v_lds[row0 + (lane & 15)][(lane >> 4) * 8]);

  f32x4 dk_acc[AM_D / 16], dv_acc[AM_D / 16];
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) {
    dk_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  for (int qq0 = 0; qq0 < Q; qq0 += AM_KT) {
    const int qtile = min(AM_KT, Q - qq0);
    stage_rows(q_lds, q + ((size_t)n * Q + qq0) * D, qtile, AM_KT, D, scale);
    stage_rows_t(qt_lds, q + ((size_t)n * Q + qq0) * D, qtile, AM_KT, D, 1.f);
    stage_rows(do_lds, dout + ((size_t)n * Q + qq0) * Dv, qtile, AM_KT, Dv,
               1.f);
    for (int i = threadIdx.x; i < AM_KT; i += AM_BLOCK) {
      const bool in = qq0 + i < Q;
      lse_lds[i] = in ? lse[(size_t)n * Q + qq0 + i] : 0.f;
      dl_lds[i] = in ? delta[(size_t)n * Q + qq0 + i] : 0.f;
    }
    __syncthreads();

    // loop the 32-query tile in two 16-col mfma? Both handled by q subtiles
    // of 16 columns: for each 16-query group g: S^T = mfma(K16, Q^T)
    #pragma unroll
    for (int g = 0; g < 2; ++g) {
      // B-frag: col = query (lane&15) within group, k = d
      const bf16x8 qf =
          ldb8(&q_lds[g * 16 + (lane & 15)][(lane >> 4) * 8]);
      const bf16x8 dof =
          ldb8(&do_lds[g * 16 + (lane & 15)][(lane >> 4) * 8]);
      // S^T[key][q] (keys = my wave's rows): col=q, row=key
      const f32x4 st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          kfrag, qf, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
      // dP^T[key][q]
      const f32x4 dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          vfrag, dof, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
      const int myq = g * 16 + (lane & 15);
      const float ls = lse_lds[myq];
      const float dl = dl_lds[myq];
      const bool qin = (qq0 + myq < Q);
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int key = row0 + (lane >> 4) * 4 + i;
        float p = 0.f, ds = 0.f;
        if (qin && key < ktw) {
          p = __expf(st[i] - ls);
          ds = p * (dpt[i] - dl);
        }
        pt_lds[key][g * 16 + (lane & 15)] = __float2bfloat16(p);
        ds_lds[key][g * 16 + (lane & 15)] = __float2bfloat16(ds);
      }
    }
    __syncthreads();

    // dK += dS^T @ Q (unscaled Q^T staged; scale applied at store)
    const bf16x8 ptf = ldb8(&pt_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
    const bf16x8 dsf = ldb8(&ds_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
    #pragma unroll
    for (int i = 0; i < AM_D / 16; ++i) {
      const bf16x8 qtf = ldb8(&qt_lds[i * 16 + (lane & 15)][(lane >> 4) * 8]);
      dk_acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, qtf, dk_acc[i],
                                                          0, 0, 0);
    }
    __syncthreads();
    // restage dO transposed into qt_lds (reuse buffer) for dV
    stage_rows_t(qt_lds, dout + ((size_t)n * Q + qq0) * Dv, qtile, AM_KT, Dv,
                 1.f);
    __syncthreads();
    #pragma unroll
    for (int i = 0; i < AM_D / 16; ++i) {
      const bf16x8 dotf = ldb8(&qt_lds[i * 16 + (lane & 15)][(lane >> 4) * 8]);
      dv_acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dotf, dv_acc[i],
                                                          0, 0, 0);
    }
    __syncthreads();
    // restage Q^T for the next tile happens at loop top
  }

  // store dK (x scale) and dV: C/D row = key, col = d
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) {
    const int d = i * 16 + (lane & 15);
    #pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int key = k0 + row0 + (lane >> 4) * 4 + rr;
      if (key < K) {
        if (d < D)
          stf(dk + ((size_t)n * K + key) * D + d, dk_acc[i][rr] * scale);
        if (d < Dv) stf(dv + ((size_t)n * K + key) * Dv + d, dv_acc[i][rr]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ: workgroup owns AM_TQ queries, loops 32-key tiles
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(AM_BLOCK) npf_attn_mfma_bwd_dq(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const T* __restrict__ dout, T* __restrict__ dq, int N, int Q, int K,
    int D, int Dv, float scale) {
  __shared__ __align__(16) __hip_bfloat16 q_lds[AM_TQ][AM_STRIDE];  // scaled
  __shared__ __align__(16) __hip_bfloat16 do_lds[AM_TQ][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 k_lds[AM_KT][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 v_lds[AM_KT][AM_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 kt_lds[AM_D][AM_KT + AM_PAD];
  __shared__ __align__(16) __hip_bfloat16 ds_lds[AM_TQ][AM_PSTRIDE];
  __shared__ float lse_lds[AM_TQ];
  __shared__ float dl_lds[AM_TQ];

  const int n = blockIdx.x;
  const int q0 = blockIdx.y * AM_TQ;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row0 = wave * 16;

  const int qt = min(AM_TQ, Q - q0);
  stage_rows(q_lds, q + ((size_t)n * Q + q0) * D, qt, AM_TQ, D, scale);
  stage_rows(do_lds, dout + ((size_t)n * Q + q0) * Dv, qt, AM_TQ, Dv, 1.f);
  for (int i = threadIdx.x; i < AM_TQ; i += AM_BLOCK) {
    const bool in = q0 + i < Q;
    lse_lds[i] = in ? lse[(size_t)n * Q + q0 + i] : 0.f;
    dl_lds[i] = in ? delta[(size_t)n * Q + q0 + i] : 0.f;
  }
  __syncthreads();

  const bf16x8 qfrag = ldb8(&q_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
  const bf16x8 dofrag = ldb8(&do_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
  float my_lse[4], my_dl[4];
  #pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    my_lse[rr] = lse_lds[row0 + (lane >> 4) * 4 + rr];
    my_dl[rr] = dl_lds[row0 + (lane >> 4) * 4 + rr];
  }

  f32x4 dq_acc[AM_D / 16];
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) dq_acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += AM_KT) {
    const int kt = min(AM_KT, K - k0);
    stage_rows(k_lds, k + ((size_t)n * K + k0) * D, kt, AM_KT, D, 1.f);
    stage_rows(v_lds, v + ((size_t)n * K + k0) * Dv, kt, AM_KT, Dv, 1.f);
    stage_rows_t(kt_lds, k + ((size_t)n * K + k0) * D, kt, AM_KT, D, 1.f);
    __syncthreads();

    // S[q][key] and dP[q][key] per 16-key group; dS staged to LDS
    #pragma unroll
    for (int g = 0; g < 2; ++g) {
      const bf16x8 kf = ldb8(&k_lds[g * 16 + (lane & 15)][(lane >> 4) * 8]);
      const bf16x8 vf = ldb8(&v_lds[g * 16 + (lane & 15)][(lane >> 4) * 8]);
      const f32x4 st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qfrag, kf, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
      const f32x4 dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          dofrag, vf, f32x4{0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
      const int key = g * 16 + (lane & 15);
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int myq = row0 + (lane >> 4) * 4 + rr;
        float ds = 0.f;
        if (q0 + myq < Q && k0 + key < K) {
          const float p = __expf(st[rr] - my_lse[rr]);
          ds = p * (dpt[rr] - my_dl[rr]);
        }
        ds_lds[myq][key] = __float2bfloat16(ds);
      }
    }
    __syncthreads();

    // dQ += dS @ K  (B-frag: col = d, k = key from K^T)
    const bf16x8 dsf = ldb8(&ds_lds[row0 + (lane & 15)][(lane >> 4) * 8]);
    #pragma unroll
    for (int i = 0; i < AM_D / 16; ++i) {
      const bf16x8 ktf = ldb8(&kt_lds[i * 16 + (lane & 15)][(lane >> 4) * 8]);
      dq_acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, ktf, dq_acc[i],
                                                          0, 0, 0);
    }
    __syncthreads();
  }

  // store dQ x scale: C/D row = q, col = d
  #pragma unroll
  for (int i = 0; i < AM_D / 16; ++i) {
    const int d = i * 16 + (lane & 15);
    if (d >= D) continue;
    #pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int qq = q0 + row0 + (lane >> 4) * 4 + rr;
      if (qq < Q) stf(dq + ((size_t)n * Q + qq) * D + d, dq_acc[i][rr] * scale);
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

extern "C" void npf_attn_mfma_fwd_launch_f32(
    const void* q, const void* k, const void* v, void* out, float* lse, int N,
    int Q, int K, int D, int Dv, float scale, hipStream_t stream) {
  dim3 grid(N, (Q + AM_TQ - 1) / AM_TQ);
  hipLaunchKernelGGL((npf_attn_mfma_fwd<float>), grid, dim3(AM_BLOCK), 0,
                     stream, (const float*)q, (const float*)k,
                     (const float*)v, (float*)out, lse, N, Q, K, D, Dv, scale);
}

extern "C" void npf_attn_mfma_fwd_launch_bf16(
    const void* q, const void* k, const void* v, void* out, float* lse, int N,
    int Q, int K, int D, int Dv, float scale, hipStream_t stream) {
  dim3 grid(N, (Q + AM_TQ - 1) / AM_TQ);
  hipLaunchKernelGGL((npf_attn_mfma_fwd<__hip_bfloat16>), grid,
                     dim3(AM_BLOCK), 0, stream, (const __hip_bfloat16*)q,
                     (const __hip_bfloat16*)k, (const __hip_bfloat16*)v,
                     (__hip_bfloat16*)out, lse, N, Q, K, D, Dv, scale);
}

template <typename T>
static void attn_mfma_bwd_launch(const void* q, const void* k, const void* v,
                                 const void* out, const float* lse,
                                 const void* dout, float* delta, void* dq,
                                 void* dk, void* dv, int N, int Q, int K,
                                 int D, int Dv, float scale,
                                 hipStream_t stream) {
  const long rows = (long)N * Q;
  hipLaunchKernelGGL((npf_attn_delta<T>),
                     dim3((unsigned)((rows + 255) / 256)), dim3(256), 0,
                     stream, (const T*)dout, (const T*)out, delta, rows, Dv);
  dim3 gkv(N, (K + AM_TK - 1) / AM_TK);
  hipLaunchKernelGGL((npf_attn_mfma_bwd_dkv<T>), gkv, dim3(AM_BLOCK), 0,
                     stream, (const T*)q, (const T*)k, (const T*)v, lse,
                     delta, (const T*)dout, (T*)dk, (T*)dv, N, Q, K, D, Dv,
                     scale);
  dim3 gq(N, (Q + AM_TQ - 1) / AM_TQ);
  hipLaunchKernelGGL((npf_attn_mfma_bwd_dq<T>), gq, dim3(AM_BLOCK), 0, stream,
                     (const T*)q, (const T*)k, (const T*)v, lse, delta,
                     (const T*)dout, (T*)dq, N, Q, K, D, Dv, scale);
}

extern "C" void npf_attn_mfma_bwd_launch_f32(
    const void* q, const void* k, const void* v, const void* out,
    const float* lse, const void* dout, float* delta, void* dq, void* dk,
    void* dv, int N, int Q, int K, int D, int Dv, float scale,
    hipStream_t stream) {
  attn_mfma_bwd_launch<float>(q, k, v, out, lse, dout, delta, dq, dk, dv, N,
                              Q, K, D, Dv, scale, stream);
}

extern "C" void npf_attn_mfma_bwd_launch_bf16(
    const void* q, const void* k, const void* v, const void* out,
    const float* lse, const void* dout, float* delta, void* dq, void* dk,
    void* dv, int N, int Q, int K, int D, int Dv, float scale,
    hipStream_t stream) {
  attn_mfma_bwd_launch<__hip_bfloat16>(q, k, v, out, lse, dout, delta, dq,
                                       dk, dv, N, Q, K, D, Dv, scale, stream);
}
