#include "hip/hip_runtime.h"
// Fused scaled-dot cross-attention for the NP shape regime:
//   out[n,q,:] = softmax_k(scale * Q[n,q,:] . K[n,k,:]) @ V[n,k,:]
//
// Shape regime (SURVEY.md §2.3 "Cross-attention"): per-head dims D, Dv <= 32
// (the shipped configs use 128-dim / 8 heads = 16), keys = contexts (small,
// 10..1300), queries = targets (large, 128..4096), batch = B*heads.
//
// Design (MI355X/CDNA4):
// - forward: THREAD-per-query online softmax.  K/V tiles are staged in LDS
//   (all 64 lanes of a wave read the same (k,d) element => LDS broadcast, no
//   bank conflicts), the query row and output accumulator live in registers
//   (<= 2*DMAX floats).  The [Q,K] score matrix never exists in HBM; per-row
//   logsumexp is written for the backward recompute (flash-style).
// - backward: two kernels with the same staging discipline —
//   dq: thread-per-query, recomputing w = exp(s - lse);
//   dk/dv: thread-per-key over LDS-staged query tiles (q, dout, lse, delta).
//
// This is a latency/bandwidth-shaped problem (D=16 dots), not an MFMA one:
// at C~50 the whole K/V set is ~6 KB — MFMA tiles would be >90% padding.

#include "common.h"

#define ATTN_BLOCK 256
#define ATTN_KTILE 128   // keys staged per LDS round
#define ATTN_QTILE 64    // queries staged per LDS round (backward dkv)
#define NEG_INF (-1e30f)

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <typename T, int DMAX>
__global__ void __launch_bounds__(ATTN_BLOCK) npf_attn_fwd_tpq(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    T* __restrict__ out, float* __restrict__ lse,
    int N, int Q, int K, int D, int Dv, float scale) {
  __shared__ float ks[ATTN_KTILE * DMAX];
  __shared__ float vs[ATTN_KTILE * DMAX];

  const int n = blockIdx.x;
  const int qi = blockIdx.y * ATTN_BLOCK + threadIdx.x;

  float qreg[DMAX];
  float acc[DMAX];
  if (qi < Q) {
    const T* qp = q + ((size_t)n * Q + qi) * D;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) qreg[d] = (d < D) ? ldf(qp + d) * scale : 0.f;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) acc[d] = 0.f;
  }
  float m = NEG_INF, l = 0.f;

  for (int k0 = 0; k0 < K; k0 += ATTN_KTILE) {
    const int kt = min(ATTN_KTILE, K - k0);
    // cooperative K/V tile staging (coalesced: consecutive threads read
    // consecutive elements of the [kt, D] row-major block)
    for (int i = threadIdx.x; i < kt * D; i += ATTN_BLOCK)
      ks[(i / D) * DMAX + (i % D)] = ldf(k + ((size_t)n * K + k0) * D + i);
    for (int i = threadIdx.x; i < kt * Dv; i += ATTN_BLOCK)
      vs[(i / Dv) * DMAX + (i % Dv)] = ldf(v + ((size_t)n * K + k0) * Dv + i);
    __syncthreads();

    if (qi < Q) {
      for (int c = 0; c < kt; ++c) {
        float s = 0.f;
        #pragma unroll
        for (int d = 0; d < DMAX; ++d)
          if (d < D) s += qreg[d] * ks[c * DMAX + d];
        if (s > m) {
          const float r = __expf(m - s);
          l *= r;
          #pragma unroll
          for (int d = 0; d < DMAX; ++d) acc[d] *= r;
          m = s;
        }
        const float w = __expf(s - m);
        l += w;
        #pragma unroll
        for (int d = 0; d < DMAX; ++d)
          if (d < Dv) acc[d] += w * vs[c * DMAX + d];
      }
    }
    __syncthreads();
  }

  if (qi < Q) {
    T* op = out + ((size_t)n * Q + qi) * Dv;
    const float inv_l = 1.f / l;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d)
      if (d < Dv) stf(op + d, acc[d] * inv_l);
    lse[(size_t)n * Q + qi] = m + __logf(l);
  }
}

// ---------------------------------------------------------------------------
// backward: dq (thread-per-query)
//   dq[q,d] = scale * sum_c w_qc (g_qc - delta_q) k[c,d]
//   with w_qc = exp(scale q.k - lse_q), g_qc = dout[q,:].v[c,:],
//   delta_q = dout[q,:].out[q,:]
// ---------------------------------------------------------------------------

template <typename T, int DMAX>
__global__ void __launch_bounds__(ATTN_BLOCK) npf_attn_bwd_dq(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const T* __restrict__ out, const float* __restrict__ lse,
    const T* __restrict__ dout, T* __restrict__ dq,
    int N, int Q, int K, int D, int Dv, float scale) {
  __shared__ float ks[ATTN_KTILE * DMAX];
  __shared__ float vs[ATTN_KTILE * DMAX];

  const int n = blockIdx.x;
  const int qi = blockIdx.y * ATTN_BLOCK + threadIdx.x;

  float qreg[DMAX], dreg[DMAX], acc[DMAX];
  float delta = 0.f, lse_q = 0.f;
  if (qi < Q) {
    const T* qp = q + ((size_t)n * Q + qi) * D;
    const T* dp = dout + ((size_t)n * Q + qi) * Dv;
    const T* op = out + ((size_t)n * Q + qi) * Dv;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) {
      qreg[d] = (d < D) ? ldf(qp + d) * scale : 0.f;
      dreg[d] = (d < Dv) ? ldf(dp + d) : 0.f;
      acc[d] = 0.f;
      if (d < Dv) delta += dreg[d] * ldf(op + d);
    }
    lse_q = lse[(size_t)n * Q + qi];
  }

  for (int k0 = 0; k0 < K; k0 += ATTN_KTILE) {
    const int kt = min(ATTN_KTILE, K - k0);
    for (int i = threadIdx.x; i < kt * D; i += ATTN_BLOCK)
      ks[(i / D) * DMAX + (i % D)] = ldf(k + ((size_t)n * K + k0) * D + i);
    for (int i = threadIdx.x; i < kt * Dv; i += ATTN_BLOCK)
      vs[(i / Dv) * DMAX + (i % Dv)] = ldf(v + ((size_t)n * K + k0) * Dv + i);
    __syncthreads();

    if (qi < Q) {
      for (int c = 0; c < kt; ++c) {
        float s = 0.f, g = 0.f;
        #pragma unroll
        for (int d = 0; d < DMAX; ++d) {
          if (d < D) s += qreg[d] * ks[c * DMAX + d];
          if (d < Dv) g += dreg[d] * vs[c * DMAX + d];
        }
        const float w = __expf(s - lse_q);
        const float f = w * (g - delta);
        #pragma unroll
        for (int d = 0; d < DMAX; ++d)
          if (d < D) acc[d] += f * ks[c * DMAX + d];
      }
    }
    __syncthreads();
  }

  if (qi < Q) {
    T* gp = dq + ((size_t)n * Q + qi) * D;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d)
      if (d < D) stf(gp + d, acc[d] * scale);  // dlogits/dq = scale * k
  }
}

// ---------------------------------------------------------------------------
// backward: dk, dv (thread-per-key, LDS-staged query tiles)
//   dv[c,d] = sum_q w_qc dout[q,d]
//   dk[c,d] = scale * sum_q w_qc (g_qc - delta_q) q[q,d]
// ---------------------------------------------------------------------------

template <typename T, int DMAX>
__global__ void __launch_bounds__(ATTN_BLOCK) npf_attn_bwd_dkv(
    const T* __restrict__ q, const T* __restrict__ k, const T* __restrict__ v,
    const T* __restrict__ out, const float* __restrict__ lse,
    const T* __restrict__ dout, T* __restrict__ dk, T* __restrict__ dv,
    int N, int Q, int K, int D, int Dv, float scale) {
  __shared__ float qs[ATTN_QTILE * DMAX];
  __shared__ float ds[ATTN_QTILE * DMAX];
  __shared__ float ls[ATTN_QTILE];
  __shared__ float dl[ATTN_QTILE];

  const int n = blockIdx.x;
  const int ki = blockIdx.y * ATTN_BLOCK + threadIdx.x;

  float kreg[DMAX], vreg[DMAX], dk_acc[DMAX], dv_acc[DMAX];
  if (ki < K) {
    const T* kp = k + ((size_t)n * K + ki) * D;
    const T* vp = v + ((size_t)n * K + ki) * Dv;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) {
      kreg[d] = (d < D) ? ldf(kp + d) * scale : 0.f;
      vreg[d] = (d < Dv) ? ldf(vp + d) : 0.f;
      dk_acc[d] = 0.f;
      dv_acc[d] = 0.f;
    }
  }

  for (int q0 = 0; q0 < Q; q0 += ATTN_QTILE) {
    const int qt = min(ATTN_QTILE, Q - q0);
    for (int i = threadIdx.x; i < qt * D; i += ATTN_BLOCK)
      qs[(i / D) * DMAX + (i % D)] = ldf(q + ((size_t)n * Q + q0) * D + i);
    for (int i = threadIdx.x; i < qt * Dv; i += ATTN_BLOCK)
      ds[(i / Dv) * DMAX + (i % Dv)] = ldf(dout + ((size_t)n * Q + q0) * Dv + i);
    for (int i = threadIdx.x; i < qt; i += ATTN_BLOCK) {
      ls[i] = lse[(size_t)n * Q + q0 + i];
      const T* dp = dout + ((size_t)n * Q + q0 + i) * Dv;
      const T* op = out + ((size_t)n * Q + q0 + i) * Dv;
      float dd = 0.f;
      for (int d = 0; d < Dv; ++d) dd += ldf(dp + d) * ldf(op + d);
      dl[i] = dd;
    }
    __syncthreads();

    if (ki < K) {
      for (int qq = 0; qq < qt; ++qq) {
        float s = 0.f, g = 0.f;
        #pragma unroll
        for (int d = 0; d < DMAX; ++d) {
          if (d < D) s += kreg[d] * qs[qq * DMAX + d];
          if (d < Dv) g += vreg[d] * ds[qq * DMAX + d];
        }
        const float w = __expf(s - ls[qq]);
        const float f = w * (g - dl[qq]) * scale;
        #pragma unroll
        for (int d = 0; d < DMAX; ++d) {
          if (d < D) dk_acc[d] += f * qs[qq * DMAX + d];
          if (d < Dv) dv_acc[d] += w * ds[qq * DMAX + d];
        }
      }
    }
    __syncthreads();
  }

  if (ki < K) {
    T* kp = dk + ((size_t)n * K + ki) * D;
    T* vp = dv + ((size_t)n * K + ki) * Dv;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) {
      if (d < D) stf(kp + d, dk_acc[d]);
      if (d < Dv) stf(vp + d, dv_acc[d]);
    }
  }
}

// ---------------------------------------------------------------------------
// launchers (extern "C" entry points used by ext.cpp)
// ---------------------------------------------------------------------------

template <typename T>
static void attn_fwd_launch_t(const void* q, const void* k, const void* v,
                              void* out, float* lse, int N, int Q, int K,
                              int D, int Dv, float scale, hipStream_t stream) {
  dim3 grid(N, (Q + ATTN_BLOCK - 1) / ATTN_BLOCK);
  if (D <= 16 && Dv <= 16) {
    hipLaunchKernelGGL((npf_attn_fwd_tpq<T, 16>), grid, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v, (T*)out,
                       lse, N, Q, K, D, Dv, scale);
  } else {
    hipLaunchKernelGGL((npf_attn_fwd_tpq<T, 32>), grid, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v, (T*)out,
                       lse, N, Q, K, D, Dv, scale);
  }
}

template <typename T>
static void attn_bwd_launch_t(const void* q, const void* k, const void* v,
                              const void* out, const float* lse,
                              const void* dout, void* dq, void* dk, void* dv,
                              int N, int Q, int K, int D, int Dv, float scale,
                              hipStream_t stream) {
  dim3 gq(N, (Q + ATTN_BLOCK - 1) / ATTN_BLOCK);
  dim3 gk(N, (K + ATTN_BLOCK - 1) / ATTN_BLOCK);
  if (D <= 16 && Dv <= 16) {
    hipLaunchKernelGGL((npf_attn_bwd_dq<T, 16>), gq, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v,
                       (const T*)out, lse, (const T*)dout, (T*)dq, N, Q, K, D,
                       Dv, scale);
    hipLaunchKernelGGL((npf_attn_bwd_dkv<T, 16>), gk, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v,
                       (const T*)out, lse, (const T*)dout, (T*)dk, (T*)dv, N,
                       Q, K, D, Dv, scale);
  } else {
    hipLaunchKernelGGL((npf_attn_bwd_dq<T, 32>), gq, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v,
                       (const T*)out, lse, (const T*)dout, (T*)dq, N, Q, K, D,
                       Dv, scale);
    hipLaunchKernelGGL((npf_attn_bwd_dkv<T, 32>), gk, dim3(ATTN_BLOCK), 0,
                       stream, (const T*)q, (const T*)k, (const T*)v,
                       (const T*)out, lse, (const T*)dout, (T*)dk, (T*)dv, N,
                       Q, K, D, Dv, scale);
  }
}

extern "C" void npf_attn_fwd_launch_f32(const void* q, const void* k,
                                        const void* v, void* out, float* lse,
                                        int N, int Q, int K, int D, int Dv,
                                        float scale, hipStream_t stream) {
  attn_fwd_launch_t<float>(q, k, v, out, lse, N, Q, K, D, Dv, scale, stream);
}
extern "C" void npf_attn_fwd_launch_bf16(const void* q, const void* k,
                                         const void* v, void* out, float* lse,
                                         int N, int Q, int K, int D, int Dv,
                                         float scale, hipStream_t stream) {
  attn_fwd_launch_t<__hip_bfloat16>(q, k, v, out, lse, N, Q, K, D, Dv, scale,
                                    stream);
}
extern "C" void npf_attn_bwd_launch_f32(const void* q, const void* k,
                                        const void* v, const void* out,
                                        const float* lse, const void* dout,
                                        void* dq, void* dk, void* dv, int N,
                                        int Q, int K, int D, int Dv,
                                        float scale, hipStream_t stream) {
  attn_bwd_launch_t<float>(q, k, v, out, lse, dout, dq, dk, dv, N, Q, K, D,
                           Dv, scale, stream);
}
extern "C" void npf_attn_bwd_launch_bf16(const void* q, const void* k,
                                         const void* v, const void* out,
                                         const float* lse, const void* dout,
                                         void* dq, void* dk, void* dv, int N,
                                         int Q, int K, int D, int Dv,
                                         float scale, hipStream_t stream) {
  attn_bwd_launch_t<__hip_bfloat16>(q, k, v, out, lse, dout, dq, dk, dv, N, Q,
                                    K, D, Dv, scale, stream);
}
