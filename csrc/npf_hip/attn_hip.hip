#include "hip/hip_runtime.h"
// Fused scaled-dot cross-attention for the NP shape regime:
//   out[n,q,:] = softmax_k(scale * Q[n,q,:] . K[n,k,:]) @ V[n,k,:]
//
// Shape regime (SURVEY.md §2.3 "Cross-attention"): per-head dims D, Dv <= 32
// (the shipped configs use 128-dim / 8 heads = 16), keys = contexts (small,
// 10..1300), queries = targets (large, 128..4096), batch = B*heads (~256).
//
// Design (MI355X/CDNA4):
// - THREAD-per-query online softmax; K/V staged in LDS once per block and
//   read as float4 (ds_read_b128, broadcast across lanes -> conflict-free);
//   the [Q,K] score matrix never exists in HBM; per-row logsumexp is written
//   for the backward recompute (flash-style).
// - The work per thread is tiny (K*D fused-mul-adds), so the limiter is the
//   LATENCY of LDS-dependent chains, not FLOPs: all inner loops use float4
//   loads + split accumulators (2-way ILP; the forward runs TWO independent
//   online softmaxes over even/odd keys and merges them at the end).
// - backward: dq thread-per-query (w recomputed from lse); dk/dv
//   thread-per-key over LDS-staged query tiles.
//
// Not an MFMA problem: at C~50 the whole K/V set is ~6 KB and MFMA tiles
// would be >90% padding (guide §5: stage only what pays).

#include "common.h"

#define ATTN_BLOCK 256
#define ATTN_KTILE 128   // keys staged per LDS round (fwd / dq)
#define ATTN_QTILE 64    // queries staged per LDS round (bwd dkv)
#define NEG_INF (-1e30f)

// stage a row-major [rows, width] fp32/bf16 global block into LDS as fp32
// with row stride `DMAX` floats (width <= DMAX); cooperative, coalesced.
template <typename T, int DMAX>
__device__ __forceinline__ void stage_tile(float* dst, const T* src, int rows,
                                           int width) {
  for (int i = threadIdx.x; i < rows * width; i += ATTN_BLOCK)
    dst[(i / width) * DMAX + (i % width)] = ldf(src + i);
}

template <int DMAX>
__device__ __forceinline__ float dot_f4(const float4* a, const float4* b) {
  float s = 0.f;
  #pragma unroll
  for (int j = 0; j < DMAX / 4; ++j) {
    const float4 x = a[j], y = b[j];
    s += x.x * y.x + x.y * y.y + x.z * y.z + x.w * y.w;
  }
  return s;
}

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

  float4 qreg[DMAX / 4];
  // two independent online-softmax states (even / odd keys) for ILP
  float acc0[DMAX], acc1[DMAX];
  if (qi < Q) {
    const T* qp = q + ((size_t)n * Q + qi) * D;
    float tmp[DMAX];
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) tmp[d] = (d < D) ? ldf(qp + d) * scale : 0.f;
    #pragma unroll
    for (int j = 0; j < DMAX / 4; ++j)
      qreg[j] = make_float4(tmp[4 * j], tmp[4 * j + 1], tmp[4 * j + 2],
                            tmp[4 * j + 3]);
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) acc0[d] = acc1[d] = 0.f;
  }
  float m0 = NEG_INF, l0 = 0.f, m1 = NEG_INF, l1 = 0.f;

  const float4* ks4 = reinterpret_cast<const float4*>(ks);
  const float4* vs4 = reinterpret_cast<const float4*>(vs);

  for (int k0 = 0; k0 < K; k0 += ATTN_KTILE) {
    const int kt = min(ATTN_KTILE, K - k0);
    stage_tile<T, DMAX>(ks, k + ((size_t)n * K + k0) * D, kt, D);
    stage_tile<T, DMAX>(vs, v + ((size_t)n * K + k0) * Dv, kt, Dv);
    __syncthreads();

    if (qi < Q) {
      int c = 0;
      for (; c + 1 < kt; c += 2) {
        const float s0 = dot_f4<DMAX>(qreg, ks4 + c * (DMAX / 4));
        const float s1 = dot_f4<DMAX>(qreg, ks4 + (c + 1) * (DMAX / 4));
        // branchless online update, independent chains
        const float n0 = fmaxf(m0, s0), n1 = fmaxf(m1, s1);
        const float r0 = __expf(m0 - n0), r1 = __expf(m1 - n1);
        const float w0 = __expf(s0 - n0), w1 = __expf(s1 - n1);
        l0 = l0 * r0 + w0;
        l1 = l1 * r1 + w1;
        m0 = n0;
        m1 = n1;
        const float4* v0 = vs4 + c * (DMAX / 4);
        const float4* v1 = vs4 + (c + 1) * (DMAX / 4);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = v0[j], b = v1[j];
          acc0[4 * j] = acc0[4 * j] * r0 + w0 * a.x;
          acc0[4 * j + 1] = acc0[4 * j + 1] * r0 + w0 * a.y;
          acc0[4 * j + 2] = acc0[4 * j + 2] * r0 + w0 * a.z;
          acc0[4 * j + 3] = acc0[4 * j + 3] * r0 + w0 * a.w;
          acc1[4 * j] = acc1[4 * j] * r1 + w1 * b.x;
          acc1[4 * j + 1] = acc1[4 * j + 1] * r1 + w1 * b.y;
          acc1[4 * j + 2] = acc1[4 * j + 2] * r1 + w1 * b.z;
          acc1[4 * j + 3] = acc1[4 * j + 3] * r1 + w1 * b.w;
        }
      }
      if (c < kt) {  // odd tail into state 0
        const float s0 = dot_f4<DMAX>(qreg, ks4 + c * (DMAX / 4));
        const float n0 = fmaxf(m0, s0);
        const float r0 = __expf(m0 - n0), w0 = __expf(s0 - n0);
        l0 = l0 * r0 + w0;
        m0 = n0;
        const float4* v0 = vs4 + c * (DMAX / 4);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = v0[j];
          acc0[4 * j] = acc0[4 * j] * r0 + w0 * a.x;
          acc0[4 * j + 1] = acc0[4 * j + 1] * r0 + w0 * a.y;
          acc0[4 * j + 2] = acc0[4 * j + 2] * r0 + w0 * a.z;
          acc0[4 * j + 3] = acc0[4 * j + 3] * r0 + w0 * a.w;
        }
      }
    }
    __syncthreads();
  }

  if (qi < Q) {
    // merge the two online states
    const float m = fmaxf(m0, m1);
    const float r0 = __expf(m0 - m), r1 = __expf(m1 - m);
    const float l = l0 * r0 + l1 * r1;
    const float inv_l = 1.f / l;
    T* op = out + ((size_t)n * Q + qi) * Dv;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d)
      if (d < Dv) stf(op + d, (acc0[d] * r0 + acc1[d] * r1) * inv_l);
    lse[(size_t)n * Q + qi] = m + __logf(l);
  }
}

// ---------------------------------------------------------------------------
// backward: dq (thread-per-query)
//   dq[q,d] = scale * sum_c w_qc (g_qc - delta_q) k[c,d]
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

  float4 qreg[DMAX / 4], dreg[DMAX / 4];
  float acc0[DMAX], acc1[DMAX];
  float delta = 0.f, lse_q = 0.f;
  if (qi < Q) {
    const T* qp = q + ((size_t)n * Q + qi) * D;
    const T* dp = dout + ((size_t)n * Q + qi) * Dv;
    const T* op = out + ((size_t)n * Q + qi) * Dv;
    float tq[DMAX], td[DMAX];
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) {
      tq[d] = (d < D) ? ldf(qp + d) * scale : 0.f;
      td[d] = (d < Dv) ? ldf(dp + d) : 0.f;
      if (d < Dv) delta += td[d] * ldf(op + d);
      acc0[d] = acc1[d] = 0.f;
    }
    #pragma unroll
    for (int j = 0; j < DMAX / 4; ++j) {
      qreg[j] = make_float4(tq[4 * j], tq[4 * j + 1], tq[4 * j + 2], tq[4 * j + 3]);
      dreg[j] = make_float4(td[4 * j], td[4 * j + 1], td[4 * j + 2], td[4 * j + 3]);
    }
    lse_q = lse[(size_t)n * Q + qi];
  }

  const float4* ks4 = reinterpret_cast<const float4*>(ks);
  const float4* vs4 = reinterpret_cast<const float4*>(vs);

  for (int k0 = 0; k0 < K; k0 += ATTN_KTILE) {
    const int kt = min(ATTN_KTILE, K - k0);
    stage_tile<T, DMAX>(ks, k + ((size_t)n * K + k0) * D, kt, D);
    stage_tile<T, DMAX>(vs, v + ((size_t)n * K + k0) * Dv, kt, Dv);
    __syncthreads();

    if (qi < Q) {
      int c = 0;
      for (; c + 1 < kt; c += 2) {
        const float4* k0p = ks4 + c * (DMAX / 4);
        const float4* k1p = ks4 + (c + 1) * (DMAX / 4);
        const float f0 =
            __expf(dot_f4<DMAX>(qreg, k0p) - lse_q) *
            (dot_f4<DMAX>(dreg, vs4 + c * (DMAX / 4)) - delta);
        const float f1 =
            __expf(dot_f4<DMAX>(qreg, k1p) - lse_q) *
            (dot_f4<DMAX>(dreg, vs4 + (c + 1) * (DMAX / 4)) - delta);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = k0p[j], b = k1p[j];
          acc0[4 * j] += f0 * a.x;
          acc0[4 * j + 1] += f0 * a.y;
          acc0[4 * j + 2] += f0 * a.z;
          acc0[4 * j + 3] += f0 * a.w;
          acc1[4 * j] += f1 * b.x;
          acc1[4 * j + 1] += f1 * b.y;
          acc1[4 * j + 2] += f1 * b.z;
          acc1[4 * j + 3] += f1 * b.w;
        }
      }
      if (c < kt) {
        const float4* k0p = ks4 + c * (DMAX / 4);
        const float f0 =
            __expf(dot_f4<DMAX>(qreg, k0p) - lse_q) *
            (dot_f4<DMAX>(dreg, vs4 + c * (DMAX / 4)) - delta);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = k0p[j];
          acc0[4 * j] += f0 * a.x;
          acc0[4 * j + 1] += f0 * a.y;
          acc0[4 * j + 2] += f0 * a.z;
          acc0[4 * j + 3] += f0 * a.w;
        }
      }
    }
    __syncthreads();
  }

  if (qi < Q) {
    T* gp = dq + ((size_t)n * Q + qi) * D;
    #pragma unroll
    for (int d = 0; d < DMAX; ++d)
      if (d < D) stf(gp + d, (acc0[d] + acc1[d]) * scale);  // dlogits/dq = scale*k
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

  float4 kreg[DMAX / 4], vreg[DMAX / 4];
  float dk0[DMAX], dk1[DMAX], dv0[DMAX], dv1[DMAX];
  if (ki < K) {
    const T* kp = k + ((size_t)n * K + ki) * D;
    const T* vp = v + ((size_t)n * K + ki) * Dv;
    float tk[DMAX], tv[DMAX];
    #pragma unroll
    for (int d = 0; d < DMAX; ++d) {
      tk[d] = (d < D) ? ldf(kp + d) * scale : 0.f;
      tv[d] = (d < Dv) ? ldf(vp + d) : 0.f;
      dk0[d] = dk1[d] = dv0[d] = dv1[d] = 0.f;
    }
    #pragma unroll
    for (int j = 0; j < DMAX / 4; ++j) {
      kreg[j] = make_float4(tk[4 * j], tk[4 * j + 1], tk[4 * j + 2], tk[4 * j + 3]);
      vreg[j] = make_float4(tv[4 * j], tv[4 * j + 1], tv[4 * j + 2], tv[4 * j + 3]);
    }
  }

  const float4* qs4 = reinterpret_cast<const float4*>(qs);
  const float4* ds4 = reinterpret_cast<const float4*>(ds);

  for (int q0 = 0; q0 < Q; q0 += ATTN_QTILE) {
    const int qt = min(ATTN_QTILE, Q - q0);
    stage_tile<T, DMAX>(qs, q + ((size_t)n * Q + q0) * D, qt, D);
    stage_tile<T, DMAX>(ds, dout + ((size_t)n * Q + q0) * Dv, qt, Dv);
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
      int qq = 0;
      for (; qq + 1 < qt; qq += 2) {
        const float4* q0p = qs4 + qq * (DMAX / 4);
        const float4* q1p = qs4 + (qq + 1) * (DMAX / 4);
        const float4* d0p = ds4 + qq * (DMAX / 4);
        const float4* d1p = ds4 + (qq + 1) * (DMAX / 4);
        const float w0 = __expf(dot_f4<DMAX>(kreg, q0p) - ls[qq]);
        const float w1 = __expf(dot_f4<DMAX>(kreg, q1p) - ls[qq + 1]);
        const float f0 = w0 * (dot_f4<DMAX>(vreg, d0p) - dl[qq]);
        const float f1 = w1 * (dot_f4<DMAX>(vreg, d1p) - dl[qq + 1]);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = q0p[j], b = q1p[j];
          const float4 x = d0p[j], y = d1p[j];
          dk0[4 * j] += f0 * a.x;
          dk0[4 * j + 1] += f0 * a.y;
          dk0[4 * j + 2] += f0 * a.z;
          dk0[4 * j + 3] += f0 * a.w;
          dk1[4 * j] += f1 * b.x;
          dk1[4 * j + 1] += f1 * b.y;
          dk1[4 * j + 2] += f1 * b.z;
          dk1[4 * j + 3] += f1 * b.w;
          dv0[4 * j] += w0 * x.x;
          dv0[4 * j + 1] += w0 * x.y;
          dv0[4 * j + 2] += w0 * x.z;
          dv0[4 * j + 3] += w0 * x.w;
          dv1[4 * j] += w1 * y.x;
          dv1[4 * j + 1] += w1 * y.y;
          dv1[4 * j + 2] += w1 * y.z;
          dv1[4 * j + 3] += w1 * y.w;
        }
      }
      if (qq < qt) {
        const float4* q0p = qs4 + qq * (DMAX / 4);
        const float4* d0p = ds4 + qq * (DMAX / 4);
        const float w0 = __expf(dot_f4<DMAX>(kreg, q0p) - ls[qq]);
        const float f0 = w0 * (dot_f4<DMAX>(vreg, d0p) - dl[qq]);
        #pragma unroll
        for (int j = 0; j < DMAX / 4; ++j) {
          const float4 a = q0p[j], x = d0p[j];
          dk0[4 * j] += f0 * a.x;
          dk0[4 * j + 1] += f0 * a.y;
          dk0[4 * j + 2] += f0 * a.z;
          dk0[4 * j + 3] += f0 * a.w;
          dv0[4 * j] += w0 * x.x;
          dv0[4 * j + 1] += w0 * x.y;
          dv0[4 * j + 2] += w0 * x.z;
          dv0[4 * j + 3] += w0 * x.w;
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
      if (d < D) stf(kp + d, (dk0[d] + dk1[d]) * scale);  // dlogits/dk = scale*q
      if (d < Dv) stf(vp + d, dv0[d] + dv1[d]);
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
