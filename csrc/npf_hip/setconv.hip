// Fused Gaussian SetConv (grid & ungrid directions share this kernel).
//
//   s_qk    = -((x_k - x_q)/sigma)^2
//   w_qk    = softmax_k(s_qk)                       (density-normalized RBF)
//   out[q]  = [ sum_k w_qk * v_k  ;  density_q = sum_k exp(s_qk) ]
//
// This is the ConvCNP/ConvLNP hot loop (SURVEY.md §2.3 rows "SetConv
// grid/ungrid"; reference setcnn.py:234-268 + ExpRBF :126-142).  The
// reference materializes the [B, Q, K, 1] pairwise tensor on HBM; here the
// scores live in LDS only (one [QT, K] stripe per workgroup) and each output
// is produced in a single kernel.
//
// Layout: one workgroup = (batch b, tile of QT=4 queries), 256 threads =
// 4 waves; wave w owns query (tile*QT + w).  Phase 1 computes the per-query
// score row (lanes stride over keys, shuffle-allreduce for max/sum).
// Phase 2 computes the value reduction with threads striding over (q, c)
// pairs (consecutive threads -> consecutive channels: coalesced V reads, LDS
// broadcast on the weights).  Backward uses the same staging with two more
// LDS stripes and atomics on the shared dk / dv / dsigma.
//
// x_dim == 1 (the SetConv contract, reference setcnn.py:226); fp32 only —
// positions at sigma ~ 4e-3 need fp32 resolution, values are cast by the
// python wrapper (npf/ops/functional.py).

#include "common.h"

#define SC_BLOCK 256
#define SC_QT 4          // queries per workgroup (= waves per block)
#define SC_MAX_K 8192    // wrapper falls back above this (LDS stripe cap)
#define NEG_INF (-1e30f)

// dynamic smem layout (floats):
//   w   [SC_QT][K]          scores -> exp(s - m)
//   g   [SC_QT][K]          (backward only) dot products -> dL/ds
//   mq  [SC_QT], lq [SC_QT], xq [SC_QT], hq [SC_QT], dD [SC_QT]
//   red [16]

extern "C" __global__ void __launch_bounds__(SC_BLOCK)
npf_setconv_fwd(const float* __restrict__ xk, const float* __restrict__ xq,
                const float* __restrict__ v, float* __restrict__ out,
                int B, int K, int Q, int C,
                const float* __restrict__ sigma_ptr) {
  const float sigma = *sigma_ptr;
  extern __shared__ float smem[];
  float* w = smem;                       // [SC_QT][K]
  float* mq = smem + SC_QT * K;          // [SC_QT]
  float* lq = mq + SC_QT;                // [SC_QT]

  const int b = blockIdx.x;
  const int q0 = blockIdx.y * SC_QT;
  const int wid = threadIdx.x / NPF_WAVE;
  const int lane = threadIdx.x & (NPF_WAVE - 1);
  const int qi = q0 + wid;
  const float inv_s2 = 1.f / (sigma * sigma);

  // ---- phase 1: score rows, one wave per query ----
  if (qi < Q) {
    const float x_q = xq[(size_t)b * Q + qi];
    float m = NEG_INF;
    for (int c = lane; c < K; c += NPF_WAVE) {
      const float u = xk[(size_t)b * K + c] - x_q;
      const float s = -u * u * inv_s2;
      w[wid * K + c] = s;
      m = fmaxf(m, s);
    }
    m = wave_allreduce_max(m);
    float l = 0.f;
    for (int c = lane; c < K; c += NPF_WAVE) {
      const float p = __expf(w[wid * K + c] - m);
      w[wid * K + c] = p;
      l += p;
    }
    l = wave_allreduce_sum(l);
    if (lane == 0) {
      mq[wid] = m;
      lq[wid] = l;
    }
  }
  __syncthreads();

  // ---- phase 2: value reduction, threads stride (q, c) ----
  const int qt = min(SC_QT, Q - q0);
  for (int idx = threadIdx.x; idx < qt * C; idx += SC_BLOCK) {
    const int qq = idx / C;
    const int cc = idx % C;
    const float* vrow = v + (size_t)b * K * C + cc;
    float acc = 0.f;
    for (int c = 0; c < K; ++c) acc += w[qq * K + c] * vrow[(size_t)c * C];
    out[((size_t)b * Q + q0 + qq) * (C + 1) + cc] = acc / lq[qq];
  }
  // density channel
  for (int qq = threadIdx.x; qq < qt; qq += SC_BLOCK)
    out[((size_t)b * Q + q0 + qq) * (C + 1) + C] = __expf(mq[qq]) * lq[qq];
}

extern "C" __global__ void __launch_bounds__(SC_BLOCK)
npf_setconv_bwd(const float* __restrict__ xk, const float* __restrict__ xq,
                const float* __restrict__ v, const float* __restrict__ dout,
                float* __restrict__ dxk, float* __restrict__ dxq,
                float* __restrict__ dv, float* __restrict__ dsigma,
                int B, int K, int Q, int C,
                const float* __restrict__ sigma_ptr) {
  const float sigma = *sigma_ptr;
  extern __shared__ float smem[];
  float* w = smem;                        // [SC_QT][K] normalized weights
  float* g = smem + SC_QT * K;            // [SC_QT][K] dots -> dL/ds
  float* mq = g + SC_QT * K;
  float* lq = mq + SC_QT;
  float* xqs = lq + SC_QT;
  float* hq = xqs + SC_QT;
  float* dD = hq + SC_QT;
  float* red = dD + SC_QT;                // [16]

  const int b = blockIdx.x;
  const int q0 = blockIdx.y * SC_QT;
  const int wid = threadIdx.x / NPF_WAVE;
  const int lane = threadIdx.x & (NPF_WAVE - 1);
  const int qi = q0 + wid;
  const int qt = min(SC_QT, Q - q0);
  const float inv_s2 = 1.f / (sigma * sigma);

  // ---- pass 1: recompute normalized weights ----
  if (qi < Q) {
    const float x_q = xq[(size_t)b * Q + qi];
    float m = NEG_INF;
    for (int c = lane; c < K; c += NPF_WAVE) {
      const float u = xk[(size_t)b * K + c] - x_q;
      const float s = -u * u * inv_s2;
      w[wid * K + c] = s;
      m = fmaxf(m, s);
    }
    m = wave_allreduce_max(m);
    float l = 0.f;
    for (int c = lane; c < K; c += NPF_WAVE) {
      const float p = __expf(w[wid * K + c] - m);
      w[wid * K + c] = p;
      l += p;
    }
    l = wave_allreduce_sum(l);
    const float inv_l = 1.f / l;
    for (int c = lane; c < K; c += NPF_WAVE) w[wid * K + c] *= inv_l;
    if (lane == 0) {
      mq[wid] = m;
      lq[wid] = l;
      xqs[wid] = x_q;
      dD[wid] = dout[((size_t)b * Q + qi) * (C + 1) + C];
    }
  }
  __syncthreads();

  // ---- pass 2: g[q][k] = dout[q,:C] . v[k,:] ----
  for (int idx = threadIdx.x; idx < qt * K; idx += SC_BLOCK) {
    const int qq = idx / K;
    const int c = idx % K;
    const float* vrow = v + ((size_t)b * K + c) * C;
    const float* drow = dout + ((size_t)b * Q + q0 + qq) * (C + 1);
    float acc = 0.f;
    for (int ch = 0; ch < C; ++ch) acc += drow[ch] * vrow[ch];
    g[qq * K + c] = acc;
  }
  __syncthreads();

  // ---- pass 3: h_q = sum_k w g ----
  if (qi < Q) {
    float h = 0.f;
    for (int c = lane; c < K; c += NPF_WAVE) h += w[wid * K + c] * g[wid * K + c];
    h = wave_allreduce_sum(h);
    if (lane == 0) hq[wid] = h;
  }
  __syncthreads();

  // ---- pass 4: dL/ds; accumulate dxk (atomic), dsigma (block-reduced) ----
  float dsig_part = 0.f;
  for (int idx = threadIdx.x; idx < qt * K; idx += SC_BLOCK) {
    const int qq = idx / K;
    const int c = idx % K;
    const float ww = w[qq * K + c];
    // d density / d s = exp(s) = w * l * exp(m)
    const float dLds =
        ww * (g[qq * K + c] - hq[qq] + dD[qq] * lq[qq] * __expf(mq[qq]));
    g[qq * K + c] = dLds;
    const float u = xk[(size_t)b * K + c] - xqs[qq];
    const float du = dLds * 2.f * u * inv_s2;  // = dL/d(x_q); dL/d(x_k) = -du
    atomicAdd(&dxk[(size_t)b * K + c], -du);
    dsig_part += dLds * 2.f * u * u * inv_s2 / sigma;
  }
  dsig_part = block_reduce_sum(dsig_part, red);
  if (threadIdx.x == 0) atomicAdd(dsigma, dsig_part);
  __syncthreads();

  // ---- pass 5: dxq (wave per query) ----
  if (qi < Q) {
    const float x_q = xqs[wid];
    float acc = 0.f;
    for (int c = lane; c < K; c += NPF_WAVE) {
      const float u = xk[(size_t)b * K + c] - x_q;
      acc += g[wid * K + c] * 2.f * u * inv_s2;
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) dxq[(size_t)b * Q + qi] = acc;
  }

  // ---- pass 6: dv[k,c] += sum_{q in tile} w dout ----
  for (int idx = threadIdx.x; idx < K * C; idx += SC_BLOCK) {
    const int c = idx / C;
    const int ch = idx % C;
    float acc = 0.f;
    for (int qq = 0; qq < qt; ++qq)
      acc += w[qq * K + c] * dout[((size_t)b * Q + q0 + qq) * (C + 1) + ch];
    atomicAdd(&dv[((size_t)b * K + c) * C + ch], acc);
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

extern "C" void npf_setconv_fwd_launch(const float* xk, const float* xq,
                                       const float* v, float* out, int B,
                                       int K, int Q, int C, const float* sigma,
                                       hipStream_t stream) {
  dim3 grid(B, (Q + SC_QT - 1) / SC_QT);
  const size_t smem = (size_t)(SC_QT * K + 2 * SC_QT) * sizeof(float);
  hipLaunchKernelGGL(npf_setconv_fwd, grid, dim3(SC_BLOCK), smem, stream, xk,
                     xq, v, out, B, K, Q, C, sigma);
}

extern "C" void npf_setconv_bwd_launch(const float* xk, const float* xq,
                                       const float* v, const float* dout,
                                       float* dxk, float* dxq, float* dv,
                                       float* dsigma, int B, int K, int Q,
                                       int C, const float* sigma,
                                       hipStream_t stream) {
  dim3 grid(B, (Q + SC_QT - 1) / SC_QT);
  const size_t smem =
      (size_t)(2 * SC_QT * K + 5 * SC_QT + 16) * sizeof(float);
  hipLaunchKernelGGL(npf_setconv_bwd, grid, dim3(SC_BLOCK), smem, stream, xk,
                     xq, v, dout, dxk, dxq, dv, dsigma, B, K, Q, C, sigma);
}
