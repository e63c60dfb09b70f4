// Fused pre-activation depthwise conv block (2D) — the GridConvCNP/LNP CNN
// hot loop (reference cnn.py:204-215 with Conv2d/BatchNorm2d, k=9, on
// [N, C, H, W] = [Z*B, 128, 32..64, 32..64]).
//
//   a = relu(batchnorm(x));  y = dwconv2d(a) (+bias) (+res)
//
// Same structure as the 1D kernels (convblock.hip): the per-channel stats
// and the elementwise BN-backward kernel are shared with 1D (L = H*W);
// only the two stencil kernels differ.  One workgroup per (n, c) plane,
// 256 threads striding H*W; the activation plane is staged in LDS with its
// (k/2)-halo ((64+8)^2 floats = 20 KB of the 160 KB LDS), and the k*k
// weights live in LDS (k^2 register accumulators would blow VGPR budget);
// dW is accumulated as one block-reduction per weight position, reading
// both operands from LDS.

#include "common.h"

#define CB2_BLOCK 256
#define CB2_MAX_K 13
#define CB2_TR 8   // plane rows per workgroup: small LDS tiles -> enough
                   // resident waves to hide LDS latency (a full 64x64 plane
                   // per WG was 41.5 KB -> 3 waves/SIMD -> 8.7ms bwd; row
                   // tiles re-read the k/2 halo rows but run latency-hidden)

template <typename T, int KT>
__global__ void __launch_bounds__(CB2_BLOCK)
npf_cb2d_fwd(const T* __restrict__ x, const T* __restrict__ res,
             const float* __restrict__ w, const float* __restrict__ bias,
             const float* __restrict__ gamma, const float* __restrict__ beta,
             const float* __restrict__ mean, const float* __restrict__ rstd,
             T* __restrict__ y, int N, int C, int H, int W, int K_rt) {
  const int K = (KT > 0) ? KT : K_rt;
  extern __shared__ float smem[];
  const int pad = K / 2;
  const int TRP = CB2_TR + 2 * pad, WP = W + 2 * pad;
  float* a = smem;             // [TRP * WP] activation row-tile with halo
  float* ws = smem + TRP * WP; // [K * K]

  const int n = blockIdx.x / C;
  const int c = blockIdx.x % C;
  const int r0 = blockIdx.y * CB2_TR;
  const T* xpl = x + ((long)n * C + c) * H * W;

  const bool has_bn = gamma != nullptr;
  const float mu = has_bn ? mean[c] : 0.f;
  const float gscale = has_bn ? rstd[c] * gamma[c] : 1.f;
  const float gshift = has_bn ? beta[c] : 0.f;

  for (int i = threadIdx.x; i < K * K; i += CB2_BLOCK) ws[i] = w[c * K * K + i];
  for (int i = threadIdx.x; i < TRP * WP; i += CB2_BLOCK) {
    const int r = r0 + i / WP - pad, col = i % WP - pad;
    float v = 0.f;
    if (r >= 0 && r < H && col >= 0 && col < W)
      v = fmaxf((ldf(xpl + r * W + col) - mu) * gscale + gshift, 0.f);
    a[i] = v;
  }
  __syncthreads();

  const int tr = min(CB2_TR, H - r0);
  const float b = (bias != nullptr) ? bias[c] : 0.f;
  T* ypl = y + ((long)n * C + c) * H * W;
  const T* rpl =
      (res != nullptr) ? res + ((long)n * C + c) * H * W : nullptr;
  // div-free pixel walk: software integer div/mod per element was ~8x the
  // useful VALU work (SQ_INSTS_VALU 5075/wave vs ~650 needed)
  {
    int r = threadIdx.x / W, col = threadIdx.x % W;
    const int dr = CB2_BLOCK / W, dc = CB2_BLOCK % W;
    while (r < tr) {
      float acc = b;
      #pragma unroll
      for (int kr = 0; kr < K; ++kr) {
        const float* arow = a + (r + kr) * WP + col;
        #pragma unroll
        for (int kc = 0; kc < K; ++kc) acc += ws[kr * K + kc] * arow[kc];
      }
      const long gp = (long)(r0 + r) * W + col;
      if (rpl != nullptr) acc += ldf(rpl + gp);
      stf(ypl + gp, acc);
      r += dr;
      col += dc;
      if (col >= W) { col -= W; ++r; }
    }
  }
}

// backward stencil: dact (stored), dW/db block-reduced then global atomics,
// BN channel partial sums (global atomics)
template <typename T, int KT>
__global__ void __launch_bounds__(CB2_BLOCK)
npf_cb2d_bwd_dact(const T* __restrict__ x, const float* __restrict__ w,
                  const T* __restrict__ dy, const float* __restrict__ gamma,
                  const float* __restrict__ beta, const float* __restrict__ mean,
                  const float* __restrict__ rstd, T* __restrict__ dact,
                  float* __restrict__ dw, float* __restrict__ db,
                  float* __restrict__ sum_dxhat,
                  float* __restrict__ sum_dxhat_xhat,
                  float* __restrict__ dgamma, float* __restrict__ dbeta,
                  int N, int C, int H, int W, int K_rt) {
  const int K = (KT > 0) ? KT : K_rt;
  extern __shared__ float smem[];
  const int pad = K / 2;
  const int TRP = CB2_TR + 2 * pad, WP = W + 2 * pad;
  float* a = smem;                      // [TRP*WP] activations (halo)
  float* dys = smem + TRP * WP;         // [TRP*WP] dY (halo)
  float* ws = dys + TRP * WP;           // [K*K]
  __shared__ float red[16];

  const int n = blockIdx.x / C;
  const int c = blockIdx.x % C;
  const int r0 = blockIdx.y * CB2_TR;
  const T* xpl = x + ((long)n * C + c) * H * W;
  const T* dypl = dy + ((long)n * C + c) * H * W;

  const bool has_bn = gamma != nullptr;
  const float mu = has_bn ? mean[c] : 0.f;
  const float rs = has_bn ? rstd[c] : 1.f;
  const float gm = has_bn ? gamma[c] : 1.f;
  const float gscale = has_bn ? rs * gm : 1.f;
  const float gshift = has_bn ? beta[c] : 0.f;

  for (int i = threadIdx.x; i < K * K; i += CB2_BLOCK) ws[i] = w[c * K * K + i];
  for (int i = threadIdx.x; i < TRP * WP; i += CB2_BLOCK) {
    const int r = r0 + i / WP - pad, col = i % WP - pad;
    const bool in = (r >= 0 && r < H && col >= 0 && col < W);
    dys[i] = in ? ldf(dypl + r * W + col) : 0.f;
    float v = 0.f;
    if (in) v = fmaxf((ldf(xpl + r * W + col) - mu) * gscale + gshift, 0.f);
    a[i] = v;
  }
  __syncthreads();

  const int tr = min(CB2_TR, H - r0);
  T* dactpl = dact + ((long)n * C + c) * H * W;
  float s_dxhat = 0.f, s_dxhat_xhat = 0.f, s_dg = 0.f, s_db = 0.f, dbp = 0.f;
  {
    int r = threadIdx.x / W, col = threadIdx.x % W;
    const int drr = CB2_BLOCK / W, dcc = CB2_BLOCK % W;
    while (r < tr) {
      const long gp = (long)(r0 + r) * W + col;
      const float dyl = ldf(dypl + gp);
      dbp += dyl;
      // transposed conv: da[p] = sum_k w[k] dY[p + pad - k]
      float da = 0.f;
      #pragma unroll
      for (int kr = 0; kr < K; ++kr) {
        const float* drow = dys + (r + K - 1 - kr) * WP + col;
        #pragma unroll
        for (int kc = 0; kc < K; ++kc)
          da += ws[kr * K + kc] * drow[K - 1 - kc];
      }
      const float act = a[(r + pad) * WP + col + pad];
      const float dr = (act > 0.f) ? da : 0.f;
      stf(dactpl + gp, dr);
      if (has_bn) {
        const float xhat = (ldf(xpl + gp) - mu) * rs;
        const float dxh = dr * gm;
        s_dxhat += dxh;
        s_dxhat_xhat += dxh * xhat;
        s_dg += dr * xhat;
        s_db += dr;
      }
      r += drr;
      col += dcc;
      if (col >= W) { col -= W; ++r; }
    }
  }
  __syncthreads();
  // dW[kr,kc] = sum_p dY[p] * a[p + (kr,kc)]: K^2 positions split across
  // waves, shuffle-only reduction, one atomic per (row-tile, position).
  // Row/col loops instead of a strided pixel walk: the inner loop is pure
  // lane-strided adds (the walk's per-element index fixups were measured
  // as several x the stencil VALU work in PMC counters)
  {
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    const int nw = CB2_BLOCK / 64;
    for (int kk = wv; kk < K * K; kk += nw) {
      const int kr = kk / K, kc = kk % K;
      float psum = 0.f;
      for (int r = 0; r < tr; ++r) {
        const float* dyrow = dys + (r + pad) * WP + pad;
        const float* arow = a + (r + kr) * WP + kc;
        for (int col = lane; col < W; col += 64)
          psum += dyrow[col] * arow[col];
      }
      psum = wave_reduce_sum(psum);
      if (lane == 0) atomicAdd(&dw[c * K * K + kk], psum);
    }
  }
  {
    const float v = block_reduce_sum(dbp, red);
    __syncthreads();
    if (threadIdx.x == 0 && db != nullptr) atomicAdd(&db[c], v);
  }
  if (has_bn) {
    float v = block_reduce_sum(s_dxhat, red);
    __syncthreads();
    if (threadIdx.x == 0) atomicAdd(&sum_dxhat[c], v);
    v = block_reduce_sum(s_dxhat_xhat, red);
    __syncthreads();
    if (threadIdx.x == 0) atomicAdd(&sum_dxhat_xhat[c], v);
    v = block_reduce_sum(s_dg, red);
    __syncthreads();
    if (threadIdx.x == 0) atomicAdd(&dgamma[c], v);
    v = block_reduce_sum(s_db, red);
    __syncthreads();
    if (threadIdx.x == 0) atomicAdd(&dbeta[c], v);
  }
}

// ---------------------------------------------------------------------------
// host launchers (stats + bn-backward-elementwise reuse the 1D kernels with
// L = H*W, launched from ext.cpp)
// ---------------------------------------------------------------------------

template <typename T>
static void cb2d_fwd_dispatch(const void* x, const void* res, const float* w,
                              const float* bias, const float* gamma,
                              const float* beta, const float* mean,
                              const float* rstd, void* y, int N, int C, int H,
                              int W, int K, hipStream_t stream) {
  const int pad = K / 2;
  const size_t smem =
      ((size_t)(CB2_TR + 2 * pad) * (W + 2 * pad) + K * K) * sizeof(float);
  const dim3 grid((unsigned)N * C, (unsigned)((H + CB2_TR - 1) / CB2_TR)),
      blk(CB2_BLOCK);
#define CB2D_FWD(KT)                                                        \
  hipLaunchKernelGGL((npf_cb2d_fwd<T, KT>), grid, blk, smem, stream,        \
                     (const T*)x, (const T*)res, w, bias, gamma, beta,      \
                     mean, rstd, (T*)y, N, C, H, W, K)
  switch (K) {  // compile-time stencil: full unroll + hoisted addresses
    case 5: CB2D_FWD(5); break;
    case 9: CB2D_FWD(9); break;
    case 11: CB2D_FWD(11); break;
    default: CB2D_FWD(0);
  }
#undef CB2D_FWD
}

extern "C" void npf_cb2d_fwd_launch(const void* x, const void* res,
                                    const float* w, const float* bias,
                                    const float* gamma, const float* beta,
                                    const float* mean, const float* rstd,
                                    void* y, int N, int C, int H, int W,
                                    int K, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    cb2d_fwd_dispatch<__hip_bfloat16>(x, res, w, bias, gamma, beta, mean,
                                      rstd, y, N, C, H, W, K, stream);
  else
    cb2d_fwd_dispatch<float>(x, res, w, bias, gamma, beta, mean, rstd, y, N,
                             C, H, W, K, stream);
}

template <typename T>
static void cb2d_bwd_dispatch(const void* x, const float* w, const void* dy,
                              const float* gamma, const float* beta,
                              const float* mean, const float* rstd,
                              void* dact, float* dw, float* db,
                              float* sum_dxhat, float* sum_dxhat_xhat,
                              float* dgamma, float* dbeta, int N, int C,
                              int H, int W, int K, hipStream_t stream) {
  const int pad = K / 2;
  const size_t smem =
      (2 * (size_t)(CB2_TR + 2 * pad) * (W + 2 * pad) + K * K) * sizeof(float);
  const dim3 grid((unsigned)N * C, (unsigned)((H + CB2_TR - 1) / CB2_TR)),
      blk(CB2_BLOCK);
#define CB2D_BWD(KT)                                                        \
  hipLaunchKernelGGL((npf_cb2d_bwd_dact<T, KT>), grid, blk, smem, stream,   \
                     (const T*)x, w, (const T*)dy, gamma, beta, mean, rstd, \
                     (T*)dact, dw, db, sum_dxhat, sum_dxhat_xhat, dgamma,   \
                     dbeta, N, C, H, W, K)
  switch (K) {
    case 5: CB2D_BWD(5); break;
    case 9: CB2D_BWD(9); break;
    case 11: CB2D_BWD(11); break;
    default: CB2D_BWD(0);
  }
#undef CB2D_BWD
}

extern "C" void npf_cb2d_bwd_dact_launch(
    const void* x, const float* w, const void* dy, const float* gamma,
    const float* beta, const float* mean, const float* rstd, void* dact,
    float* dw, float* db, float* sum_dxhat, float* sum_dxhat_xhat,
    float* dgamma, float* dbeta, int N, int C, int H, int W, int K,
    int is_bf16, hipStream_t stream) {
  if (is_bf16)
    cb2d_bwd_dispatch<__hip_bfloat16>(x, w, dy, gamma, beta, mean, rstd,
                                      dact, dw, db, sum_dxhat,
                                      sum_dxhat_xhat, dgamma, dbeta, N, C, H,
                                      W, K, stream);
  else
    cb2d_bwd_dispatch<float>(x, w, dy, gamma, beta, mean, rstd, dact, dw, db,
                             sum_dxhat, sum_dxhat_xhat, dgamma, dbeta, N, C,
                             H, W, K, stream);
}
