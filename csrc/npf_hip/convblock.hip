// Fused pre-activation depthwise conv block (1D), the ConvNP CNN hot loop.
//
//   a      = relu(batchnorm(x))          (batchnorm optional)
//   y[l]   = sum_j w[c,j] * a[l+j-pad] (+ b[c]) (+ x[l] if residual)
//
// Covers the norm2/act/conv2_depthwise/(+X) chain of ResConvBlock
// (reference cnn.py:204-215; our npf/architectures/cnn.py:132-137) on
// [N, C, L] tensors (N = Z*B up to ~512, C = r_dim = 128, L = n_induced =
// 192 for the 1D ConvCNP/ConvLNP configs, k = 19).  The pointwise conv that
// follows stays a library GEMM (hipBLASLt).
//
// Why fused: at these sizes every op is HBM/dispatch-bound, so the eager
// chain (norm stats, normalize, relu, im2col/miopen dwconv, add) is ~12
// kernels each reading+writing the whole [N,C,L] tensor; this file does the
// training forward in 2 kernels (channel stats + fused apply) and the
// backward in 3, keeping the activation row staged in LDS with its conv
// halo.
//
// Launch shape: the forward apply kernel uses one workgroup per (n, c) row
// (N*C = 4k-65k workgroups fills 256 CUs); the backward stencil tiles CB_TN
// batch rows per workgroup so its per-weight block-reductions amortize;
// stats/reduction kernels use one workgroup per channel.
//
// fp32 compute (BN statistics need it; tensors are fp32 master or bf16 —
// the python wrapper casts bf16 I/O, keeping the op numerically identical
// to the eager fp32 reference path).

#include "common.h"

#define CB_BLOCK 256
#define CB_MAX_K 31

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

// per-channel batch mean/var over [N, C, L] (training mode), two stages:
// partial (sum, sumsq) atomics over a (C x batch-slice) grid — a single
// workgroup per channel was latency-bound (1.1ms at [256,128,4096]) —
// then a tiny finalize kernel.
#define CB_STATS_SLICE 16   // batch rows per partial-reduction workgroup

template <typename T>
__global__ void __launch_bounds__(CB_BLOCK)
npf_cb_stats_partial(const T* __restrict__ x, float* __restrict__ sums,
                     int N, int C, int L) {
  __shared__ float red[16];
  const int c = blockIdx.x % C;
  const int n0 = (blockIdx.x / C) * CB_STATS_SLICE;
  const int nt = min(CB_STATS_SLICE, N - n0);
  float s = 0.f, s2 = 0.f;
  for (long i = threadIdx.x; i < (long)nt * L; i += CB_BLOCK) {
    const long n = n0 + i / L, l = i % L;
    const float v = ldf(x + (n * C + c) * (long)L + l);
    s += v;
    s2 += v * v;
  }
  s = block_reduce_sum(s, red);
  __syncthreads();
  s2 = block_reduce_sum(s2, red);
  if (threadIdx.x == 0) {
    atomicAdd(&sums[c], s);
    atomicAdd(&sums[C + c], s2);
  }
}

extern "C" __global__ void npf_cb_stats_finalize(
    const float* __restrict__ sums, float* __restrict__ mean,
    float* __restrict__ rstd, float* __restrict__ save_var,
    float* __restrict__ running_mean, float* __restrict__ running_var, int N,
    int C, int L, float eps, float momentum) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float M = (float)N * (float)L;
  const float mu = sums[c] / M;
  const float var = fmaxf(sums[C + c] / M - mu * mu, 0.f);
  mean[c] = mu;
  save_var[c] = var;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    // torch semantics: running update uses the UNBIASED batch variance
    const float unbiased = (M > 1.f) ? var * M / (M - 1.f) : var;
    running_mean[c] += momentum * (mu - running_mean[c]);
    running_var[c] += momentum * (unbiased - running_var[c]);
  }
}

// y = dwconv(relu(bn(x))) (+bias) (+x); one workgroup per (n, c) row
extern "C" __global__ void __launch_bounds__(CB_BLOCK)
npf_cb_fwd(const float* __restrict__ x, const float* __restrict__ res,
           const float* __restrict__ w,
           const float* __restrict__ bias, const float* __restrict__ gamma,
           const float* __restrict__ beta, const float* __restrict__ mean,
           const float* __restrict__ rstd, float* __restrict__ y,
           int N, int C, int L, int K) {
  extern __shared__ float a[];  // [L + K - 1] activation row with halo
  const int n = blockIdx.x / C;
  const int c = blockIdx.x % C;
  const int pad = K / 2;
  const float* xrow = x + ((long)n * C + c) * L;

  const float mu = (gamma != nullptr) ? mean[c] : 0.f;
  const float gscale = (gamma != nullptr) ? rstd[c] * gamma[c] : 1.f;
  const float gshift = (gamma != nullptr) ? beta[c] : 0.f;

  for (int l = threadIdx.x; l < L + 2 * pad; l += CB_BLOCK) {
    const int src = l - pad;
    float v = 0.f;
    if (src >= 0 && src < L) {
      v = fmaxf((xrow[src] - mu) * gscale + gshift, 0.f);
    }
    a[l] = v;
  }
  __syncthreads();

  float wreg[CB_MAX_K];
  #pragma unroll 4
  for (int j = 0; j < K; ++j) wreg[j] = w[c * K + j];
  const float b = (bias != nullptr) ? bias[c] : 0.f;

  float* yrow = y + ((long)n * C + c) * L;
  for (int l = threadIdx.x; l < L; l += CB_BLOCK) {
    float acc = b;
    #pragma unroll 4
    for (int j = 0; j < K; ++j) acc += wreg[j] * a[l + j];
    if (res != nullptr) acc += res[((long)n * C + c) * L + l];
    yrow[l] = acc;
  }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------

// B1: dact = relu'(a) * corr(w, dY); per-channel partials for BN/weight grads
//     dact is stored; channel sums go through atomics on [C] buffers.
//     One workgroup covers CB_TN batch rows of ONE channel so the K
//     block-reductions for dW amortize over the tile (measured 72us -> the
//     per-(n,c) version spent most of its time in 19 serialized reductions).
#define CB_TN 4
extern "C" __global__ void __launch_bounds__(CB_BLOCK)
npf_cb_bwd_dact(const float* __restrict__ x, const float* __restrict__ w,
                const float* __restrict__ dy, const float* __restrict__ gamma,
                const float* __restrict__ beta, const float* __restrict__ mean,
                const float* __restrict__ rstd, float* __restrict__ dact,
                float* __restrict__ dw, float* __restrict__ db,
                float* __restrict__ sum_dxhat, float* __restrict__ sum_dxhat_xhat,
                float* __restrict__ dgamma, float* __restrict__ dbeta,
                int N, int C, int L, int K) {
  extern __shared__ float smem[];
  const int pad = K / 2;
  const int LP = L + 2 * pad;
  float* dys = smem;            // [CB_TN][LP] dY rows with halo
  float* as = smem + CB_TN * LP;  // [CB_TN][LP] activation rows with halo
  __shared__ float red[16];

  const int c = blockIdx.x % C;
  const int n0 = (blockIdx.x / C) * CB_TN;
  const int tn = min(CB_TN, N - n0);

  const bool has_bn = gamma != nullptr;
  const float mu = has_bn ? mean[c] : 0.f;
  const float rs = has_bn ? rstd[c] : 1.f;
  const float gm = has_bn ? gamma[c] : 1.f;
  const float gscale = has_bn ? rs * gm : 1.f;
  const float gshift = has_bn ? beta[c] : 0.f;

  for (int i = threadIdx.x; i < tn * LP; i += CB_BLOCK) {
    const int t = i / LP, src = i % LP - pad;
    const long base = ((long)(n0 + t) * C + c) * L;
    const bool in = (src >= 0 && src < L);
    dys[i] = in ? dy[base + src] : 0.f;
    as[i] = in ? fmaxf((x[base + src] - mu) * gscale + gshift, 0.f) : 0.f;
  }
  __syncthreads();

  float wreg[CB_MAX_K];
  #pragma unroll 4
  for (int j = 0; j < K; ++j) wreg[j] = w[c * K + j];

  // dW[c,j] = sum_{n,l} dY * a[l+j-pad]; db[c] = sum dY
  float dwp[CB_MAX_K];
  #pragma unroll 4
  for (int j = 0; j < K; ++j) dwp[j] = 0.f;
  float dbp = 0.f;

  float s_dxhat = 0.f, s_dxhat_xhat = 0.f, s_dg = 0.f, s_db = 0.f;
  for (int i = threadIdx.x; i < tn * L; i += CB_BLOCK) {
    const int t = i / L, l = i % L;
    const float* dyrow_s = dys + t * LP;
    const float* arow_s = as + t * LP;
    const float dyl = dyrow_s[l + pad];
    dbp += dyl;
    #pragma unroll 4
    for (int j = 0; j < K; ++j) dwp[j] += dyl * arow_s[l + j];
    // transposed conv: da[l] = sum_j w[j] * dY[l - j + pad]
    float da = 0.f;
    #pragma unroll 4
    for (int j = 0; j < K; ++j) da += wreg[j] * dyrow_s[l + (K - 1 - j)];
    const float act = arow_s[l + pad];
    const float dr = (act > 0.f) ? da : 0.f;  // through relu
    dact[((long)(n0 + t) * C + c) * L + l] = dr;
    if (has_bn) {
      const float xhat = (x[((long)(n0 + t) * C + c) * L + l] - mu) * rs;
      const float dxh = dr * gm;
      s_dxhat += dxh;
      s_dxhat_xhat += dxh * xhat;
      s_dg += dr * xhat;
      s_db += dr;
    }
  }

  #pragma unroll 4
  for (int j = 0; j < K; ++j) {
    const float v = block_reduce_sum(dwp[j], red);
    __syncthreads();
    if (threadIdx.x == 0) atomicAdd(&dw[c * K + j], v);
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

// B2: dx = bn_backward(dact) (+ dY residual)
template <typename T>
__global__ void __launch_bounds__(CB_BLOCK)
npf_cb_bwd_dx(const T* __restrict__ x, const T* __restrict__ dact,
              const T* __restrict__ dy, const float* __restrict__ gamma,
              const float* __restrict__ mean, const float* __restrict__ rstd,
              const float* __restrict__ sum_dxhat,
              const float* __restrict__ sum_dxhat_xhat, T* __restrict__ dx,
              int N, int C, int L, int training) {
  const int n = blockIdx.x / C;
  const int c = blockIdx.x % C;
  const long off = ((long)n * C + c) * L;
  const bool has_bn = gamma != nullptr;
  const float M = (float)N * (float)L;
  const float mu = has_bn ? mean[c] : 0.f;
  const float rs = has_bn ? rstd[c] : 1.f;
  const float g = has_bn ? gamma[c] : 1.f;
  const float mean_dxh = has_bn ? sum_dxhat[c] / M : 0.f;
  const float mean_dxh_xh = has_bn ? sum_dxhat_xhat[c] / M : 0.f;
  for (int l = threadIdx.x; l < L; l += CB_BLOCK) {
    float d;
    if (has_bn) {
      const float dxh = ldf(dact + off + l) * g;
      if (training) {
        const float xhat = (ldf(x + off + l) - mu) * rs;
        d = rs * (dxh - mean_dxh - xhat * mean_dxh_xh);
      } else {
        d = rs * dxh;  // eval mode: stats are constants
      }
    } else {
      d = ldf(dact + off + l);
    }
    stf(dx + off + l, d);
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

extern "C" void npf_cb_stats_launch(const void* x, int is_bf16,
                                    float* sums_ws, float* mean, float* rstd,
                                    float* save_var, float* running_mean,
                                    float* running_var, int N, int C, int L,
                                    float eps, float momentum,
                                    hipStream_t stream) {
  const unsigned slices = (unsigned)((N + CB_STATS_SLICE - 1) / CB_STATS_SLICE);
  if (is_bf16)
    hipLaunchKernelGGL((npf_cb_stats_partial<__hip_bfloat16>),
                       dim3(slices * C), dim3(CB_BLOCK), 0, stream,
                       (const __hip_bfloat16*)x, sums_ws, N, C, L);
  else
    hipLaunchKernelGGL((npf_cb_stats_partial<float>), dim3(slices * C),
                       dim3(CB_BLOCK), 0, stream, (const float*)x, sums_ws,
                       N, C, L);
  hipLaunchKernelGGL(npf_cb_stats_finalize, dim3((C + 255) / 256), dim3(256),
                     0, stream, sums_ws, mean, rstd, save_var, running_mean,
                     running_var, N, C, L, eps, momentum);
}

extern "C" void npf_cb_fwd_launch(const float* x, const float* res,
                                  const float* w,
                                  const float* bias, const float* gamma,
                                  const float* beta, const float* mean,
                                  const float* rstd, float* y, int N, int C,
                                  int L, int K, hipStream_t stream) {
  const size_t smem = (size_t)(L + K - 1) * sizeof(float);
  hipLaunchKernelGGL(npf_cb_fwd, dim3((unsigned)N * C), dim3(CB_BLOCK), smem,
                     stream, x, res, w, bias, gamma, beta, mean, rstd, y, N,
                     C, L, K);
}

extern "C" void npf_cb_bwd_launch(const float* x, const float* w,
                                  const float* dy, const float* gamma,
                                  const float* beta, const float* mean,
                                  const float* rstd, float* dact, float* dw,
                                  float* db, float* sum_dxhat,
                                  float* sum_dxhat_xhat, float* dgamma,
                                  float* dbeta, float* dx, int N, int C,
                                  int L, int K, int training,
                                  hipStream_t stream) {
  const size_t smem = (size_t)(2 * CB_TN * (L + K - 1)) * sizeof(float);
  const unsigned ntiles = (unsigned)((N + CB_TN - 1) / CB_TN);
  hipLaunchKernelGGL(npf_cb_bwd_dact, dim3(ntiles * C), dim3(CB_BLOCK),
                     smem, stream, x, w, dy, gamma, beta, mean, rstd, dact,
                     dw, db, sum_dxhat, sum_dxhat_xhat, dgamma, dbeta, N, C,
                     L, K);
  hipLaunchKernelGGL((npf_cb_bwd_dx<float>), dim3((unsigned)N * C),
                     dim3(CB_BLOCK), 0, stream, x, dact, dy, gamma, mean,
                     rstd, sum_dxhat, sum_dxhat_xhat, dx, N, C, L, training);
}

// standalone launcher for the elementwise BN-backward kernel (shared by the
// 2D block, which flattens L = H*W); dtype-dispatched
extern "C" void npf_cb_bwd_dx_launch(const void* x, const void* dact,
                                     const void* dy, const float* gamma,
                                     const float* mean, const float* rstd,
                                     const float* sum_dxhat,
                                     const float* sum_dxhat_xhat, void* dx,
                                     int N, int C, int L, int training,
                                     int is_bf16, hipStream_t stream) {
  if (is_bf16)
    hipLaunchKernelGGL((npf_cb_bwd_dx<__hip_bfloat16>),
                       dim3((unsigned)N * C), dim3(CB_BLOCK), 0, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dact,
                       (const __hip_bfloat16*)dy, gamma, mean, rstd,
                       sum_dxhat, sum_dxhat_xhat, (__hip_bfloat16*)dx, N, C,
                       L, training);
  else
    hipLaunchKernelGGL((npf_cb_bwd_dx<float>), dim3((unsigned)N * C),
                       dim3(CB_BLOCK), 0, stream, (const float*)x,
                       (const float*)dact, (const float*)dy, gamma, mean,
                       rstd, sum_dxhat, sum_dxhat_xhat, (float*)dx, N, C, L,
                       training);
}
