// Fused diagonal-Gaussian log-likelihood + target-set reduction.
//
// Computes out[z,b] = sum_m [ -(y-loc)^2/(2 s^2) - log s - 0.5 log 2pi ]
// over the flattened target dims m (SURVEY.md §2.3 "Gaussian NLL +
// reductions"; the reference composes Independent(Normal).log_prob with a
// view+sum, reference npf/losses.py:18-24).
//
// One workgroup per (z,b) row, grid-stride over the row, wave+LDS reduce:
// a single kernel replaces log_prob's ~6 elementwise kernels + a reduction.

#include "common.h"

#define LL_BLOCK 256
#define HALF_LOG_2PI 0.9189385332046727f

extern "C" __global__ void __launch_bounds__(LL_BLOCK)
npf_gauss_ll_fwd(const float* __restrict__ loc, const float* __restrict__ scale,
                 const float* __restrict__ y, float* __restrict__ out,
                 long long rows, long long m) {
  __shared__ float red[16];
  const long long row = blockIdx.x;
  if (row >= rows) return;
  const float* l = loc + row * m;
  const float* s = scale + row * m;
  const float* t = y + row * m;
  float acc = 0.f;
  for (long long i = threadIdx.x; i < m; i += LL_BLOCK) {
    const float d = t[i] - l[i];
    const float sc = s[i];
    acc += -d * d / (2.f * sc * sc) - __logf(sc);
  }
  acc = block_reduce_sum(acc, red);
  if (threadIdx.x == 0) out[row] = acc - HALF_LOG_2PI * (float)m;
}

extern "C" __global__ void __launch_bounds__(LL_BLOCK)
npf_gauss_ll_bwd(const float* __restrict__ loc, const float* __restrict__ scale,
                 const float* __restrict__ y, const float* __restrict__ dout,
                 float* __restrict__ dloc, float* __restrict__ dscale,
                 long long rows, long long m) {
  const long long row = blockIdx.x;
  if (row >= rows) return;
  const float g = dout[row];
  const long long base = row * m;
  for (long long i = threadIdx.x; i < m; i += LL_BLOCK) {
    const float d = y[base + i] - loc[base + i];
    const float sc = scale[base + i];
    const float inv_s = 1.f / sc;
    const float inv_s2 = inv_s * inv_s;
    dloc[base + i] = g * d * inv_s2;
    dscale[base + i] = g * (d * d * inv_s2 - 1.f) * inv_s;
  }
}

// ---------------------------------------------------------------------------
// NPML epilogue: out[b] = logsumexp_z w[z,b] - log Z  (SURVEY.md §2.3 "NPML
// objective" row; reference npf/losses.py:169-203 does this with ~6 torch
// kernels).  Z <= 32 in every shipped config: one wave per b, each lane owns
// a z stripe, shuffle-reduce max then sum.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(64)
npf_lse_z_fwd(const float* __restrict__ w, float* __restrict__ out,
              long long Z, long long B) {
  const long long b = blockIdx.x;
  if (b >= B) return;
  float mx = -INFINITY;
  for (long long z = threadIdx.x; z < Z; z += 64)
    mx = fmaxf(mx, w[z * B + b]);
  for (int off = 32; off; off >>= 1)
    mx = fmaxf(mx, __shfl_down(mx, off));
  mx = __shfl(mx, 0);
  float s = 0.f;
  for (long long z = threadIdx.x; z < Z; z += 64)
    s += expf(w[z * B + b] - mx);
  for (int off = 32; off; off >>= 1)
    s += __shfl_down(s, off);
  if (threadIdx.x == 0) out[b] = mx + logf(s) - logf((float)Z);
}

// dw[z,b] = dout[b] * softmax_z(w)[z,b] = dout[b] * exp(w - (out[b] + log Z))
extern "C" __global__ void __launch_bounds__(256)
npf_lse_z_bwd(const float* __restrict__ w, const float* __restrict__ out,
              const float* __restrict__ dout, float* __restrict__ dw,
              long long Z, long long B) {
  const long long i = (long long)blockIdx.x * 256 + threadIdx.x;
  if (i >= Z * B) return;
  const long long b = i % B;
  const float lse = out[b] + logf((float)Z);
  dw[i] = dout[b] * expf(w[i] - lse);
}

extern "C" void npf_lse_z_fwd_launch(const float* w, float* out, long long Z,
                                     long long B, hipStream_t stream) {
  hipLaunchKernelGGL(npf_lse_z_fwd, dim3((unsigned)B), dim3(64), 0, stream, w,
                     out, Z, B);
}

extern "C" void npf_lse_z_bwd_launch(const float* w, const float* out,
                                     const float* dout, float* dw, long long Z,
                                     long long B, hipStream_t stream) {
  const long long n = Z * B;
  hipLaunchKernelGGL(npf_lse_z_bwd, dim3((unsigned)((n + 255) / 256)),
                     dim3(256), 0, stream, w, out, dout, dw, Z, B);
}

extern "C" void npf_gauss_ll_fwd_launch(const float* loc, const float* scale,
                                        const float* y, float* out,
                                        long long rows, long long m,
                                        hipStream_t stream) {
  hipLaunchKernelGGL(npf_gauss_ll_fwd, dim3((unsigned)rows), dim3(LL_BLOCK), 0,
                     stream, loc, scale, y, out, rows, m);
}

extern "C" void npf_gauss_ll_bwd_launch(const float* loc, const float* scale,
                                        const float* y, const float* dout,
                                        float* dloc, float* dscale,
                                        long long rows, long long m,
                                        hipStream_t stream) {
  hipLaunchKernelGGL(npf_gauss_ll_bwd, dim3((unsigned)rows), dim3(LL_BLOCK), 0,
                     stream, loc, scale, y, dout, dloc, dscale, rows, m);
}
