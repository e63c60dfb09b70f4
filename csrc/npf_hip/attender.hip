// Fused TransformerAttender block pieces (SURVEY.md §2.3 "Cross-attention"
// row; reference attention.py:375-527 MultiheadAttender + :530-588
// TransformerAttender).  Two kernel families:
//
// 1) npf_qkv_fwd / npf_qkv_bwd — the K/Q/V projections as ONE MFMA kernel
//    over up to 3 (input, weight) problems, writing the HEAD-SPLIT layout
//    [H*B, N, hs] directly: the reference's 3 cast+GEMM launches plus 3
//    permute+contiguous kernels collapse into one launch.  hs = 16 (every
//    shipped config: kq=128, 8 heads), so a head tile aligns exactly with
//    one MFMA 16-col fragment and the head split is a pure store-index map.
//    Backward: one kernel produces dX (MFMA, W staged transposed) and
//    re-materializes dZ in row-major [R, D] (coalesced) so the dW rank-128
//    GEMMs run on hipBLASLt; q-bias grads are atomics like mlp_chain.
//
// 2) npf_add_ln_fwd / npf_add_ln_bwd — y = LayerNorm(a + b) with an
//    optional head-split gather on `a`: LN1(context + queries) consumes the
//    attention output [H*B, Q, 16] WITHOUT a head-merge permute kernel, and
//    LN2(h + FFN(h)) reuses the same kernels in plain layout.  One wave per
//    row (D <= 128: 2 elems/lane), fp32 statistics, saved (s, mean, rstd)
//    for the standard LN backward; dgamma/dbeta are fp32 atomics.
//
// Conventions (fragment layout, staging, padding) follow mlp_chain.hip.

#include "common.h"

#define QK_MAX_D 128
#define QK_TR 32
#define QK_BLOCK 128  // 2 waves
#define QK_PAD 8
#define QK_STRIDE (QK_MAX_D + QK_PAD)

typedef __attribute__((ext_vector_type(8))) short qk_bf16x8;
typedef __attribute__((ext_vector_type(4))) float qk_f32x4;

struct QkvProblem {
  const __hip_bfloat16* x;     // [R, D] row-major
  const float* w;              // [D, D] torch Linear weight (out, in)
  const float* b;              // [D] or null
  __hip_bfloat16* out;         // head-split [H*B, N, hs]
  const __hip_bfloat16* dout;  // bwd in: head-split grad [H*B, N, hs]
  __hip_bfloat16* dz;          // bwd out: re-materialized [R, D]
  __hip_bfloat16* dx;          // bwd out: [R, D]
  float* db;                   // bwd: [D] atomics or null
  long R;                      // B * N
  int B;
  int N;
};

struct QkvParams {
  QkvProblem p[3];
  int n_problems;
  int D;        // model width (= H * hs)
  int H;        // heads
};

__device__ __forceinline__ qk_bf16x8 qk_ld16(const __hip_bfloat16* p) {
  return *reinterpret_cast<const qk_bf16x8*>(p);
}

// stage [QK_TR, D] rows r0.. of a [R, D] bf16 tensor into LDS (zero-padded)
__device__ __forceinline__ void qk_stage_x(
    const __hip_bfloat16* __restrict__ src, __hip_bfloat16 (*dst)[QK_STRIDE],
    long r0, long R, int d, int d_p) {
  const int tr = (int)((R - r0) < QK_TR ? (R - r0) : QK_TR);
  const int qn = d >> 3;  // d % 8 == 0 (d is a multiple of 16)
  for (int i = threadIdx.x; i < tr * qn; i += QK_BLOCK) {
    const int r = i / qn, k = (i % qn) << 3;
    *reinterpret_cast<qk_bf16x8*>(&dst[r][k]) =
        *reinterpret_cast<const qk_bf16x8*>(&src[(r0 + r) * d + k]);
  }
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  const int kpad = d_p - d;
  if (kpad > 0)
    for (int i = threadIdx.x; i < QK_TR * kpad; i += QK_BLOCK)
      dst[i / kpad][d + i % kpad] = z;
  for (int i = threadIdx.x; i < (QK_TR - tr) * d; i += QK_BLOCK)
    dst[tr + i / d][i % d] = z;
}

// stage the head-split tensor rows r0.. (logical [R, D] view) into LDS:
// src[(h*B + b), n, hs] with r = b*N + n, c = h*hs + j
__device__ __forceinline__ void qk_stage_headsplit(
    const __hip_bfloat16* __restrict__ src, __hip_bfloat16 (*dst)[QK_STRIDE],
    long r0, long R, int d, int d_p, int B, int N, int H) {
  const int hs = d / H;
  const int tr = (int)((R - r0) < QK_TR ? (R - r0) : QK_TR);
  for (int i = threadIdx.x; i < tr * d; i += QK_BLOCK) {
    const int r = i / d, c = i % d;
    const long rr = r0 + r;
    const int b = (int)(rr / N), n = (int)(rr % N);
    const int h = c / hs, j = c % hs;
    dst[r][c] = src[((long)(h * B + b) * N + n) * hs + j];
  }
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  const int kpad = d_p - d;
  if (kpad > 0)
    for (int i = threadIdx.x; i < QK_TR * kpad; i += QK_BLOCK)
      dst[i / kpad][d + i % kpad] = z;
  for (int i = threadIdx.x; i < (QK_TR - tr) * d; i += QK_BLOCK)
    dst[tr + i / d][i % d] = z;
}

// stage [dout, din] fp32 weight -> bf16 LDS, optional transpose, zero-pad
__device__ __forceinline__ void qk_stage_w(
    const float* __restrict__ w, __hip_bfloat16 (*ws)[QK_STRIDE], int d,
    int d_p, bool transpose) {
  const int qn = d >> 2;
  for (int i = threadIdx.x; i < d * qn; i += QK_BLOCK) {
    const int o = i / qn, k = (i % qn) << 2;
    const float4 v = reinterpret_cast<const float4*>(w)[i];
    if (transpose) {
      ws[k][o] = __float2bfloat16(v.x);
      ws[k + 1][o] = __float2bfloat16(v.y);
      ws[k + 2][o] = __float2bfloat16(v.z);
      ws[k + 3][o] = __float2bfloat16(v.w);
    } else {
      ws[o][k] = __float2bfloat16(v.x);
      ws[o][k + 1] = __float2bfloat16(v.y);
      ws[o][k + 2] = __float2bfloat16(v.z);
      ws[o][k + 3] = __float2bfloat16(v.w);
    }
  }
  const __hip_bfloat16 z = __float2bfloat16(0.f);
  for (int i = threadIdx.x; i < d_p * (d_p - d); i += QK_BLOCK) {
    const int o = i / (d_p - d), k = d + i % (d_p - d);
    ws[o][k] = z;
    ws[k][o] = z;
  }
}

__device__ __forceinline__ void qk_tile_gemm(
    const __hip_bfloat16 (*a)[QK_STRIDE], const __hip_bfloat16 (*w)[QK_STRIDE],
    qk_f32x4* acc, int d_p, int row0, int lane) {
  for (int kk = 0; kk < d_p; kk += 32) {
    const qk_bf16x8 av = qk_ld16(&a[row0 + (lane & 15)][kk + (lane >> 4) * 8]);
    for (int n = 0; n < d_p / 16; ++n) {
      const qk_bf16x8 bv = qk_ld16(&w[n * 16 + (lane & 15)][kk + (lane >> 4) * 8]);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc[n], 0, 0, 0);
    }
  }
}

// ---------------------------------------------------------------------------
// forward: out_p = headsplit(x_p @ W_p^T + b_p)
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(QK_BLOCK)
npf_qkv_fwd(QkvParams prm) {
  __shared__ __align__(16) __hip_bfloat16 a_lds[QK_TR][QK_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 w_lds[QK_MAX_D][QK_STRIDE];
  __shared__ float b_lds[QK_MAX_D];

  const QkvProblem& pb = prm.p[blockIdx.y];
  const long r0 = (long)blockIdx.x * QK_TR;
  if (r0 >= pb.R) return;
  const int d = prm.D;
  const int d_p = (d + 31) & ~31;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hs = d / prm.H;

  qk_stage_x(pb.x, a_lds, r0, pb.R, d, d_p);
  qk_stage_w(pb.w, w_lds, d, d_p, false);
  for (int i = threadIdx.x; i < d_p; i += QK_BLOCK)
    b_lds[i] = (pb.b != nullptr && i < d) ? pb.b[i] : 0.f;
  __syncthreads();

  const int row0 = wave * 16;
  qk_f32x4 acc[QK_MAX_D / 16];
  #pragma unroll
  for (int n = 0; n < QK_MAX_D / 16; ++n) acc[n] = qk_f32x4{0.f, 0.f, 0.f, 0.f};
  qk_tile_gemm(a_lds, w_lds, acc, d_p, row0, lane);

  const int col = lane & 15;
  const int rbase = row0 + (lane >> 4) * 4;
  for (int n = 0; n < d_p / 16; ++n) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const long r = r0 + rbase + i;
      const int c = n * 16 + col;
      if (r < pb.R && c < d) {
        const int b = (int)(r / pb.N), nn = (int)(r % pb.N);
        const int h = c / hs, j = c % hs;
        pb.out[((long)(h * pb.B + b) * pb.N + nn) * hs + j] =
            __float2bfloat16(acc[n][i] + b_lds[c]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward: dX_p = dOut_p @ W_p (dOut gathered from head-split), dZ_p
// re-materialized row-major for the host-side dW GEMMs, db atomics
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(QK_BLOCK)
npf_qkv_bwd(QkvParams prm) {
  __shared__ __align__(16) __hip_bfloat16 a_lds[QK_TR][QK_STRIDE];
  __shared__ __align__(16) __hip_bfloat16 w_lds[QK_MAX_D][QK_STRIDE];

  const QkvProblem& pb = prm.p[blockIdx.y];
  const long r0 = (long)blockIdx.x * QK_TR;
  if (r0 >= pb.R) return;
  const int d = prm.D;
  const int d_p = (d + 31) & ~31;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // dz tile (gathered): logical row-major view of the head-split grad
  qk_stage_headsplit(pb.dout, a_lds, r0, pb.R, d, d_p, pb.B, pb.N, prm.H);
  qk_stage_w(pb.w, w_lds, d, d_p, true);
  __syncthreads();

  // re-materialize dz row-major (coalesced) + bias-grad atomics
  {
    const int tr = (int)((pb.R - r0) < QK_TR ? (pb.R - r0) : QK_TR);
    for (int i = threadIdx.x; i < tr * d; i += QK_BLOCK) {
      const int r = i / d, c = i % d;
      pb.dz[(r0 + r) * d + c] = a_lds[r][c];
    }
    if (pb.db != nullptr) {
      for (int c = threadIdx.x; c < d; c += QK_BLOCK) {
        float s = 0.f;
        for (int r = 0; r < tr; ++r) s += __bfloat162float(a_lds[r][c]);
        atomicAdd(&pb.db[c], s);
      }
    }
  }

  const int row0 = wave * 16;
  qk_f32x4 acc[QK_MAX_D / 16];
  #pragma unroll
  for (int n = 0; n < QK_MAX_D / 16; ++n) acc[n] = qk_f32x4{0.f, 0.f, 0.f, 0.f};
  qk_tile_gemm(a_lds, w_lds, acc, d_p, row0, lane);

  const int col = lane & 15;
  const int rbase = row0 + (lane >> 4) * 4;
  for (int n = 0; n < d_p / 16; ++n) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const long r = r0 + rbase + i;
      const int c = n * 16 + col;
      if (r < pb.R && c < d)
        pb.dx[r * d + c] = __float2bfloat16(acc[n][i]);
    }
  }
}

// ---------------------------------------------------------------------------
// add + LayerNorm: y = gamma * (s - mu) * rstd + beta,  s = a + b
// `a` optionally lives in head-split layout [H*B, N, hs] (attention output).
// One wave per row: D <= 128 => 2 elements per lane, shuffle reductions.
// ---------------------------------------------------------------------------

#define ALN_WAVES 4  // rows per workgroup

struct AddLnParams {
  const __hip_bfloat16* a;
  const __hip_bfloat16* b;
  const float* gamma;
  const float* beta;
  __hip_bfloat16* y;
  __hip_bfloat16* s;       // saved a+b
  float* mean;
  float* rstd;
  // bwd
  const __hip_bfloat16* dy;
  __hip_bfloat16* da;      // written in a's layout (head-split if flagged)
  float* dgamma;
  float* dbeta;
  long R;
  int D;
  int B, N, H;             // head-split mapping of `a` (H == 0: plain)
  float eps;
};

__device__ __forceinline__ long aln_a_index(const AddLnParams& p, long r,
                                            int c) {
  if (p.H == 0) return r * p.D + c;
  const int hs = p.D / p.H;
  const int b = (int)(r / p.N), n = (int)(r % p.N);
  const int h = c / hs, j = c % hs;
  return ((long)(h * p.B + b) * p.N + n) * hs + j;
}

extern "C" __global__ void __launch_bounds__(64 * ALN_WAVES)
npf_add_ln_fwd(AddLnParams p) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long r = (long)blockIdx.x * ALN_WAVES + wave;
  if (r >= p.R) return;
  const int D = p.D;

  float v[2] = {0.f, 0.f};
  float sum = 0.f;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = lane + 64 * i;
    if (c < D) {
      v[i] = __bfloat162float(p.a[aln_a_index(p, r, c)]) +
             __bfloat162float(p.b[r * D + c]);
      sum += v[i];
    }
  }
  for (int off = 32; off; off >>= 1) sum += __shfl_down(sum, off);
  const float mu = __shfl(sum, 0) / D;
  float var = 0.f;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = lane + 64 * i;
    if (c < D) var += (v[i] - mu) * (v[i] - mu);
  }
  for (int off = 32; off; off >>= 1) var += __shfl_down(var, off);
  const float rstd = rsqrtf(__shfl(var, 0) / D + p.eps);

  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = lane + 64 * i;
    if (c < D) {
      p.s[r * D + c] = __float2bfloat16(v[i]);
      p.y[r * D + c] = __float2bfloat16(
          p.gamma[c] * (v[i] - mu) * rstd + p.beta[c]);
    }
  }
  if (lane == 0) {
    p.mean[r] = mu;
    p.rstd[r] = rstd;
  }
}

extern "C" __global__ void __launch_bounds__(64 * ALN_WAVES)
npf_add_ln_bwd(AddLnParams p) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long r = (long)blockIdx.x * ALN_WAVES + wave;
  if (r >= p.R) return;
  const int D = p.D;
  const float mu = p.mean[r], rstd = p.rstd[r];

  float xh[2] = {0.f, 0.f}, g[2] = {0.f, 0.f};
  float s1 = 0.f, s2 = 0.f;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = lane + 64 * i;
    if (c < D) {
      xh[i] = (__bfloat162float(p.s[r * D + c]) - mu) * rstd;
      g[i] = __bfloat162float(p.dy[r * D + c]) * p.gamma[c];
      s1 += g[i];
      s2 += g[i] * xh[i];
    }
  }
  for (int off = 32; off; off >>= 1) {
    s1 += __shfl_down(s1, off);
    s2 += __shfl_down(s2, off);
  }
  const float m1 = __shfl(s1, 0) / D, m2 = __shfl(s2, 0) / D;

  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = lane + 64 * i;
    if (c < D) {
      const float ds = rstd * (g[i] - m1 - xh[i] * m2);
      const __hip_bfloat16 h = __float2bfloat16(ds);
      p.da[aln_a_index(p, r, c)] = h;       // grad w.r.t. a (a's layout)
      p.y[r * D + c] = h;                   // y reused as db (plain layout)
    }
  }
}

// dgamma[c] = sum_r dy[r,c] * xhat[r,c]; dbeta[c] = sum_r dy[r,c].
// Separate column-reduction pass: per-element atomics in the row kernel
// were 524k global atomics on 128 addresses.  Each thread owns a column
// (coalesced across the warp), each workgroup strides a row slab; one
// atomic per (workgroup, column).
extern "C" __global__ void __launch_bounds__(128)
npf_add_ln_gb(AddLnParams p) {
  const int c = threadIdx.x;
  if (c >= p.D) return;
  float sg = 0.f, sb = 0.f;
  for (long r = blockIdx.x; r < p.R; r += gridDim.x) {
    const float dyv = __bfloat162float(p.dy[r * p.D + c]);
    const float xh =
        (__bfloat162float(p.s[r * p.D + c]) - p.mean[r]) * p.rstd[r];
    sg += dyv * xh;
    sb += dyv;
  }
  atomicAdd(&p.dgamma[c], sg);
  atomicAdd(&p.dbeta[c], sb);
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

extern "C" void npf_qkv_fwd_launch(const void* const* xs,
                                   const float* const* ws,
                                   const float* const* bs, void* const* outs,
                                   const long* Rs, const int* Bs,
                                   const int* Ns, int n_problems, int D,
                                   int H, hipStream_t stream) {
  QkvParams p = {};
  long rmax = 0;
  for (int i = 0; i < n_problems; ++i) {
    p.p[i].x = (const __hip_bfloat16*)xs[i];
    p.p[i].w = ws[i];
    p.p[i].b = bs[i];
    p.p[i].out = (__hip_bfloat16*)outs[i];
    p.p[i].R = Rs[i];
    p.p[i].B = Bs[i];
    p.p[i].N = Ns[i];
    if (Rs[i] > rmax) rmax = Rs[i];
  }
  p.n_problems = n_problems;
  p.D = D;
  p.H = H;
  dim3 grid((unsigned)((rmax + QK_TR - 1) / QK_TR), n_problems);
  hipLaunchKernelGGL(npf_qkv_fwd, grid, dim3(QK_BLOCK), 0, stream, p);
}

extern "C" void npf_qkv_bwd_launch(const void* const* douts,
                                   void* const* dzs, const float* const* ws,
                                   void* const* dxs, float* const* dbs,
                                   const long* Rs, const int* Bs,
                                   const int* Ns, int n_problems, int D,
                                   int H, hipStream_t stream) {
  QkvParams p = {};
  long rmax = 0;
  for (int i = 0; i < n_problems; ++i) {
    p.p[i].dout = (const __hip_bfloat16*)douts[i];
    p.p[i].dz = (__hip_bfloat16*)dzs[i];
    p.p[i].w = ws[i];
    p.p[i].dx = (__hip_bfloat16*)dxs[i];
    p.p[i].db = dbs[i];
    p.p[i].R = Rs[i];
    p.p[i].B = Bs[i];
    p.p[i].N = Ns[i];
    if (Rs[i] > rmax) rmax = Rs[i];
  }
  p.n_problems = n_problems;
  p.D = D;
  p.H = H;
  dim3 grid((unsigned)((rmax + QK_TR - 1) / QK_TR), n_problems);
  hipLaunchKernelGGL(npf_qkv_bwd, grid, dim3(QK_BLOCK), 0, stream, p);
}

extern "C" void npf_add_ln_fwd_launch(const void* a, const void* b,
                                      const float* gamma, const float* beta,
                                      void* y, void* s, float* mean,
                                      float* rstd, long R, int D, int B,
                                      int N, int H, float eps,
                                      hipStream_t stream) {
  AddLnParams p = {};
  p.a = (const __hip_bfloat16*)a;
  p.b = (const __hip_bfloat16*)b;
  p.gamma = gamma;
  p.beta = beta;
  p.y = (__hip_bfloat16*)y;
  p.s = (__hip_bfloat16*)s;
  p.mean = mean;
  p.rstd = rstd;
  p.R = R;
  p.D = D;
  p.B = B;
  p.N = N;
  p.H = H;
  p.eps = eps;
  const unsigned grid = (unsigned)((R + ALN_WAVES - 1) / ALN_WAVES);
  hipLaunchKernelGGL(npf_add_ln_fwd, dim3(grid), dim3(64 * ALN_WAVES), 0,
                     stream, p);
}

extern "C" void npf_add_ln_bwd_launch(const void* s, const void* dy,
                                      const float* gamma, const float* mean,
                                      const float* rstd, void* da, void* db,
                                      float* dgamma, float* dbeta, long R,
                                      int D, int B, int N, int H,
                                      hipStream_t stream) {
  AddLnParams p = {};
  p.s = (__hip_bfloat16*)s;
  p.dy = (const __hip_bfloat16*)dy;
  p.gamma = gamma;
  p.mean = (float*)mean;  // read-only in the backward kernel
  p.rstd = (float*)rstd;
  p.da = (__hip_bfloat16*)da;
  p.y = (__hip_bfloat16*)db;
  p.dgamma = dgamma;
  p.dbeta = dbeta;
  p.R = R;
  p.D = D;
  p.B = B;
  p.N = N;
  p.H = H;
  const unsigned grid = (unsigned)((R + ALN_WAVES - 1) / ALN_WAVES);
  hipLaunchKernelGGL(npf_add_ln_bwd, dim3(grid), dim3(64 * ALN_WAVES), 0,
                     stream, p);
  // enough workgroups to cover the XCDs (the 64-WG version profiled at
  // 21 us with most of the chip idle); one atomic per (WG, column)
  const unsigned gb_grid =
      (unsigned)(R < 4096 ? ((unsigned)(R + 7) / 8 + 1) : 512);
  hipLaunchKernelGGL(npf_add_ln_gb, dim3(gb_grid), dim3(128), 0, stream, p);
}
