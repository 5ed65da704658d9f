// Fused BatchNorm2d (+ReLU) (+residual add) for NHWC (channels_last) bf16,
// MI355X (gfx950, CDNA4).
//
// Motivation (profiles/r01): MIOpen's 4-kernel spatial BN plus the unfused
// eager add/relu/clamp glue is 53% of a ResNet-50 bf16 step.  This file
// collapses each BN layer-pass to:
//   forward:  bn_sum_partial  (per-channel sum/sumsq, atomics)
//             bn_fwd_finalize (C-thread: mean/var/running stats + scale/bias)
//             bn_fwd_apply    (y = relu(x*scale + bias [+ residual]))
//   backward: bn_bwd_reduce   (per-channel sum(dy_eff), sum(dy_eff*xhat),
//                              dy_eff = relu-masked dy; atomics)
//             bn_bwd_finalize (C-thread: dgamma/dbeta + dx coefficients)
//             bn_bwd_apply    (dx = a*dy_eff + b + d*(x-mean) [, dres])
//
// Geometry: channels_last puts C innermost, so a wave reads adjacent
// channels: 64 lanes x 2 channels (ushort2, 4 B/lane) for the reductions,
// 8 bf16 (16 B/lane) for the elementwise passes (Guideline 13).  Per-block
// partials reduce through LDS, one float atomicAdd per channel per block
// (Guideline 12).  Two-pass statistics are exact (fp32 accumulation).
#include <hip/hip_runtime.h>
#include <cstdint>

#define BN_THREADS 256
#define BN_LANES 64          // lanes across channels
#define BN_ROWS 4            // rows per block iteration (BN_THREADS/BN_LANES)
#define BN_CHUNK (BN_LANES * 2)  // channels per block (2 per lane)

__device__ __forceinline__ float bnb2f(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t bnf2b(float f) {
  union { uint32_t u; float f; } v;
  v.f = f;
  uint32_t u = v.u + (0x7FFF + ((v.u >> 16) & 1));
  return (uint16_t)(u >> 16);
}

// ---------------------------------------------------------------- forward

// LDS tree-fold of two 8-wide per-thread accumulator vectors across the
// thread groups that share a channel set (group stride G = C/8, power of
// 2), then ONE partial row per block: p[c * NB + blockIdx.x].  Global
// atomics are NOT used — per-address atomic serialization ran at
// 32-930 GB/s vs 6.2 TB/s for this scheme (scripts/bnrepro.hip).  A second
// kernel (bn_fold_partials) folds the NB partials per channel.
__device__ __forceinline__ void bn_block_fold8(
    float acc0[8], float acc1[8], int C, int c, int NB,
    float* __restrict__ p0, float* __restrict__ p1) {
  __shared__ float lds0[BN_THREADS][8];
  __shared__ float lds1[BN_THREADS][8];
  const int t = threadIdx.x;
  const int G = C / 8;  // distinct channel sets per block
  const long long b = blockIdx.x;
  if ((G & (G - 1)) == 0 && G <= BN_THREADS) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      lds0[t][j] = acc0[j];
      lds1[t][j] = acc1[j];
    }
    __syncthreads();
    for (int s = BN_THREADS / 2; s >= G; s >>= 1) {
      if (t < s) {
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          lds0[t][j] += lds0[t + s][j];
          lds1[t][j] += lds1[t + s][j];
        }
      }
      __syncthreads();
    }
    if (t < G) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        p0[(long long)(t * 8 + j) * NB + b] = lds0[t][j];
        p1[(long long)(t * 8 + j) * NB + b] = lds1[t][j];
      }
    }
  }
  // non-power-of-2 C never reaches the fused path (gated in
  // ops/fused_bn.py::_use_fused); (void)c silences unused in that branch
  (void)c;
}

// wave64 + LDS block sum used by bn_fold_partials
__device__ __forceinline__ float block_reduce_sum2(float val) {
  for (int off = 32; off > 0; off >>= 1)
    val += __shfl_down(val, off, 64);
  __shared__ float lds[BN_THREADS / 64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  float out = 0.f;
  if (threadIdx.x == 0) {
    #pragma unroll
    for (int i = 0; i < BN_THREADS / 64; ++i) out += lds[i];
  }
  __syncthreads();
  return out;
}

// Fold NB partials per channel: block c sums p[c*NB .. c*NB+NB) (float4
// coalesced: 256 threads x float4 covers NB=1024 in one load).
extern "C" __global__ __launch_bounds__(BN_THREADS)
void bn_fold_partials(const float* __restrict__ p0,
                      const float* __restrict__ p1, int NB, int C,
                      float* __restrict__ out0, float* __restrict__ out1) {
  const int c = blockIdx.x;
  float s0 = 0.f, s1 = 0.f;
  const long long base = (long long)c * NB;
  for (int i = threadIdx.x * 4; i < NB; i += BN_THREADS * 4) {
    const float4 a = *(const float4*)(p0 + base + i);
    const float4 b = *(const float4*)(p1 + base + i);
    s0 += a.x + a.y + a.z + a.w;
    s1 += b.x + b.y + b.z + b.w;
  }
  s0 = block_reduce_sum2(s0);
  s1 = block_reduce_sum2(s1);
  if (threadIdx.x == 0) {
    out0[c] = s0;
    out1[c] = s1;
  }
}

// Per-channel sum and sum-of-squares over the flat tensor (16 B/lane,
// fixed-channel stride contract like the apply kernels).  Writes one
// partial row per block into p0/p1 ([C][NB] layout); MUST be launched
// with gridDim.x == NB (idle blocks still write their zero partials).
extern "C" __global__ __launch_bounds__(BN_THREADS)
void bn_sum_partial(const uint16_t* __restrict__ x, long long total, int C,
                    int NB, float* __restrict__ p0,
                    float* __restrict__ p1) {
  const long long e0 =
      ((long long)blockIdx.x * BN_THREADS + threadIdx.x) * 8;
  const long long stride = (long long)gridDim.x * BN_THREADS * 8;
  const int c = (int)(e0 % C);
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float q[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (long long e = e0; e < total; e += stride) {
    const uint4 raw = *(const uint4*)(x + e);
    const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float a = bnb2f((uint16_t)(w[j] & 0xFFFF));
      const float b = bnb2f((uint16_t)(w[j] >> 16));
      s[j * 2] += a;     q[j * 2] = fmaf(a, a, q[j * 2]);
      s[j * 2 + 1] += b; q[j * 2 + 1] = fmaf(b, b, q[j * 2 + 1]);
    }
  }
  bn_block_fold8(s, q, C, c, NB, p0, p1);
}

// One block of >=C threads: statistics + running-stat update + the apply
// coefficients scale/shift.
extern "C" __global__
void bn_fwd_finalize(const float* __restrict__ sums,
                     const float* __restrict__ sumsqs, long long M, int C,
                     const float* __restrict__ gamma,
                     const float* __restrict__ beta,
                     float* __restrict__ running_mean,
                     float* __restrict__ running_var, float momentum,
                     float eps, int training,
                     float* __restrict__ save_mean,
                     float* __restrict__ save_inv_std,
                     float* __restrict__ scale,
                     float* __restrict__ shift) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    float mean, var;
    if (training) {
      mean = sums[c] / (float)M;
      var = fmaxf(sumsqs[c] / (float)M - mean * mean, 0.0f);
      // unbiased running var (torch semantics)
      const float unbiased = M > 1 ? var * (float)M / (float)(M - 1) : var;
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] +
                       momentum * unbiased;
    } else {
      mean = running_mean[c];
      var = running_var[c];
    }
    const float inv_std = rsqrtf(var + eps);
    save_mean[c] = mean;
    save_inv_std[c] = inv_std;
    const float sc = gamma[c] * inv_std;
    scale[c] = sc;
    shift[c] = beta[c] - mean * sc;
  }
}

// y = act(x * scale[c] + shift[c] [+ residual]); act = relu when relu != 0.
// 8 bf16 per lane.  REQUIRES the element stride (gridDim.x * BN_THREADS
// * 8) to be a multiple of C (the host wrapper guarantees it): then every
// thread touches the SAME 8 channels on every iteration, so the per-channel
// coefficients load into registers once — the previous per-iteration gather
// of 16 scalar loads made this kernel issue-bound (profiles/r02).
extern "C" __global__ __launch_bounds__(BN_THREADS)
void bn_fwd_apply_vec8(const uint16_t* __restrict__ x,
                       const uint16_t* __restrict__ residual,
                       uint16_t* __restrict__ y, long long total, int C,
                       const float* __restrict__ scale,
                       const float* __restrict__ shift, int relu) {
  const long long e0 =
      ((long long)blockIdx.x * BN_THREADS + threadIdx.x) * 8;
  if (e0 >= total) return;
  const long long stride = (long long)gridDim.x * BN_THREADS * 8;
  const int c = (int)(e0 % C);
  float sc[8], sh[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[c + j];
    sh[j] = shift[c + j];
  }
  for (long long e = e0; e < total; e += stride) {
    const uint4 raw = *(const uint4*)(x + e);
    uint4 res;
    if (residual) res = *(const uint4*)(residual + e);
    const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
    const uint32_t rw[4] = {residual ? res.x : 0, residual ? res.y : 0,
                            residual ? res.z : 0, residual ? res.w : 0};
    uint32_t out[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = fmaf(bnb2f((uint16_t)(w[j] & 0xFFFF)), sc[j * 2],
                     sh[j * 2]);
      float b = fmaf(bnb2f((uint16_t)(w[j] >> 16)), sc[j * 2 + 1],
                     sh[j * 2 + 1]);
      if (residual) {
        a += bnb2f((uint16_t)(rw[j] & 0xFFFF));
        b += bnb2f((uint16_t)(rw[j] >> 16));
      }
      if (relu) { a = fmaxf(a, 0.f); b = fmaxf(b, 0.f); }
      out[j] = (uint32_t)bnf2b(a) | ((uint32_t)bnf2b(b) << 16);
    }
    *(uint4*)(y + e) = make_uint4(out[0], out[1], out[2], out[3]);
  }
}

// ---------------------------------------------------------------- backward

// Per-channel sum(dy_eff) and sum(dy_eff * xhat); dy_eff = dy masked by
// y>0 when relu was fused.  Flat 16 B/lane layout with fixed channels per
// thread (same stride contract as the apply kernels); outputs zeroed.
extern "C" __global__ __launch_bounds__(BN_THREADS)
void bn_bwd_reduce(const uint16_t* __restrict__ dy,
                   const uint16_t* __restrict__ x,
                   const uint16_t* __restrict__ y, long long total, int C,
                   const float* __restrict__ save_mean,
                   const float* __restrict__ save_inv_std, int relu,
                   int NB, float* __restrict__ p0,
                   float* __restrict__ p1) {
  const long long e0 =
      ((long long)blockIdx.x * BN_THREADS + threadIdx.x) * 8;
  const long long stride = (long long)gridDim.x * BN_THREADS * 8;
  const int c = (int)(e0 % C);
  float mu[8], is[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = save_mean[c + j];
    is[j] = save_inv_std[c + j];
  }
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float t[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (long long e = e0; e < total; e += stride) {
    const uint4 rd = *(const uint4*)(dy + e);
    const uint4 rx = *(const uint4*)(x + e);
    uint4 ry4;
    if (relu) ry4 = *(const uint4*)(y + e);
    const uint32_t wd[4] = {rd.x, rd.y, rd.z, rd.w};
    const uint32_t wx[4] = {rx.x, rx.y, rx.z, rx.w};
    const uint32_t wy[4] = {relu ? ry4.x : 0, relu ? ry4.y : 0,
                            relu ? ry4.z : 0, relu ? ry4.w : 0};
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float d0 = bnb2f((uint16_t)(wd[j] & 0xFFFF));
      float d1 = bnb2f((uint16_t)(wd[j] >> 16));
      if (relu) {
        if (bnb2f((uint16_t)(wy[j] & 0xFFFF)) <= 0.f) d0 = 0.f;
        if (bnb2f((uint16_t)(wy[j] >> 16)) <= 0.f) d1 = 0.f;
      }
      const float xh0 =
          (bnb2f((uint16_t)(wx[j] & 0xFFFF)) - mu[j * 2]) * is[j * 2];
      const float xh1 =
          (bnb2f((uint16_t)(wx[j] >> 16)) - mu[j * 2 + 1]) * is[j * 2 + 1];
      s[j * 2] += d0;     t[j * 2] = fmaf(d0, xh0, t[j * 2]);
      s[j * 2 + 1] += d1; t[j * 2 + 1] = fmaf(d1, xh1, t[j * 2 + 1]);
    }
  }
  bn_block_fold8(s, t, C, c, NB, p0, p1);
}

// dgamma/dbeta + the three dx coefficients:
//   dx = a[c]*dy_eff + b[c] + d[c]*(x - mean[c])
//   a = gamma*inv_std, b = -a*sum_dy/M, d = -a*inv_std^2*sum_dy_xhat/M...
// derivation: dx = a*(dy_eff - sum_dy/M - xhat*sum_dy_xhat/M),
//   xhat = (x-mean)*inv_std  =>  d = -a*inv_std*sum_dy_xhat/M.
extern "C" __global__
void bn_bwd_finalize(const float* __restrict__ sum_dy,
                     const float* __restrict__ sum_dy_xhat, long long M,
                     int C, const float* __restrict__ gamma,
                     const float* __restrict__ save_inv_std,
                     float* __restrict__ dgamma, float* __restrict__ dbeta,
                     float* __restrict__ coef_a, float* __restrict__ coef_b,
                     float* __restrict__ coef_d) {
  for (int c = threadIdx.x + blockIdx.x * blockDim.x; c < C;
       c += blockDim.x * gridDim.x) {
    const float inv_std = save_inv_std[c];
    dgamma[c] = sum_dy_xhat[c];
    dbeta[c] = sum_dy[c];
    const float a = gamma[c] * inv_std;
    coef_a[c] = a;
    coef_b[c] = -a * sum_dy[c] / (float)M;
    coef_d[c] = -a * inv_std * sum_dy_xhat[c] / (float)M;
  }
}

// dx (+ optional residual grad = dy_eff).  8 bf16/lane; same fixed-channel
// stride contract as bn_fwd_apply_vec8 — all per-channel coefficients live
// in registers across the grid-stride loop.
extern "C" __global__ __launch_bounds__(BN_THREADS)
void bn_bwd_apply_vec8(const uint16_t* __restrict__ dy,
                       const uint16_t* __restrict__ x,
                       const uint16_t* __restrict__ y,
                       uint16_t* __restrict__ dx,
                       uint16_t* __restrict__ dres, long long total, int C,
                       const float* __restrict__ save_mean,
                       const float* __restrict__ coef_a,
                       const float* __restrict__ coef_b,
                       const float* __restrict__ coef_d, int relu) {
  const long long e0 =
      ((long long)blockIdx.x * BN_THREADS + threadIdx.x) * 8;
  if (e0 >= total) return;
  const long long stride = (long long)gridDim.x * BN_THREADS * 8;
  const int c = (int)(e0 % C);
  float mu[8], ca[8], cb[8], cd[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = save_mean[c + j];
    ca[j] = coef_a[c + j];
    cb[j] = coef_b[c + j];
    cd[j] = coef_d[c + j];
  }
  for (long long e = e0; e < total; e += stride) {
    const uint4 rd = *(const uint4*)(dy + e);
    const uint4 rx = *(const uint4*)(x + e);
    uint4 ry4;
    if (relu) ry4 = *(const uint4*)(y + e);
    const uint32_t wd[4] = {rd.x, rd.y, rd.z, rd.w};
    const uint32_t wx[4] = {rx.x, rx.y, rx.z, rx.w};
    const uint32_t wy[4] = {relu ? ry4.x : 0, relu ? ry4.y : 0,
                            relu ? ry4.z : 0, relu ? ry4.w : 0};
    uint32_t odx[4], odr[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float d0 = bnb2f((uint16_t)(wd[j] & 0xFFFF));
      float d1 = bnb2f((uint16_t)(wd[j] >> 16));
      if (relu) {
        if (bnb2f((uint16_t)(wy[j] & 0xFFFF)) <= 0.f) d0 = 0.f;
        if (bnb2f((uint16_t)(wy[j] >> 16)) <= 0.f) d1 = 0.f;
      }
      const float x0 = bnb2f((uint16_t)(wx[j] & 0xFFFF));
      const float x1 = bnb2f((uint16_t)(wx[j] >> 16));
      const float r0 = fmaf(cd[j * 2], x0 - mu[j * 2],
                            fmaf(ca[j * 2], d0, cb[j * 2]));
      const float r1 = fmaf(cd[j * 2 + 1], x1 - mu[j * 2 + 1],
                            fmaf(ca[j * 2 + 1], d1, cb[j * 2 + 1]));
      odx[j] = (uint32_t)bnf2b(r0) | ((uint32_t)bnf2b(r1) << 16);
      if (dres)
        odr[j] = (uint32_t)bnf2b(d0) | ((uint32_t)bnf2b(d1) << 16);
    }
    *(uint4*)(dx + e) = make_uint4(odx[0], odx[1], odx[2], odx[3]);
    if (dres)
      *(uint4*)(dres + e) = make_uint4(odr[0], odr[1], odr[2], odr[3]);
  }
}

// ------------------------------------------------- host launch wrappers

// Max partial rows for the two-stage reduction (allocation layout size).
// The actual row count adapts: no more blocks than the sweep needs, and
// capped so the partial buffer stays <= ~4 MB for large C (2*C*NB*4B).
// Any power-of-2 NB keeps the fixed-channel stride contract because every
// power-of-2 C <= 2048 divides BN_THREADS*8.
#define BN_NB 1024

static inline int bn_pick_nb(long long total, int C) {
  long long nb = (total / 8 + BN_THREADS - 1) / BN_THREADS;
  long long cap = (4LL << 20) / (8LL * C);  // partials <= 4 MB
  if (nb > cap) nb = cap;
  if (nb > BN_NB) nb = BN_NB;
  if (nb < 64) nb = 64;
  // round down to a multiple of 4 (fold kernel reads float4 rows)
  nb &= ~3LL;
  return (int)nb;
}

static inline long long bn_gcd(long long a, long long b) {
  while (b) { long long t = a % b; a = b; b = t; }
  return a;
}

// grid for the vec8 apply kernels; the element stride grid*BN_THREADS*8
// must be a multiple of C (fixed-channel contract).  For ResNet's
// power-of-2 channel counts any grid works; general C rounds up to a
// multiple of C/gcd(C, 2048).
static inline int bn_elem_grid(long long total, int C) {
  long long g = (total / 8 + BN_THREADS - 1) / BN_THREADS;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  const long long m = C / bn_gcd((long long)C, (long long)BN_THREADS * 8);
  g = ((g + m - 1) / m) * m;
  return (int)g;
}

extern "C" void launch_bn_fwd(
    const void* x, const void* residual, void* y, long long M, int C,
    const void* gamma, const void* beta, void* running_mean,
    void* running_var, float momentum, float eps, int training, int relu,
    void* workspace,  // 6*C floats: sums, sumsqs, save_mean, save_inv_std,
                      // scale, shift
    void* partials,   // 2 * C * BN_NB floats (training only)
    hipStream_t stream) {
  float* ws = (float*)workspace;
  float* sums = ws;
  float* sumsqs = ws + C;
  float* save_mean = ws + 2 * C;
  float* save_inv_std = ws + 3 * C;
  float* scale = ws + 4 * C;
  float* shift = ws + 5 * C;
  if (training) {
    const int nb = bn_pick_nb(M * (long long)C, C);
    float* p0 = (float*)partials;
    float* p1 = p0 + (long long)C * BN_NB;  // fixed allocation layout
    hipLaunchKernelGGL(bn_sum_partial, dim3(nb), dim3(BN_THREADS), 0,
                       stream, (const uint16_t*)x, M * (long long)C, C,
                       nb, p0, p1);
    hipLaunchKernelGGL(bn_fold_partials, dim3(C), dim3(BN_THREADS), 0,
                       stream, p0, p1, nb, C, sums, sumsqs);
  }
  hipLaunchKernelGGL(bn_fwd_finalize, dim3((C + 255) / 256), dim3(256), 0,
                     stream, sums, sumsqs, M, C, (const float*)gamma,
                     (const float*)beta, (float*)running_mean,
                     (float*)running_var, momentum, eps, training,
                     save_mean, save_inv_std, scale, shift);
  hipLaunchKernelGGL(bn_fwd_apply_vec8, dim3(bn_elem_grid(M * (long long)C, C)),
                     dim3(BN_THREADS), 0, stream, (const uint16_t*)x,
                     (const uint16_t*)residual, (uint16_t*)y,
                     M * (long long)C, C, scale, shift, relu);
}

extern "C" void launch_bn_bwd(
    const void* dy, const void* x, const void* y, void* dx, void* dres,
    long long M, int C, const void* gamma, const void* save_mean,
    const void* save_inv_std, int relu, void* dgamma, void* dbeta,
    void* workspace,  // 5*C floats: sum_dy, sum_dy_xhat, coef_a/b/d
    void* partials,   // 2 * C * BN_NB floats
    hipStream_t stream) {
  float* ws = (float*)workspace;
  float* sum_dy = ws;
  float* sum_dy_xhat = ws + C;
  float* coef_a = ws + 2 * C;
  float* coef_b = ws + 3 * C;
  float* coef_d = ws + 4 * C;
  const int nb = bn_pick_nb(M * (long long)C, C);
  float* p0 = (float*)partials;
  float* p1 = p0 + (long long)C * BN_NB;  // fixed allocation layout
  hipLaunchKernelGGL(bn_bwd_reduce, dim3(nb), dim3(BN_THREADS), 0,
                     stream, (const uint16_t*)dy, (const uint16_t*)x,
                     (const uint16_t*)y, M * (long long)C, C,
                     (const float*)save_mean, (const float*)save_inv_std,
                     relu, nb, p0, p1);
  hipLaunchKernelGGL(bn_fold_partials, dim3(C), dim3(BN_THREADS), 0,
                     stream, p0, p1, nb, C, sum_dy, sum_dy_xhat);
  hipLaunchKernelGGL(bn_bwd_finalize, dim3((C + 255) / 256), dim3(256), 0,
                     stream, sum_dy, sum_dy_xhat, M, C, (const float*)gamma,
                     (const float*)save_inv_std, (float*)dgamma,
                     (float*)dbeta, coef_a, coef_b, coef_d);
  hipLaunchKernelGGL(bn_bwd_apply_vec8, dim3(bn_elem_grid(M * (long long)C, C)),
                     dim3(BN_THREADS), 0, stream, (const uint16_t*)dy,
                     (const uint16_t*)x, (const uint16_t*)y, (uint16_t*)dx,
                     (uint16_t*)dres, M * (long long)C, C,
                     (const float*)save_mean, coef_a, coef_b, coef_d, relu);
}
