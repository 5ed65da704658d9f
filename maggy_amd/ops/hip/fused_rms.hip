// Fused RMSNorm for row-contiguous bf16 [R, D], MI355X (gfx950).
//
// Replaces the eager fp32 RMSNorm of the Llama path (x.float() -> pow ->
// mean -> rsqrt -> two muls -> cast back: ~6 full-tensor sweeps in fp32)
// with a single bf16 read + bf16 write per direction, fp32 accumulation.
//
//   fwd: one workgroup per row; x lives in registers between the
//        sum-of-squares reduce and the normalize+scale write; saves
//        inv_rms[R] (fp32) for backward.
//   bwd: grid-stride over rows (NB blocks); per row
//        dx = s * (w*dy - xhat * mean_d(dy*w*xhat)), and each block
//        accumulates its dgamma partial in registers (fixed columns per
//        thread), writing one partial row -> rms_fold sums them.
//
// Constraints: D % (256*8) == 0 handled via K = D/2048 vec8 chunks per
// thread, K in [1, RMS_MAXK]; D power-of-2-multiple of 2048 covers the
// Llama configs (2048, 4096, 8192).
#include <hip/hip_runtime.h>
#include <cstdint>

#define RMS_THREADS 256
#define RMS_MAXK 4
#define RMS_NB 1024  // dgamma partial rows

__device__ __forceinline__ float rb2f(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t rf2b(float f) {
  union { uint32_t u; float f; } v;
  v.f = f;
  uint32_t u = v.u + (0x7FFF + ((v.u >> 16) & 1));
  return (uint16_t)(u >> 16);
}

__device__ __forceinline__ float rms_block_sum(float val) {
  for (int off = 32; off > 0; off >>= 1)
    val += __shfl_down(val, off, 64);
  __shared__ float lds[RMS_THREADS / 64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  float out = lds[0] + lds[1] + lds[2] + lds[3];
  __syncthreads();
  return out;  // every thread gets the sum
}

// ---------------------------------------------------------------- forward

extern "C" __global__ __launch_bounds__(RMS_THREADS)
void rms_fwd(const uint16_t* __restrict__ x, const float* __restrict__ w,
             uint16_t* __restrict__ y, float* __restrict__ inv_rms,
             long long R, int D, float eps) {
  const int K = D / (RMS_THREADS * 8);
  const long long row = blockIdx.x;
  if (row >= R) return;
  const uint16_t* xr = x + row * D;
  uint16_t* yr = y + row * D;
  float xv[RMS_MAXK][8];
  float ss = 0.f;
  #pragma unroll
  for (int k = 0; k < RMS_MAXK; ++k) {
    if (k >= K) break;
    const int base = (threadIdx.x + k * RMS_THREADS) * 8;
    const uint4 raw = *(const uint4*)(xr + base);
    const uint32_t wd[4] = {raw.x, raw.y, raw.z, raw.w};
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float a = rb2f((uint16_t)(wd[j] & 0xFFFF));
      const float b = rb2f((uint16_t)(wd[j] >> 16));
      xv[k][j * 2] = a;
      xv[k][j * 2 + 1] = b;
      ss = fmaf(a, a, ss);
      ss = fmaf(b, b, ss);
    }
  }
  ss = rms_block_sum(ss);
  const float s = rsqrtf(ss / (float)D + eps);
  if (threadIdx.x == 0) inv_rms[row] = s;
  #pragma unroll
  for (int k = 0; k < RMS_MAXK; ++k) {
    if (k >= K) break;
    const int base = (threadIdx.x + k * RMS_THREADS) * 8;
    uint32_t out[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float a = xv[k][j * 2] * s * w[base + j * 2];
      const float b = xv[k][j * 2 + 1] * s * w[base + j * 2 + 1];
      out[j] = (uint32_t)rf2b(a) | ((uint32_t)rf2b(b) << 16);
    }
    *(uint4*)(yr + base) = make_uint4(out[0], out[1], out[2], out[3]);
  }
}

// ---------------------------------------------------------------- backward

extern "C" __global__ __launch_bounds__(RMS_THREADS)
void rms_bwd(const uint16_t* __restrict__ dy,
             const uint16_t* __restrict__ x, const float* __restrict__ w,
             const float* __restrict__ inv_rms, uint16_t* __restrict__ dx,
             float* __restrict__ dw_partials,  // [D][NB]
             long long R, int D, int NB) {
  const int K = D / (RMS_THREADS * 8);
  float wv[RMS_MAXK][8];
  float dwacc[RMS_MAXK][8];
  #pragma unroll
  for (int k = 0; k < RMS_MAXK; ++k) {
    if (k >= K) break;
    const int base = (threadIdx.x + k * RMS_THREADS) * 8;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      wv[k][j] = w[base + j];
      dwacc[k][j] = 0.f;
    }
  }
  for (long long row = blockIdx.x; row < R; row += gridDim.x) {
    const uint16_t* dyr = dy + row * D;
    const uint16_t* xr = x + row * D;
    uint16_t* dxr = dx + row * D;
    const float s = inv_rms[row];
    float dyv[RMS_MAXK][8], xhv[RMS_MAXK][8];
    float dot = 0.f;
    #pragma unroll
    for (int k = 0; k < RMS_MAXK; ++k) {
      if (k >= K) break;
      const int base = (threadIdx.x + k * RMS_THREADS) * 8;
      const uint4 rdy = *(const uint4*)(dyr + base);
      const uint4 rx = *(const uint4*)(xr + base);
      const uint32_t wd[4] = {rdy.x, rdy.y, rdy.z, rdy.w};
      const uint32_t wx[4] = {rx.x, rx.y, rx.z, rx.w};
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float d0 = rb2f((uint16_t)(wd[j] & 0xFFFF));
        const float d1 = rb2f((uint16_t)(wd[j] >> 16));
        const float xh0 = rb2f((uint16_t)(wx[j] & 0xFFFF)) * s;
        const float xh1 = rb2f((uint16_t)(wx[j] >> 16)) * s;
        dyv[k][j * 2] = d0;     xhv[k][j * 2] = xh0;
        dyv[k][j * 2 + 1] = d1; xhv[k][j * 2 + 1] = xh1;
        dot = fmaf(d0 * wv[k][j * 2], xh0, dot);
        dot = fmaf(d1 * wv[k][j * 2 + 1], xh1, dot);
        dwacc[k][j * 2] = fmaf(d0, xh0, dwacc[k][j * 2]);
        dwacc[k][j * 2 + 1] = fmaf(d1, xh1, dwacc[k][j * 2 + 1]);
      }
    }
    const float c = rms_block_sum(dot) / (float)D;
    #pragma unroll
    for (int k = 0; k < RMS_MAXK; ++k) {
      if (k >= K) break;
      const int base = (threadIdx.x + k * RMS_THREADS) * 8;
      uint32_t out[4];
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float r0 = s * fmaf(wv[k][j * 2], dyv[k][j * 2],
                                  -xhv[k][j * 2] * c);
        const float r1 = s * fmaf(wv[k][j * 2 + 1], dyv[k][j * 2 + 1],
                                  -xhv[k][j * 2 + 1] * c);
        out[j] = (uint32_t)rf2b(r0) | ((uint32_t)rf2b(r1) << 16);
      }
      *(uint4*)(dxr + base) = make_uint4(out[0], out[1], out[2], out[3]);
    }
  }
  // one dgamma partial row per block
  #pragma unroll
  for (int k = 0; k < RMS_MAXK; ++k) {
    if (k >= K) break;
    const int base = (threadIdx.x + k * RMS_THREADS) * 8;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      dw_partials[(long long)(base + j) * NB + blockIdx.x] = dwacc[k][j];
    }
  }
}

// fold the [D][NB] dgamma partials: one block per column
extern "C" __global__ __launch_bounds__(RMS_THREADS)
void rms_fold(const float* __restrict__ partials, int NB, int D,
              float* __restrict__ dw) {
  const int d = blockIdx.x;
  float acc = 0.f;
  const long long base = (long long)d * NB;
  for (int i = threadIdx.x * 4; i < NB; i += RMS_THREADS * 4) {
    const float4 a = *(const float4*)(partials + base + i);
    acc += a.x + a.y + a.z + a.w;
  }
  acc = rms_block_sum(acc);
  if (threadIdx.x == 0) dw[d] = acc;
}

// ---------------------------------------------------------------- SwiGLU

// out = silu(g) * u, 8 bf16/lane (the eager path is 3 sweeps + sigmoid
// kernels; this is one read-pair + one write)
extern "C" __global__ __launch_bounds__(RMS_THREADS)
void swiglu_fwd(const uint16_t* __restrict__ g,
                const uint16_t* __restrict__ u, uint16_t* __restrict__ out,
                long long total) {
  const long long stride = (long long)gridDim.x * RMS_THREADS * 8;
  for (long long e = ((long long)blockIdx.x * RMS_THREADS + threadIdx.x)
                     * 8;
       e < total; e += stride) {
    const uint4 rg = *(const uint4*)(g + e);
    const uint4 ru = *(const uint4*)(u + e);
    const uint32_t wg[4] = {rg.x, rg.y, rg.z, rg.w};
    const uint32_t wu[4] = {ru.x, ru.y, ru.z, ru.w};
    uint32_t o[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float g0 = rb2f((uint16_t)(wg[j] & 0xFFFF));
      const float g1 = rb2f((uint16_t)(wg[j] >> 16));
      const float u0 = rb2f((uint16_t)(wu[j] & 0xFFFF));
      const float u1 = rb2f((uint16_t)(wu[j] >> 16));
      const float s0 = g0 / (1.f + __expf(-g0));
      const float s1 = g1 / (1.f + __expf(-g1));
      o[j] = (uint32_t)rf2b(s0 * u0) | ((uint32_t)rf2b(s1 * u1) << 16);
    }
    *(uint4*)(out + e) = make_uint4(o[0], o[1], o[2], o[3]);
  }
}

// dg = dy*u*sig*(1 + g*(1-sig)); du = dy*silu(g)
extern "C" __global__ __launch_bounds__(RMS_THREADS)
void swiglu_bwd(const uint16_t* __restrict__ dy,
                const uint16_t* __restrict__ g,
                const uint16_t* __restrict__ u,
                uint16_t* __restrict__ dg, uint16_t* __restrict__ du,
                long long total) {
  const long long stride = (long long)gridDim.x * RMS_THREADS * 8;
  for (long long e = ((long long)blockIdx.x * RMS_THREADS + threadIdx.x)
                     * 8;
       e < total; e += stride) {
    const uint4 rd = *(const uint4*)(dy + e);
    const uint4 rg = *(const uint4*)(g + e);
    const uint4 ru = *(const uint4*)(u + e);
    const uint32_t wd[4] = {rd.x, rd.y, rd.z, rd.w};
    const uint32_t wg[4] = {rg.x, rg.y, rg.z, rg.w};
    const uint32_t wu[4] = {ru.x, ru.y, ru.z, ru.w};
    uint32_t og[4], ou[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float d0 = rb2f((uint16_t)(wd[j] & 0xFFFF));
      const float d1 = rb2f((uint16_t)(wd[j] >> 16));
      const float g0 = rb2f((uint16_t)(wg[j] & 0xFFFF));
      const float g1 = rb2f((uint16_t)(wg[j] >> 16));
      const float u0 = rb2f((uint16_t)(wu[j] & 0xFFFF));
      const float u1 = rb2f((uint16_t)(wu[j] >> 16));
      const float s0 = 1.f / (1.f + __expf(-g0));
      const float s1 = 1.f / (1.f + __expf(-g1));
      const float dg0 = d0 * u0 * s0 * fmaf(g0, 1.f - s0, 1.f);
      const float dg1 = d1 * u1 * s1 * fmaf(g1, 1.f - s1, 1.f);
      og[j] = (uint32_t)rf2b(dg0) | ((uint32_t)rf2b(dg1) << 16);
      ou[j] = (uint32_t)rf2b(d0 * g0 * s0) |
              ((uint32_t)rf2b(d1 * g1 * s1) << 16);
    }
    *(uint4*)(dg + e) = make_uint4(og[0], og[1], og[2], og[3]);
    *(uint4*)(du + e) = make_uint4(ou[0], ou[1], ou[2], ou[3]);
  }
}

// ------------------------------------------------- host launch wrappers

static inline int rms_elem_grid(long long total) {
  long long g = (total / 8 + RMS_THREADS - 1) / RMS_THREADS;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

extern "C" void launch_swiglu_fwd(const void* g, const void* u, void* out,
                                  long long total, hipStream_t stream) {
  hipLaunchKernelGGL(swiglu_fwd, dim3(rms_elem_grid(total)),
                     dim3(RMS_THREADS), 0, stream, (const uint16_t*)g,
                     (const uint16_t*)u, (uint16_t*)out, total);
}

extern "C" void launch_swiglu_bwd(const void* dy, const void* g,
                                  const void* u, void* dg, void* du,
                                  long long total, hipStream_t stream) {
  hipLaunchKernelGGL(swiglu_bwd, dim3(rms_elem_grid(total)),
                     dim3(RMS_THREADS), 0, stream, (const uint16_t*)dy,
                     (const uint16_t*)g, (const uint16_t*)u, (uint16_t*)dg,
                     (uint16_t*)du, total);
}

extern "C" void launch_rms_fwd(const void* x, const void* w, void* y,
                               void* inv_rms, long long R, int D,
                               float eps, hipStream_t stream) {
  hipLaunchKernelGGL(rms_fwd, dim3((unsigned)R), dim3(RMS_THREADS), 0,
                     stream, (const uint16_t*)x, (const float*)w,
                     (uint16_t*)y, (float*)inv_rms, R, D, eps);
}

extern "C" void launch_rms_bwd(const void* dy, const void* x, const void* w,
                               const void* inv_rms, void* dx,
                               void* dw_partials, void* dw, long long R,
                               int D, hipStream_t stream) {
  int nb = RMS_NB;
  if (R < nb) {
    // still launch RMS_NB? no: partial rows must match fold's NB; use
    // exactly min(R, RMS_NB) blocks and tell the fold kernel
    nb = (int)R;
    nb = (nb + 3) & ~3;  // fold reads float4 rows
    if (nb > RMS_NB) nb = RMS_NB;
    if (nb < 4) nb = 4;
  }
  // blocks beyond R write zero partials only if launched; launch exactly
  // nb blocks and have rows < blocks covered by the loop guard, while
  // every block still writes its (possibly zero) dwacc row
  hipLaunchKernelGGL(rms_bwd, dim3(nb), dim3(RMS_THREADS), 0, stream,
                     (const uint16_t*)dy, (const uint16_t*)x,
                     (const float*)w, (const float*)inv_rms, (uint16_t*)dx,
                     (float*)dw_partials, R, D, nb);
  hipLaunchKernelGGL(rms_fold, dim3(D), dim3(RMS_THREADS), 0, stream,
                     (float*)dw_partials, nb, D, (float*)dw);
}

// ------------------------------------------------------------------ RoPE
// Rotary position embedding on an interleaved-pair layout, bf16 in/out,
// fp32 cos/sin tables: for row r (= flattened [B, T, H]) and pair i,
//   out[2i]   = x[2i]*cos[t][i] - sign*x[2i+1]*sin[t][i]
//   out[2i+1] = x[2i+1]*cos[t][i] + sign*x[2i]*sin[t][i]
// with t = pos + (r / H) % T.  sign=+1 is the forward rotation; sign=-1
// is its transpose = the backward (rotation by -theta).  One bf16 read +
// one write per element replaces the eager path's ~6 sliced sweeps and 8
// launches per call (profiles/r10: elementwise glue was 10% of the 8B
// step).  Each thread handles 2 pairs = one dword-aligned 8-byte x
// vector and an 8-byte cos/sin pair read.
#define ROPE_THREADS 256

extern "C" __global__ __launch_bounds__(ROPE_THREADS)
void rope_apply(const uint16_t* __restrict__ x, uint16_t* __restrict__ out,
                const float* __restrict__ cost,
                const float* __restrict__ sint,
                long long n_quads,  // total D/4-element groups = R * D/4
                int quads_per_row,  // D/4
                int H, int T, int pos, float sign) {
  const long long g0 = (long long)blockIdx.x * ROPE_THREADS + threadIdx.x;
  if (g0 >= n_quads) return;
  const long long row = g0 / quads_per_row;
  const int q = (int)(g0 % quads_per_row);     // 2 pairs per quad
  const int t = pos + (int)((row / H) % T);
  const long long base = row * (long long)quads_per_row * 4 + q * 4;
  const int cbase = t * (quads_per_row * 2) + q * 2;  // D/2 pairs per t
  // load 4 bf16 (two pairs) as one 8-byte vector
  ushort v4[4];
  *(unsigned long long*)v4 = *(const unsigned long long*)(x + base);
  float c0 = cost[cbase], s0 = sint[cbase] * sign;
  float c1 = cost[cbase + 1], s1 = sint[cbase + 1] * sign;
  union { uint32_t u; float f; } a, b;
  ushort o4[4];
  a.u = (uint32_t)v4[0] << 16; b.u = (uint32_t)v4[1] << 16;
  float r0 = a.f * c0 - b.f * s0;
  float r1 = b.f * c0 + a.f * s0;
  a.u = (uint32_t)v4[2] << 16; b.u = (uint32_t)v4[3] << 16;
  float r2 = a.f * c1 - b.f * s1;
  float r3 = b.f * c1 + a.f * s1;
  union { uint32_t u; float f; } w;
  w.f = r0; o4[0] = (ushort)((w.u + (0x7FFF + ((w.u >> 16) & 1))) >> 16);
  w.f = r1; o4[1] = (ushort)((w.u + (0x7FFF + ((w.u >> 16) & 1))) >> 16);
  w.f = r2; o4[2] = (ushort)((w.u + (0x7FFF + ((w.u >> 16) & 1))) >> 16);
  w.f = r3; o4[3] = (ushort)((w.u + (0x7FFF + ((w.u >> 16) & 1))) >> 16);
  *(unsigned long long*)(out + base) = *(unsigned long long*)o4;
}

extern "C" void launch_rope(const void* x, void* out, const void* cost,
                            const void* sint, long long rows, int D, int H,
                            int T, int pos, float sign,
                            hipStream_t stream) {
  const int qpr = D / 4;
  const long long n = rows * qpr;
  const long long blocks = (n + ROPE_THREADS - 1) / ROPE_THREADS;
  hipLaunchKernelGGL(rope_apply, dim3((unsigned)blocks), dim3(ROPE_THREADS),
                     0, stream, (const uint16_t*)x, (uint16_t*)out,
                     (const float*)cost, (const float*)sint, n, qpr, H, T,
                     pos, sign);
}
