// Standalone repro: why is the BN reduce loop 4x slower than the apply loop?
// Variants of the same 16B/lane grid-stride sweep over a bf16 tensor:
//   v0: pure load + accumulate, no epilogue (baseline)
//   v1: + LDS fold8 epilogue
//   v2: + per-thread atomics epilogue (no LDS)
//   v3: + LDS fold8 + atomics (the shipped bn_sum_partial)
//   v4: apply-style: load + fma + store (for reference)
// hipcc --offload-arch=gfx950 -O3 scripts/bnrepro.hip -o bnrepro
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

#define T 256

__device__ __forceinline__ float b2f(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

template <int EPI>  // 0 none, 1 lds, 2 atomic, 3 lds+atomic
__global__ __launch_bounds__(T) void sweep(const uint16_t* __restrict__ x,
                                           long long total, int C,
                                           float* __restrict__ out0,
                                           float* __restrict__ out1) {
  const long long e0 = ((long long)blockIdx.x * T + threadIdx.x) * 8;
  const long long stride = (long long)gridDim.x * T * 8;
  const int c = (int)(e0 % C);
  float s[8] = {0}, q[8] = {0};
  for (long long e = e0; e < total; e += stride) {
    const uint4 raw = *(const uint4*)(x + e);
    const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float a = b2f((uint16_t)(w[j] & 0xFFFF));
      const float b = b2f((uint16_t)(w[j] >> 16));
      s[j * 2] += a;     q[j * 2] = fmaf(a, a, q[j * 2]);
      s[j * 2 + 1] += b; q[j * 2 + 1] = fmaf(b, b, q[j * 2 + 1]);
    }
  }
  if (EPI == 0) {
    // fold into one value and write per-block (keeps the loop alive)
    float acc = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += s[j] + q[j];
    if (threadIdx.x == 0) out0[blockIdx.x] = acc;
    return;
  }
  if (EPI == 2) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      atomicAdd(&out0[c + j], s[j]);
      atomicAdd(&out1[c + j], q[j]);
    }
    return;
  }
  __shared__ float l0[T][8];
  __shared__ float l1[T][8];
  const int t = threadIdx.x;
  const int G = C / 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) { l0[t][j] = s[j]; l1[t][j] = q[j]; }
  __syncthreads();
  for (int st = T / 2; st >= G; st >>= 1) {
    if (t < st) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        l0[t][j] += l0[t + st][j];
        l1[t][j] += l1[t + st][j];
      }
    }
    __syncthreads();
  }
  if (t < G) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (EPI == 3) {
        atomicAdd(&out0[t * 8 + j], l0[t][j]);
        atomicAdd(&out1[t * 8 + j], l1[t][j]);
      } else {
        out0[blockIdx.x % 64 * 2048 + t * 8 + j] = l0[t][j];
        out1[blockIdx.x % 64 * 2048 + t * 8 + j] = l1[t][j];
      }
    }
  }
}

__global__ __launch_bounds__(T) void apply_like(
    const uint16_t* __restrict__ x, uint16_t* __restrict__ y,
    long long total, int C, const float* __restrict__ sc) {
  const long long e0 = ((long long)blockIdx.x * T + threadIdx.x) * 8;
  if (e0 >= total) return;
  const long long stride = (long long)gridDim.x * T * 8;
  const int c = (int)(e0 % C);
  float k[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) k[j] = sc[c + j];
  for (long long e = e0; e < total; e += stride) {
    const uint4 raw = *(const uint4*)(x + e);
    uint32_t o[4];
    const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = b2f((uint16_t)(w[j] & 0xFFFF)) * k[j * 2];
      float b = b2f((uint16_t)(w[j] >> 16)) * k[j * 2 + 1];
      union { uint32_t u; float f; } ua, ub;
      ua.f = a; ub.f = b;
      o[j] = (ua.u >> 16) | (ub.u & 0xFFFF0000u);
    }
    *(uint4*)(y + e) = make_uint4(o[0], o[1], o[2], o[3]);
  }
}

static float timeit(void (*launch)(int), int grid, int iters) {
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  launch(grid);  // warm
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int i = 0; i < iters; ++i) launch(grid);
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms;
  hipEventElapsedTime(&ms, a, b);
  return ms / iters;
}

uint16_t* g_x;
uint16_t* g_y;
float* g_o0;
float* g_o1;
float* g_sc;
long long g_total;
int g_C;

template <int EPI> void launch_sweep(int grid) {
  hipLaunchKernelGGL(sweep<EPI>, dim3(grid), dim3(T), 0, 0, g_x, g_total,
                     g_C, g_o0, g_o1);
}
void launch_apply(int grid) {
  hipLaunchKernelGGL(apply_like, dim3(grid), dim3(T), 0, 0, g_x, g_y,
                     g_total, g_C, g_sc);
}

int main() {
  const struct { int C; long long rows; } shapes[] = {
      {64, 256LL * 112 * 112}, {256, 256LL * 56 * 56},
      {2048, 256LL * 7 * 7}};
  for (auto& sh : shapes) {
    g_C = sh.C;
    g_total = sh.rows * sh.C;
    hipMalloc(&g_x, g_total * 2);
    hipMalloc(&g_y, g_total * 2);
    hipMalloc(&g_o0, 64 * 2048 * 4 + 4096);
    hipMalloc(&g_o1, 64 * 2048 * 4 + 4096);
    hipMalloc(&g_sc, 4096 * 4);
    hipMemset(g_x, 0x3f, g_total * 2);
    const int grid = 2048;
    const double gb = g_total * 2.0 / 1e9;
    printf("C=%4d total=%lldM read=%.2fGB\n", g_C, g_total / 1000000, gb);
    printf("  v0 load-only   : %7.3f ms  %6.0f GB/s\n",
           timeit(launch_sweep<0>, grid, 20),
           gb / timeit(launch_sweep<0>, grid, 20) * 1e3);
    printf("  v1 +lds        : %7.3f ms  %6.0f GB/s\n",
           timeit(launch_sweep<1>, grid, 20),
           gb / timeit(launch_sweep<1>, grid, 20) * 1e3);
    printf("  v2 +atomics    : %7.3f ms  %6.0f GB/s\n",
           timeit(launch_sweep<2>, grid, 20),
           gb / timeit(launch_sweep<2>, grid, 20) * 1e3);
    printf("  v3 +lds+atomic : %7.3f ms  %6.0f GB/s\n",
           timeit(launch_sweep<3>, grid, 20),
           gb / timeit(launch_sweep<3>, grid, 20) * 1e3);
    printf("  v4 apply-like  : %7.3f ms  %6.0f GB/s (rw)\n",
           timeit(launch_apply, grid, 20),
           2 * gb / timeit(launch_apply, grid, 20) * 1e3);
    hipFree(g_x); hipFree(g_y); hipFree(g_o0); hipFree(g_o1);
    hipFree(g_sc);
  }
  return 0;
}
