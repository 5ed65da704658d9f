// maggy_amd HIP kernels for MI355X (gfx950, CDNA4).
//
// Replaces the accelerator-touching call sites of the reference framework
// (SURVEY.md §2.9): N7 (optimizer.step of torch.optim.* behind the ZeRO
// wrappers, /root/reference/maggy/core/patching/optim.py:54-117) becomes a
// single multi-tensor fused Adam/SGD launch with in-kernel global-norm
// clipping; N8 (grad-norm + metric reduction behind reporter.broadcast,
// /root/reference/maggy/core/reporter.py:77) becomes the hierarchical
// wave->LDS->atomic reduction kernels here.
//
// Design (per /opt/skills/guides/cdna_hip_programming.md):
//  - all kernels are memory-bound elementwise/reduction: 256-thread blocks,
//    16 B/lane vectorized access (float4 / 8 x bf16), grid-stride over a
//    flat chunk table so one launch covers every parameter tensor
//  - wave64 shuffle reductions (__shfl_down with warpSize=64), one LDS
//    round across the 4 waves of a block, one global atomic per block
//  - bf16 handled via bit ops (u16 << 16 widen, RNE pack) — hipcc does not
//    auto-vectorize scalar bf16 loads (Guideline 13)
//  - the Adam/SGD kernels read the pre-computed grad-norm from device
//    memory and derive the clip scale in-kernel: no host round-trip between
//    the norm pass and the update pass.
#include <hip/hip_runtime.h>
#include <cstdint>

#define THREADS 256
#define CHUNK 32768           // elements per chunk (table granularity)
#define MAX_GRID 2048         // grid-stride beyond this (Guideline 11)

// ---------------------------------------------------------------- helpers

__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = ((uint32_t)h) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { uint32_t u; float f; } v;
  v.f = f;
  uint32_t u = v.u;
  uint32_t rounding = 0x7FFF + ((u >> 16) & 1);   // round-to-nearest-even
  u += rounding;
  return (uint16_t)(u >> 16);
}

// wave (64-lane) + block (4-wave) sum reduction -> lane 0 of wave 0
__device__ __forceinline__ float block_reduce_sum(float val) {
  for (int off = 32; off > 0; off >>= 1)
    val += __shfl_down(val, off, 64);
  __shared__ float lds[4];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  if (wave == 0) {
    val = (lane < (THREADS / 64)) ? lds[lane] : 0.0f;
    for (int off = 2; off > 0; off >>= 1)
      val += __shfl_down(val, off, 64);
  }
  return val;  // valid in wave 0 lane 0
}

__device__ __forceinline__ float block_reduce_max(float val) {
  for (int off = 32; off > 0; off >>= 1)
    val = fmaxf(val, __shfl_down(val, off, 64));
  __shared__ float lds[4];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  if (wave == 0) {
    val = (lane < (THREADS / 64)) ? lds[lane] : -INFINITY;
    for (int off = 2; off > 0; off >>= 1)
      val = fmaxf(val, __shfl_down(val, off, 64));
  }
  return val;
}

// ------------------------------------------------------- chunk table types

// Per-tensor pointer record (filled host-side once; pointers are stable
// across steps because grads/states are pre-allocated).
struct TensorMeta {
  void* param;       // fp32 master (or fp32 param)
  void* param_lo;    // bf16 mirror of the param, nullptr in fp32 mode
  void* grad;        // bf16 or fp32 (see grad_bf16 flag per launch)
  void* exp_avg;     // fp32 m   (Adam) / momentum buf (SGD)
  void* exp_avg_sq;  // fp32 v   (Adam) / unused (SGD)
  long long numel;
};

// Per-chunk record: which tensor, element offset of the chunk.
struct ChunkMeta {
  int tensor;
  int pad;
  long long offset;
};

// --------------------------------------------------------- grad L2 norm^2

// Accumulates sum(g^2) over every chunk into norm_sq[0] (fp32).  Call with
// norm_sq zeroed.  Vectorized 16 B loads; bf16 grads widen in-register.
extern "C" __global__ __launch_bounds__(THREADS)
void multi_l2norm_sq(const ChunkMeta* __restrict__ chunks, int n_chunks,
                     const TensorMeta* __restrict__ tensors, int grad_bf16,
                     float* __restrict__ norm_sq) {
  float acc = 0.0f;
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const ChunkMeta ck = chunks[c];
    const TensorMeta tm = tensors[ck.tensor];
    const long long base = ck.offset;
    const long long n = min((long long)CHUNK, tm.numel - base);
    if (grad_bf16) {
      const uint16_t* g = (const uint16_t*)tm.grad + base;
      // 8 bf16 per lane per iteration (16 B)
      long long nv = n & ~7LL;
      for (long long i = (long long)threadIdx.x * 8; i < nv;
           i += (long long)THREADS * 8) {
        const uint4 raw = *(const uint4*)(g + i);
        const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          float a = bf16_to_f32((uint16_t)(w[j] & 0xFFFF));
          float b = bf16_to_f32((uint16_t)(w[j] >> 16));
          acc = fmaf(a, a, acc);
          acc = fmaf(b, b, acc);
        }
      }
      for (long long i = nv + threadIdx.x; i < n; i += THREADS) {
        float a = bf16_to_f32(g[i]);
        acc = fmaf(a, a, acc);
      }
    } else {
      const float* g = (const float*)tm.grad + base;
      long long nv = n & ~3LL;
      for (long long i = (long long)threadIdx.x * 4; i < nv;
           i += (long long)THREADS * 4) {
        const float4 v = *(const float4*)(g + i);
        acc = fmaf(v.x, v.x, acc);
        acc = fmaf(v.y, v.y, acc);
        acc = fmaf(v.z, v.z, acc);
        acc = fmaf(v.w, v.w, acc);
      }
      for (long long i = nv + threadIdx.x; i < n; i += THREADS) {
        acc = fmaf(g[i], g[i], acc);
      }
    }
  }
  const float total = block_reduce_sum(acc);
  if (threadIdx.x == 0) atomicAdd(norm_sq, total);
}

// ------------------------------------------------------------- fused Adam

// One launch updates every tensor: m/v update, bias correction, optional
// decoupled weight decay (AdamW), optional global-norm clip from the
// device-resident norm_sq, bf16 param mirror write.  All state fp32.
extern "C" __global__ __launch_bounds__(THREADS)
void multi_fused_adam(const ChunkMeta* __restrict__ chunks, int n_chunks,
                      const TensorMeta* __restrict__ tensors, int grad_bf16,
                      float lr, float beta1, float beta2, float eps,
                      float weight_decay, float bc1, float bc2,
                      const float* __restrict__ norm_sq, float max_norm,
                      float grad_scale_inv) {
  // clip scale: min(1, max_norm / ||g||); norm_sq already in unscaled units
  float clip = 1.0f;
  if (norm_sq != nullptr && max_norm > 0.0f) {
    const float nrm = sqrtf(*norm_sq) * grad_scale_inv;
    if (nrm > max_norm) clip = max_norm / (nrm + 1e-6f);
  }
  const float gscale = clip * grad_scale_inv;
  const float step_size = lr / bc1;   // lr * sqrt(bc2)/bc1 applied via denom
  const float inv_sqrt_bc2 = rsqrtf(bc2);

  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const ChunkMeta ck = chunks[c];
    const TensorMeta tm = tensors[ck.tensor];
    const long long base = ck.offset;
    const long long n = min((long long)CHUNK, tm.numel - base);
    float* p = (float*)tm.param + base;
    float* m = (float*)tm.exp_avg + base;
    float* v = (float*)tm.exp_avg_sq + base;
    uint16_t* plo =
        tm.param_lo ? (uint16_t*)tm.param_lo + base : nullptr;

    for (long long i = threadIdx.x; i < n; i += THREADS) {
      float g;
      if (grad_bf16) g = bf16_to_f32(((const uint16_t*)tm.grad + base)[i]);
      else           g = ((const float*)tm.grad + base)[i];
      g *= gscale;
      float pi = p[i];
      if (weight_decay != 0.0f) pi -= lr * weight_decay * pi;  // AdamW
      float mi = fmaf(beta1, m[i], (1.0f - beta1) * g);
      float vi = fmaf(beta2, v[i], (1.0f - beta2) * g * g);
      m[i] = mi;
      v[i] = vi;
      // denom = sqrt(v/bc2) + eps
      const float denom = sqrtf(vi) * inv_sqrt_bc2 + eps;
      pi -= step_size * mi / denom;
      p[i] = pi;
      if (plo) plo[i] = f32_to_bf16(pi);
    }
  }
}

// -------------------------------------------------------------- fused SGD

extern "C" __global__ __launch_bounds__(THREADS)
void multi_fused_sgd(const ChunkMeta* __restrict__ chunks, int n_chunks,
                     const TensorMeta* __restrict__ tensors, int grad_bf16,
                     float lr, float momentum, float weight_decay,
                     float dampening, int nesterov, int first_step,
                     const float* __restrict__ norm_sq, float max_norm,
                     float grad_scale_inv) {
  float clip = 1.0f;
  if (norm_sq != nullptr && max_norm > 0.0f) {
    const float nrm = sqrtf(*norm_sq) * grad_scale_inv;
    if (nrm > max_norm) clip = max_norm / (nrm + 1e-6f);
  }
  const float gscale = clip * grad_scale_inv;

  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const ChunkMeta ck = chunks[c];
    const TensorMeta tm = tensors[ck.tensor];
    const long long base = ck.offset;
    const long long n = min((long long)CHUNK, tm.numel - base);
    float* p = (float*)tm.param + base;
    float* buf = (float*)tm.exp_avg + base;  // momentum buffer
    uint16_t* plo =
        tm.param_lo ? (uint16_t*)tm.param_lo + base : nullptr;

    for (long long i = threadIdx.x; i < n; i += THREADS) {
      float g;
      if (grad_bf16) g = bf16_to_f32(((const uint16_t*)tm.grad + base)[i]);
      else           g = ((const float*)tm.grad + base)[i];
      g *= gscale;
      float pi = p[i];
      if (weight_decay != 0.0f) g = fmaf(weight_decay, pi, g);
      if (momentum != 0.0f) {
        float b = first_step ? g
                             : fmaf(momentum, buf[i], (1.0f - dampening) * g);
        buf[i] = b;
        g = nesterov ? fmaf(momentum, b, g) : b;
      }
      pi -= lr * g;
      p[i] = pi;
      if (plo) plo[i] = f32_to_bf16(pi);
    }
  }
}

// -------------------------------------------------- scalar metric reduce

// sum / max over one contiguous tensor (fp32 or bf16) -> out[0] (fp32).
// Used by reporter.broadcast(tensor) and test checks. out must be zeroed
// (sum) or set to -inf (max) by the caller.
extern "C" __global__ __launch_bounds__(THREADS)
void reduce_sum_f32(const float* __restrict__ in, long long n,
                    float* __restrict__ out) {
  float acc = 0.0f;
  const long long nv = n & ~3LL;
  const long long stride = (long long)gridDim.x * THREADS * 4;
  for (long long i = ((long long)blockIdx.x * THREADS + threadIdx.x) * 4;
       i < nv; i += stride) {
    const float4 v = *(const float4*)(in + i);
    acc += v.x + v.y + v.z + v.w;
  }
  for (long long i = nv + blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long long)gridDim.x * THREADS)
    acc += in[i];
  const float total = block_reduce_sum(acc);
  if (threadIdx.x == 0) atomicAdd(out, total);
}

extern "C" __global__ __launch_bounds__(THREADS)
void reduce_sum_bf16(const uint16_t* __restrict__ in, long long n,
                     float* __restrict__ out) {
  float acc = 0.0f;
  const long long nv = n & ~7LL;
  const long long stride = (long long)gridDim.x * THREADS * 8;
  for (long long i = ((long long)blockIdx.x * THREADS + threadIdx.x) * 8;
       i < nv; i += stride) {
    const uint4 raw = *(const uint4*)(in + i);
    const uint32_t w[4] = {raw.x, raw.y, raw.z, raw.w};
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      acc += bf16_to_f32((uint16_t)(w[j] & 0xFFFF)) +
             bf16_to_f32((uint16_t)(w[j] >> 16));
  }
  for (long long i = nv + blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long long)gridDim.x * THREADS)
    acc += bf16_to_f32(in[i]);
  const float total = block_reduce_sum(acc);
  if (threadIdx.x == 0) atomicAdd(out, total);
}

__device__ __forceinline__ float atomicMaxFloat(float* addr, float val) {
  // monotone fp32 max via integer CAS-free trick (positive/negative split)
  int* iaddr = (int*)addr;
  if (val >= 0)
    return __int_as_float(atomicMax(iaddr, __float_as_int(val)));
  return __uint_as_float(
      atomicMin((unsigned int*)addr, __float_as_uint(val)));
}

extern "C" __global__ __launch_bounds__(THREADS)
void reduce_max_f32(const float* __restrict__ in, long long n,
                    float* __restrict__ out) {
  float acc = -INFINITY;
  for (long long i = (long long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long long)gridDim.x * THREADS)
    acc = fmaxf(acc, in[i]);
  const float total = block_reduce_max(acc);
  if (threadIdx.x == 0) atomicMaxFloat(out, total);
}

// ------------------------------------------------- host launch wrappers
// (kept in this TU so <<< >>> launches are device-compiled; the pybind
// bindings call these with the current torch stream)

static inline int pick_grid(long long work_items) {
  long long g = (work_items + THREADS - 1) / THREADS;
  if (g < 1) g = 1;
  if (g > MAX_GRID) g = MAX_GRID;
  return (int)g;
}

extern "C" void launch_multi_l2norm_sq(const void* chunks, int n_chunks,
                                       const void* tensors, int grad_bf16,
                                       float* norm_sq, hipStream_t stream) {
  int grid = n_chunks < MAX_GRID ? (n_chunks < 1 ? 1 : n_chunks) : MAX_GRID;
  hipLaunchKernelGGL(multi_l2norm_sq, dim3(grid), dim3(THREADS), 0, stream,
                     (const ChunkMeta*)chunks, n_chunks,
                     (const TensorMeta*)tensors, grad_bf16, norm_sq);
}

extern "C" void launch_multi_fused_adam(
    const void* chunks, int n_chunks, const void* tensors, int grad_bf16,
    float lr, float beta1, float beta2, float eps, float weight_decay,
    float bc1, float bc2, const float* norm_sq, float max_norm,
    float grad_scale_inv, hipStream_t stream) {
  int grid = n_chunks < MAX_GRID ? (n_chunks < 1 ? 1 : n_chunks) : MAX_GRID;
  hipLaunchKernelGGL(multi_fused_adam, dim3(grid), dim3(THREADS), 0, stream,
                     (const ChunkMeta*)chunks, n_chunks,
                     (const TensorMeta*)tensors, grad_bf16, lr, beta1, beta2,
                     eps, weight_decay, bc1, bc2, norm_sq, max_norm,
                     grad_scale_inv);
}

extern "C" void launch_multi_fused_sgd(
    const void* chunks, int n_chunks, const void* tensors, int grad_bf16,
    float lr, float momentum, float weight_decay, float dampening,
    int nesterov, int first_step, const float* norm_sq, float max_norm,
    float grad_scale_inv, hipStream_t stream) {
  int grid = n_chunks < MAX_GRID ? (n_chunks < 1 ? 1 : n_chunks) : MAX_GRID;
  hipLaunchKernelGGL(multi_fused_sgd, dim3(grid), dim3(THREADS), 0, stream,
                     (const ChunkMeta*)chunks, n_chunks,
                     (const TensorMeta*)tensors, grad_bf16, lr, momentum,
                     weight_decay, dampening, nesterov, first_step, norm_sq,
                     max_norm, grad_scale_inv);
}

extern "C" void launch_reduce_sum_f32(const float* in, long long n,
                                      float* out, hipStream_t stream) {
  hipLaunchKernelGGL(reduce_sum_f32, dim3(pick_grid(n / 4)), dim3(THREADS),
                     0, stream, in, n, out);
}

extern "C" void launch_reduce_sum_bf16(const void* in, long long n,
                                       float* out, hipStream_t stream) {
  hipLaunchKernelGGL(reduce_sum_bf16, dim3(pick_grid(n / 8)), dim3(THREADS),
                     0, stream, (const uint16_t*)in, n, out);
}

extern "C" void launch_reduce_max_f32(const float* in, long long n,
                                      float* out, hipStream_t stream) {
  hipLaunchKernelGGL(reduce_max_f32, dim3(pick_grid(n)), dim3(THREADS), 0,
                     stream, in, n, out);
}
