// Common helpers for polyrl_amd CDNA4 (gfx950) kernels.
//
// Target: MI355X only. Wave size 64, 4 SIMD-32/CU, 160 KiB LDS/CU,
// HBM3E ~8 TB/s peak. No multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE_SIZE 64
#define DEV_INLINE __device__ __forceinline__

// grid sizing rule (guide G11): cap memory-bound grids, grid-stride the rest
#define MAX_RESIDENT_BLOCKS (256 * 8)

typedef __hip_bfloat16 bf16_t;

// 8 bf16 = 16 B: the coalescing sweet spot for memory-bound kernels (G13)
struct alignas(16) bf16x8 {
  bf16_t v[8];
};
struct alignas(8) bf16x4 {
  bf16_t v[4];
};
struct alignas(16) f32x4v {
  float v[4];
};

DEV_INLINE float bf2f(bf16_t x) { return __bfloat162float(x); }
DEV_INLINE bf16_t f2bf(float x) { return __float2bfloat16(x); }

// ---------------------------------------------------------------- reductions

// full-wave (64-lane) sum
DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

DEV_INLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// reduce within a 16-lane group (lanes l, l^1, ..., l^8)
DEV_INLINE float group16_reduce_sum(float x) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

// block reduce over NW waves through LDS; every thread returns the result.
// The final value is re-broadcast through LDS: a shfl only reaches the wave
// that computed it, not the other waves of the block.
template <int NW>
DEV_INLINE float block_reduce_sum(float x, float* lds_scratch) {
  int lane = threadIdx.x & (WAVE_SIZE - 1);
  int wid = threadIdx.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = lds_scratch[0];
#pragma unroll
    for (int w = 1; w < NW; ++w) r += lds_scratch[w];
    lds_scratch[0] = r;
  }
  __syncthreads();
  return lds_scratch[0];
}

template <int NW>
DEV_INLINE float block_reduce_max(float x, float* lds_scratch) {
  int lane = threadIdx.x & (WAVE_SIZE - 1);
  int wid = threadIdx.x / WAVE_SIZE;
  x = wave_reduce_max(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = lds_scratch[0];
#pragma unroll
    for (int w = 1; w < NW; ++w) r = fmaxf(r, lds_scratch[w]);
    lds_scratch[0] = r;
  }
  __syncthreads();
  return lds_scratch[0];
}

#define HIP_CHECK_KERNEL()                                      \
  do {                                                          \
    hipError_t e = hipGetLastError();                           \
    if (e != hipSuccess) {                                      \
      TORCH_CHECK(false, "HIP kernel launch failed: ",          \
                  hipGetErrorString(e));                        \
    }                                                           \
  } while (0)
