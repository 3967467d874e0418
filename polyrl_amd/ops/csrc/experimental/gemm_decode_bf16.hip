// Experimental decode-phase projection GEMM (round-2 item #3; NOT wired).
//
//   out[M,N] (f32, split-K atomics) = x[M,K] @ W[N,K]^T,  M = 256 fixed
//
// The decode regime: M small (continuous-batching rows), W enormous —
// the kernel is WEIGHT-READ bound (N*K*2 bytes at HBM rate is the floor).
// Structure per the guide's M=256 sampling-GEMM recipe, adapted to the
// nn.Linear W[N][K] layout (contiguous-K fragments, no transpose):
//   * block tile: all 256 rows x BN=64 cols; 8 waves as 8(M) x 1(N):
//     per-wave 32 rows x 64 cols = 2x4 MFMA fragments
//   * K-loop BK=64, double-buffered glds staging of BOTH operands:
//     x-tile 256x64 (32 KiB) + W-tile 64x64 (8 KiB) -> 80 KiB LDS total
//   * split-K: gridDim.y slabs, f32 atomicAdd into out (one combine pass
//     is the round-2 upgrade — splitk-seam notes in MI355X_MICROARCH.md)
//   * st_16x32 swizzle on the x image (16 lanes hit 16 different 128 B
//     rows); the W image rows are 128 B as well — same swizzle
// Constraints: M == 256, N % 64 == 0, K % (64 * gridDim.y) == 0.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

#define DM 256
#define DN 64
#define DK 64
#define DTHREADS 512

__device__ __forceinline__ int dswz(int byte) {
  return byte ^ (((byte >> 9) & 1) << 5);
}

#define XIMG (DM * DK * 2)     // 32 KiB
#define WIMG (DN * DK * 2)     // 8 KiB
#define BUF (XIMG + WIMG)      // 40 KiB per buffer

extern "C" __global__ __launch_bounds__(DTHREADS)
void gemm_decode_bf16(const bf16* __restrict__ x,   // [256][K]
                      const bf16* __restrict__ W,   // [N][K]
                      float* __restrict__ out,      // [256][N], pre-zeroed
                      int N, int K) {
  extern __shared__ char lds[];
  const int tid = threadIdx.x;
  const int lane = tid % 64;
  const int wave = tid / 64;          // M-split: rows 32*wave..+32
  const int l15 = lane % 16;
  const int lhi = lane / 16;

  // grid: x = N/64 col-tiles (XCD-remapped), y = split-K slab
  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = orig % 8, idx = orig / 8;
  const int tn = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const long ncol0 = (long)tn * DN;
  const int kslabs = gridDim.y;
  const int kper = K / kslabs;                  // K per slab (mult of 64)
  const long k0 = (long)blockIdx.y * kper;

  auto stage_x = [&](int kt, char* img) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {            // 4 x 512 x 16B = 32 KiB
      const int flat = (tid + it * DTHREADS) * 16;
      const int sflat = dswz(flat);
      const int row = sflat >> 7;
      const int colb = sflat & 127;
      const bf16* g = x + (long)row * K + k0 + (long)kt * DK + (colb >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)g,
          (__attribute__((address_space(3))) uint32_t*)(img + flat), 16, 0, 0);
    }
  };
  auto stage_w = [&](int kt, char* img) {
    // 8 KiB: 512 threads x 16B = 1 iter
    const int flat = tid * 16;
    const int sflat = dswz(flat);
    const int row = sflat >> 7;
    const int colb = sflat & 127;
    const bf16* g = W + (ncol0 + row) * (long)K + k0 + (long)kt * DK
                    + (colb >> 1);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)g,
        (__attribute__((address_space(3))) uint32_t*)(img + flat), 16, 0, 0);
  };

  f32x4_t acc[2][4];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int ktiles = kper / DK;
  stage_x(0, lds);
  stage_w(0, lds + XIMG);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  for (int kt = 0; kt < ktiles; ++kt) {
    char* cur = lds + (kt & 1) * BUF;
    char* nxt = lds + ((kt + 1) & 1) * BUF;
    if (kt + 1 < ktiles) {
      stage_x(kt + 1, nxt);
      stage_w(kt + 1, nxt + XIMG);
    }
    char* Xs = cur;
    char* Ws = cur + XIMG;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int colb = (ks * 32 + lhi * 8) * 2;
      bf16x8_t xfrag[2];
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int row = wave * 32 + m * 16 + l15;
        xfrag[m] = *reinterpret_cast<const bf16x8_t*>(
            Xs + dswz(row * 128 + colb));
      }
      bf16x8_t wfrag[4];
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = n * 16 + l15;
        wfrag[n] = *reinterpret_cast<const bf16x8_t*>(
            Ws + dswz(row * 128 + colb));
      }
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              xfrag[m], wfrag[n], acc[m][n], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
  }

  // split-K combine: f32 atomics (kslabs==1 -> plain store)
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        const long row = wave * 32 + m * 16 + lhi * 4 + r4;
        const long col = ncol0 + n * 16 + l15;
        if (kslabs == 1)
          out[row * (long)N + col] = acc[m][n][r4];
        else
          atomicAdd(&out[row * (long)N + col], acc[m][n][r4]);
      }
    }
  }
}
