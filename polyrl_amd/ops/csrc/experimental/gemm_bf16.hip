// Experimental CDNA4 bf16 GEMM — round-2 groundwork (NOT wired, NOT built
// by setup.py; see profiles/microbench_gemm.py).
//
// Shape: C[M,N] = A[M,K] @ W[N,K]^T  (the nn.Linear forward shape: W is the
// torch weight [out,in] row-major, so BOTH operands are read with
// contiguous-K 16-byte lane loads — no transpose anywhere).
//
// Structure (guide "256^2 template", coarse-sync variant):
//   * 256x256 tile, BK=64, 512 threads = 8 waves as 2(M) x 4(N)
//   * per-wave output 128x64 = 8x4 MFMA fragments (v_mfma_f32_16x16x32_bf16)
//   * LDS 128 KiB: A-tile 256x64 + W-tile 256x64, double buffered
//   * global_load_lds dwordx4 staging (2 x 16B per thread per half-image)
//   * st_16x32 LDS swizzle (byte ^= ((byte>>9)&1)<<5), pre-swizzled on the
//     glds SOURCE address, swizzled ds_read on the consumer side
//   * ONE __syncthreads + vmcnt(0) per K-tile (safe coarse schedule; the
//     guide's fine 8-phase interleave with counted vmcnt is the round-2
//     follow-up, worth another ~7-27%)
//   * bijective XCD-aware workgroup remap
//
// Constraints (microbench/prototype): M%256==0, N%256==0, K%64==0,
// A/W 16B-aligned, ld == K.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __hip_bfloat16 bf16;
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

#define BM 256
#define BN 256
#define BK 64
#define NTHREADS 512

// st_16x32 swizzle on a flat byte offset within a 1024B-aligned region.
__device__ __forceinline__ int swz(int byte) {
  return byte ^ (((byte >> 9) & 1) << 5);
}

// one LDS image: 256 rows x 64 bf16 (128 B per row) = 32 KiB
#define IMG_BYTES (BM * BK * 2)

extern "C" __global__ __launch_bounds__(NTHREADS)
void gemm_bf16_nt(const bf16* __restrict__ A,   // [M][K]
                  const bf16* __restrict__ W,   // [N][K]
                  bf16* __restrict__ C,         // [M][N]
                  int M, int N, int K) {
  extern __shared__ char lds[];
  // layout: [buf0: A | W][buf1: A | W]
  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int l15 = lane % 16;     // fragment row/col
  const int lhi = lane / 16;     // k-group
  const int wr = wave / 4;       // wave M index (0..1)
  const int wc = wave % 4;       // wave N index (0..3)

  // ---- XCD-aware bijective remap (guide formula) ----
  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = orig % 8, idx = orig / 8;
  int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const int ntile_n = N / BN;
  const int tm = wg / ntile_n;
  const int tn = wg % ntile_n;
  const long arow0 = (long)tm * BM;
  const long wrow0 = (long)tn * BN;

  // ---- glds staging helper ----
  // each thread stages 2 x 16B per 32 KiB image (512 thr x 16B x 4 iters
  // covers 32 KiB; A and W each need 4 chunk-iters => 8 glds per tile; we
  // split as: per image, iters it=0..3: flat16 = (tid + it*512) * 16
  // dest lds offset = swz(flat16) within the image; src = row/k decode of
  // flat16 (PRE-swizzle so the landed image is swizzled).
  auto stage = [&](const bf16* src, long row0, int kt, char* img) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int flat = (tid + it * NTHREADS) * 16;      // byte offset
      const int sflat = swz(flat);
      const int row = sflat >> 7;                       // 128 B per row
      const int colb = sflat & 127;                     // byte within row
      const bf16* gsrc = src + (row0 + row) * (long)K + (long)kt * BK
                         + (colb >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gsrc,
          (__attribute__((address_space(3))) uint32_t*)(img + flat),
          16, 0, 0);
    }
  };

  f32x4_t acc[8][4];
#pragma unroll
  for (int m = 0; m < 8; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int ktiles = K / BK;

  // prologue: stage tile 0 into buf 0
  stage(A, arow0, 0, lds);
  stage(W, wrow0, 0, lds + IMG_BYTES);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  for (int kt = 0; kt < ktiles; ++kt) {
    char* cur = lds + (kt & 1) * (2 * IMG_BYTES);
    char* nxt = lds + ((kt + 1) & 1) * (2 * IMG_BYTES);
    if (kt + 1 < ktiles) {
      stage(A, arow0, kt + 1, nxt);
      stage(W, wrow0, kt + 1, nxt + IMG_BYTES);
    }
    char* As = cur;
    char* Ws = cur + IMG_BYTES;
    // compute: 2 k-steps of 32 within this 64-deep tile
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int colb = (ks * 32 + lhi * 8) * 2;   // 16B k-slice per lane
      bf16x8_t afrag[8];
#pragma unroll
      for (int m = 0; m < 8; ++m) {
        const int row = wr * 128 + m * 16 + l15;
        afrag[m] = *reinterpret_cast<const bf16x8_t*>(
            As + swz(row * 128 + colb));
      }
      bf16x8_t wfrag[4];
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int row = wc * 64 + n * 16 + l15;
        wfrag[n] = *reinterpret_cast<const bf16x8_t*>(
            Ws + swz(row * 128 + colb));
      }
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], wfrag[n], acc[m][n], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
  }

  // ---- epilogue: C[row][col], fragment row = lhi*4 + r, col = l15 ----
  const long crow0 = arow0 + wr * 128;
  const long ccol0 = wrow0 + wc * 64;
#pragma unroll
  for (int m = 0; m < 8; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        const long row = crow0 + m * 16 + lhi * 4 + r4;
        const long col = ccol0 + n * 16 + l15;
        C[row * (long)N + col] = __float2bfloat16(acc[m][n][r4]);
      }
    }
  }
}
