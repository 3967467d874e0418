// EXPERIMENTAL: MFMA paged-decode attention (bf16, GQA, context-split).
//
// Status: compile- and resource-verified for gfx950 ONLY — round 2 ended
// with zero GPU-minutes left, so this kernel is NOT wired and NOT
// numerics/perf-validated (the guide's two-lane discipline requires a
// live box before any claim).  Design + expected numbers:
// experimental/README.md "MFMA paged-decode attention".
//
// Rationale (PMC-measured, profiles/r02_attn_pmc.csv + PROFILES.md): the
// production decode kernel is exactly VALU-throughput-bound (~60 VALU ops
// per 32 B of KV: per-key dot products, shfl reduces, per-key exps).
// This version moves S = QK^T and O += P V to the matrix unit:
//  * grid (b, hk, split) like the production kernel's context split;
//    4 waves, each streaming its own 32-key tiles (wave-private LDS
//    staging, no block barriers in the loop).
//  * Q A-fragments register-resident: A row l15 = q-head (rows >= G are
//    zero and their lanes' state is never stored).
//  * S via 2x(D/32) mfma_16x16x32 per tile; per-row online softmax on the
//    C layout (4 shfl per row over its 16-lane group).
//  * P through a wave-private 16x32 LDS tile (C->A relayout), then
//    O += P V with (D/16) mfma per tile (B-fragments from row-major Vs,
//    the prefill pattern).
//  * 4-wave merge through LDS; with nsplit > 1 the block writes the
//    (m, l, o) partial in the SAME format as the production kernel, so
//    paged_decode_merge_kernel is reused unchanged.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __hip_bfloat16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
struct alignas(16) bf16x8v {
  bf16_t v[8];
};

#define KT 32                 // keys per wave tile
#define NW 4                  // waves per block

__device__ __forceinline__ float bf2f_(bf16_t x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ bf16_t f2bf_(float x) {
  return __float2bfloat16(x);
}

template <int D>
__device__ __forceinline__ int kswz(int row, int byte_col) {
  return row * (D * 2) + (byte_col ^ ((row & 7) << 4));
}

template <int D, int G>
__global__ __launch_bounds__(256) void paged_decode_attn_mfma_kernel(
    bf16_t* __restrict__ out,              // (B, Hq, D)
    float* __restrict__ partial,           // (B, Hq, S, D+2) when nsplit>1
    const bf16_t* __restrict__ q,          // (B, Hq, D)
    const bf16_t* __restrict__ k_cache,    // (pages, page_size, Hk, D)
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ page_table,    // (B, max_pages)
    const int* __restrict__ context_lens,  // (B,)
    int Hq, int Hk, int page_size, int max_pages, float scale, long ldq,
    int nsplit) {
  constexpr int KS = D / 32;               // A/B k-steps over head_dim
  constexpr int NDT = D / 16;              // 16-wide d-tiles of O
  __shared__ bf16_t Ks[NW][KT * D];        // swizzled (B-frags of S)
  __shared__ bf16_t Vs[NW][KT * D];        // row-major (B-frags of PV)
  __shared__ bf16_t Ps[NW][16 * KT];       // C->A relayout of P
  __shared__ float MrgO[NW][G][D];         // 4-wave merge
  __shared__ float MrgML[NW][G][2];

  const int b = blockIdx.x / (Hk * nsplit);
  const int rem = blockIdx.x % (Hk * nsplit);
  const int hk = rem / nsplit;
  const int sp = rem % nsplit;
  const int L = context_lens[b];
  const int Lc = (L + nsplit - 1) / nsplit;
  const int cbeg = sp * Lc;
  const int cend = (cbeg + Lc < L) ? (cbeg + Lc) : L;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;

  // ---- Q A-fragments: row l15 = q-head (zeros beyond G) ----------------
  bf16x8_t qa[KS];
#pragma unroll
  for (int ks = 0; ks < KS; ++ks) {
    if (l15 < G) {
      qa[ks] = *reinterpret_cast<const bf16x8_t*>(
          q + (long)b * ldq + (long)(hk * G + l15) * D + ks * 32 + lhi * 8);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) qa[ks][j] = (__bf16)0.f;
    }
  }

  // online-softmax state per C row (row = lhi*4 + r = q-head)
  float m_run[4], l_run[4];
  f32x4_t o_acc[NDT];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int nt = 0; nt < NDT; ++nt) o_acc[nt] = {0.f, 0.f, 0.f, 0.f};

  const int kv_stride = Hk * D;
  const int* ptab = page_table + (long)b * max_pages;
  bf16_t* ks_w = Ks[wid];
  bf16_t* vs_w = Vs[wid];
  bf16_t* ps_w = Ps[wid];

  // ---- stream this wave's 32-key tiles ---------------------------------
  for (int t0 = cbeg + wid * KT; t0 < cend; t0 += NW * KT) {
    // wave-private staging: 64 lanes x 8 elems; KT*D/(64*8) passes
#pragma unroll
    for (int it = 0; it < KT * D / (64 * 8); ++it) {
      const int flat = (lane + it * 64) * 8;
      const int row = flat / D;
      const int col = flat % D;
      const int key = t0 + row;
      bf16x8v k8, v8;
      if (key < cend) {
        const int page = ptab[key / page_size];
        const long slot = (long)page * page_size + (key % page_size);
        k8 = *reinterpret_cast<const bf16x8v*>(
            k_cache + slot * kv_stride + hk * D + col);
        v8 = *reinterpret_cast<const bf16x8v*>(
            v_cache + slot * kv_stride + hk * D + col);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) k8.v[j] = f2bf_(0.f), v8.v[j] = f2bf_(0.f);
      }
      *reinterpret_cast<bf16x8v*>(
          reinterpret_cast<char*>(ks_w) + kswz<D>(row, col * 2)) = k8;
      *reinterpret_cast<bf16x8v*>(vs_w + row * D + col) = v8;
    }
    // wave-private LDS: ds_write -> ds_read ordered by lgkmcnt, no barrier

    // ---- S = Q K^T: C [16 head][32 key] --------------------------------
    f32x4_t c[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) c[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int krow = nt * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(ks_w) +
            kswz<D>(krow, (ks * 32 + lhi * 8) * 2));
        c[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[ks], kb, c[nt],
                                                        0, 0, 0);
      }
    }

    // ---- mask tail keys + online softmax per head-row ------------------
    const bool full = (t0 + KT <= cend);
    float s[2][4];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int kpos = t0 + nt * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s[nt][r] = (full || kpos < cend) ? c[nt][r] * scale : -1e30f;
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float rm = fmaxf(s[0][r], s[1][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rm = fmaxf(rm, __shfl_xor(rm, off, 64));
      const float mn = fmaxf(m_run[r], rm);
      alpha[r] = __expf(m_run[r] - mn);
      m_run[r] = mn;
      float rs = 0.f;
      float p0 = __expf(s[0][r] - mn);
      float p1 = __expf(s[1][r] - mn);
      rs = p0 + p1;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rs += __shfl_xor(rs, off, 64);
      l_run[r] = l_run[r] * alpha[r] + rs;
      // store P into the wave tile (row = head lhi*4+r, col = key)
      ps_w[(lhi * 4 + r) * KT + 0 * 16 + l15] = f2bf_(p0);
      ps_w[(lhi * 4 + r) * KT + 1 * 16 + l15] = f2bf_(p1);
    }
#pragma unroll
    for (int nt = 0; nt < NDT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[nt][r] *= alpha[r];

    // ---- O += P V: A = Ps row l15 (head), k = 32 keys ------------------
    const bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
        ps_w + l15 * KT + lhi * 8);
#pragma unroll
    for (int nt = 0; nt < NDT; ++nt) {
      bf16x8_t bv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        bv[j] = *reinterpret_cast<const __bf16*>(
            vs_w + (lhi * 8 + j) * D + nt * 16 + l15);
      o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bv, o_acc[nt],
                                                          0, 0, 0);
    }
  }

  // ---- 4-wave merge through LDS ----------------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int h = lhi * 4 + r;
    if (h < G) {
#pragma unroll
      for (int nt = 0; nt < NDT; ++nt)
        MrgO[wid][h][nt * 16 + l15] = o_acc[nt][r];
      if (l15 == 0) {
        MrgML[wid][h][0] = m_run[r];
        MrgML[wid][h][1] = l_run[r];
      }
    }
  }
  __syncthreads();
  if (wid == 0) {
    // lanes cover D columns per head: lane covers cols l15 + 16*lhi*? ->
    // simple split: 64 lanes x (D/64) columns each
    constexpr int EPL = D / 64;
#pragma unroll
    for (int h = 0; h < G; ++h) {
      float gm = -1e30f;
#pragma unroll
      for (int w = 0; w < NW; ++w) gm = fmaxf(gm, MrgML[w][h][0]);
      float gl = 0.f;
      float acc[EPL];
#pragma unroll
      for (int j = 0; j < EPL; ++j) acc[j] = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float a = __expf(MrgML[w][h][0] - gm);
        gl += MrgML[w][h][1] * a;
#pragma unroll
        for (int j = 0; j < EPL; ++j)
          acc[j] += MrgO[w][h][lane + j * 64] * a;
      }
      if (nsplit == 1) {
        const float inv = (gl > 0.f) ? 1.0f / gl : 0.f;
#pragma unroll
        for (int j = 0; j < EPL; ++j)
          out[((long)b * Hq + hk * G + h) * D + lane + j * 64] =
              f2bf_(acc[j] * inv);
      } else {
        float* pp = partial +
            ((((long)b * Hq + hk * G + h) * nsplit) + sp) * (D + 2);
#pragma unroll
        for (int j = 0; j < EPL; ++j) pp[lane + j * 64] = acc[j];
        if (lane == 0) {
          pp[D] = gm;
          pp[D + 1] = gl;
        }
      }
    }
  }
}

// explicit instantiations for the resource check (the shapes the engine
// dispatches today)
template __global__ void paged_decode_attn_mfma_kernel<128, 4>(
    bf16_t*, float*, const bf16_t*, const bf16_t*, const bf16_t*,
    const int*, const int*, int, int, int, int, float, long, int);
template __global__ void paged_decode_attn_mfma_kernel<128, 8>(
    bf16_t*, float*, const bf16_t*, const bf16_t*, const bf16_t*,
    const int*, const int*, int, int, int, int, float, long, int);
template __global__ void paged_decode_attn_mfma_kernel<128, 1>(
    bf16_t*, float*, const bf16_t*, const bf16_t*, const bf16_t*,
    const int*, const int*, int, int, int, int, float, long, int);
template __global__ void paged_decode_attn_mfma_kernel<64, 4>(
    bf16_t*, float*, const bf16_t*, const bf16_t*, const bf16_t*,
    const int*, const int*, int, int, int, int, float, long, int);

// this file is textually included into attention_decode.hip (the extension
// builds -fno-gpu-rdc, so kernels cannot be launched across TUs) — keep the
// local tile macros from leaking into the including TU
#undef KT
#undef NW
