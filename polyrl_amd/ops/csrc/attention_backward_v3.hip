// Varlen causal flash-attention backward v3 (bf16, head_dim 64/128, MFMA).
//
// Reference capability: the training-side flash-attn backward the reference
// reaches through the flash-attn package (SURVEY.md §2.2.2 'flash-attn').
//
// v1/v2 (attention_backward.hip) are KV-stationary single-kernel designs
// whose dQ (every q-block) and dK/dV (every block) updates go through
// global fp32 atomics, at 104 KiB LDS = 1 block/CU.  Measured ~10% MFU,
// and a fully-vectorized-LDS v2 ran at exactly v1's speed — the kernel is
// atomics/occupancy-bound, not LDS-issue-bound (profiles/PROFILES.md).
//
// v3 removes every atomic by splitting the backward into two kernels that
// each OWN their output rows exclusively (the extra recompute of S/dP is
// ~1.4x FLOPs for >5x less memory traffic):
//
//  * dkv kernel — grid (kv-tile of 64 keys, kv-head).  The GQA group's G
//    q-heads are a register-accumulated loop INSIDE the block, so dK/dV
//    for the tile are complete when the block ends: one plain store pass.
//    4 waves split the 64-q-row block by q (16 rows each) for S/dP and by
//    d-slice (D/4 columns each) for the dV/dK MFMAs.  LDS 80 KiB (D=128)
//    -> 2 blocks/CU.
//  * dq kernel — grid (q-tile of 64 rows, q-head).  Waves own 16 q rows
//    for the whole kv loop and accumulate dQ in registers; one store pass
//    at the end.  LDS 56 KiB (D=128).
//
// Both kernels reuse the 64-granularity tile table the forward already
// builds (build_varlen_tiles t64), so no extra host work per batch.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define KB3 64             // keys per kv tile
#define QB3 64             // q rows per block / iteration
#define NWAVE3 4

// swizzled row-major [rows][D] image (byte addressing), 16-B granules
template <int D>
DEV_INLINE int kswz3(int row, int byte_col) {
  return row * (D * 2) + (byte_col ^ ((row & 7) << 4));
}

// transposed [d][X] image (element addressing), 8-element granules
DEV_INLINE int trid3(int d, int x, int X) {
  return d * X + (x ^ ((d & 3) << 3));
}
DEV_INLINE int trid3_vec(int d, int xblock, int X) {
  return d * X + ((xblock * 8) ^ ((d & 3) << 3));
}

// ------------------------------------------------------- delta preprocess
// delta[t, h] = sum_d dO[t,h,d] * O[t,h,d]   (fp32)
template <int D>
__global__ void attn_bwd_delta_kernel3(float* __restrict__ delta,
                                       const bf16_t* __restrict__ dout,
                                       const bf16_t* __restrict__ out,
                                       long rows /* total_q * Hq */) {
  const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const bf16x8* dp = reinterpret_cast<const bf16x8*>(dout + row * D);
  const bf16x8* op = reinterpret_cast<const bf16x8*>(out + row * D);
  float acc = 0.f;
  if (lane < D / 8) {
    bf16x8 d8 = dp[lane], o8 = op[lane];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += bf2f(d8.v[j]) * bf2f(o8.v[j]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
  if (lane == 0) delta[row] = acc;
}

// ---------------------------------------------------------------- dK/dV
template <int D>
__global__ __launch_bounds__(256) void attn_bwd_dkv_kernel(
    float* __restrict__ dk,          // (total_k, Hk, D) fp32 (fully written)
    float* __restrict__ dv,
    const bf16_t* __restrict__ q,    // packed (strided) inputs
    const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout, // (total_q, Hq, D) contiguous
    const float* __restrict__ lse,   // (total_q, Hq)
    const float* __restrict__ delta, // (total_q, Hq)
    const int* __restrict__ cu_q, const int* __restrict__ cu_k,
    const int* __restrict__ tile_seq, const int* __restrict__ tile_k0,
    int Hq, int Hk, float scale, int causal,
    long ldq, long ldk, long ldv) {
  __shared__ bf16_t KsSwz[KB3 * D];          // B-frags of S = Q K^T
  __shared__ bf16_t VsSwz[KB3 * D];          // B-frags of dP = dO V^T
  __shared__ bf16_t QsT[D * QB3];            // [d][q] for dK += dS^T Q
  __shared__ bf16_t DOsT[D * QB3];           // [d][q] for dV += P^T dO
  __shared__ bf16_t Pt[KB3 * QB3];           // [key][q]
  __shared__ bf16_t DSt[KB3 * QB3];

  const int tile = blockIdx.x;
  const int hk = blockIdx.y;
  const int G = Hq / Hk;
  const int seq = tile_seq[tile];
  const int t0 = tile_k0[tile];              // local first key of this tile
  const int qbeg = cu_q[seq], qend = cu_q[seq + 1];
  const int kbeg = cu_k[seq], kend = cu_k[seq + 1];
  const int Lq = qend - qbeg, Lk = kend - kbeg;
  const int qk_off = Lk - Lq;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;                 // 0..3
  constexpr int KS = D / 32;                 // k-steps over head_dim
  constexpr int NDT = D / 64;                // 16-wide d-tiles per wave slice

  // ---- stage K/V swizzled (once per block) ------------------------------
  {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < KB3 * D / (256 * 8); ++it) {
      const int flat = (tid + it * 256) * 8;
      const int row = flat / D;
      const int col = flat % D;
      const int krow = t0 + row;
      bf16x8 k8, v8;
      if (krow < Lk) {
        k8 = *reinterpret_cast<const bf16x8*>(
            k + (long)(kbeg + krow) * ldk + (long)hk * D + col);
        v8 = *reinterpret_cast<const bf16x8*>(
            v + (long)(kbeg + krow) * ldv + (long)hk * D + col);
      } else {
        for (int j = 0; j < 8; ++j) k8.v[j] = f2bf(0.f), v8.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(KsSwz) + kswz3<D>(row, col * 2)) = k8;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(VsSwz) + kswz3<D>(row, col * 2)) = v8;
    }
  }
  __syncthreads();

  // dK/dV accumulators: wave owns d-slice [wid*D/4, wid*D/4 + D/4);
  // rows = 4 key sub-tiles of 16; survive the g-loop (GQA row sum).
  f32x4_t dv_acc[4][NDT], dk_acc[4][NDT];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int nt = 0; nt < NDT; ++nt) {
      dv_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
      dk_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
    }

  // causal: first q row that can see key t0
  int q_start = causal ? ((t0 - qk_off) > 0 ? (t0 - qk_off) : 0) : 0;
  q_start -= q_start % QB3;

  for (int g = 0; g < G; ++g) {
    const int hq = hk * G + g;
    for (int qb = q_start; qb < Lq; qb += QB3) {
      __syncthreads();   // prior iteration's Pt/DSt/QsT/DOsT reads done
      // ---- wave w loads its 16 q rows' A-frags; stages transposed ------
      const int qrow_l = qb + wid * 16 + l15;
      const bool live = qrow_l < Lq;
      const long gq = (long)(qbeg + (live ? qrow_l : 0));
      bf16x8_t qa[KS], da[KS];
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        if (live) {
          qa[ks] = *reinterpret_cast<const bf16x8_t*>(
              q + gq * ldq + (long)hq * D + ks * 32 + lhi * 8);
          da[ks] = *reinterpret_cast<const bf16x8_t*>(
              dout + (gq * Hq + hq) * D + ks * 32 + lhi * 8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qa[ks][j] = (__bf16)0.f;
            da[ks][j] = (__bf16)0.f;
          }
        }
        const int qcol = wid * 16 + l15;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = ks * 32 + lhi * 8 + j;
          QsT[trid3(d, qcol, QB3)] = (bf16_t)qa[ks][j];
          DOsT[trid3(d, qcol, QB3)] = (bf16_t)da[ks][j];
        }
      }

      // ---- S = Q K^T, dP = dO V^T  (wave's 16 q rows x 64 keys) --------
      f32x4_t sc[4], dpc[4];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        sc[nt] = {0.f, 0.f, 0.f, 0.f};
        dpc[nt] = {0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int krow = nt * 16 + l15;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
          const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<char*>(KsSwz) +
              kswz3<D>(krow, (ks * 32 + lhi * 8) * 2));
          const bf16x8_t vb = *reinterpret_cast<const bf16x8_t*>(
              reinterpret_cast<char*>(VsSwz) +
              kswz3<D>(krow, (ks * 32 + lhi * 8) * 2));
          sc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qa[ks], kb, sc[nt], 0, 0, 0);
          dpc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              da[ks], vb, dpc[nt], 0, 0, 0);
        }
      }

      // ---- P, dS -> transposed LDS (interior tiles skip the mask:
      // VALU-bound kernels, PMC 6:1 VALU:MFMA) -----------------------
      const bool interior =
          (t0 + KB3 <= Lk) && (qb + wid * 16 + 16 <= Lq) &&
          (!causal || t0 + KB3 - 1 <= qk_off + qb + wid * 16);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_c = qb + wid * 16 + lhi * 4 + r;   // C-layout row
        const bool qdead = qrow_c >= Lq;
        const long gqc = (long)(qbeg + (qdead ? 0 : qrow_c)) * Hq + hq;
        const float lse_r = qdead ? 0.f : lse[gqc];
        const float del_r = qdead ? 0.f : delta[gqc];
        const int qpos = qk_off + qrow_c;
        const int qcol = wid * 16 + lhi * 4 + r;
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int key = nt * 16 + l15;
          float p;
          if (interior) {
            p = __expf(sc[nt][r] * scale - lse_r);
          } else {
            const int kpos = t0 + key;
            const bool dead = qdead || (kpos >= Lk) ||
                              (causal && kpos > qpos);
            p = dead ? 0.f : __expf(sc[nt][r] * scale - lse_r);
          }
          const float ds = p * (dpc[nt][r] - del_r) * scale;
          Pt[trid3(key, qcol, QB3)] = f2bf(p);
          DSt[trid3(key, qcol, QB3)] = f2bf(ds);
        }
      }
      __syncthreads();

      // ---- dV += P^T dO ; dK += dS^T Q  (wave owns d-slice wid*D/4) ----
#pragma unroll
      for (int a = 0; a < 4; ++a) {          // key sub-tile (rows)
        const int keyrow = a * 16 + l15;
#pragma unroll
        for (int kq = 0; kq < 2; ++kq) {     // k = q rows, 2 x 32
          const bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
              Pt + trid3_vec(keyrow, kq * 4 + lhi, QB3));
          const bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
              DSt + trid3_vec(keyrow, kq * 4 + lhi, QB3));
#pragma unroll
          for (int nt = 0; nt < NDT; ++nt) {
            const int d = wid * (D / 4) + nt * 16 + l15;
            const bf16x8_t dob = *reinterpret_cast<const bf16x8_t*>(
                DOsT + trid3_vec(d, kq * 4 + lhi, QB3));
            const bf16x8_t qb_ = *reinterpret_cast<const bf16x8_t*>(
                QsT + trid3_vec(d, kq * 4 + lhi, QB3));
            dv_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pa, dob, dv_acc[a][nt], 0, 0, 0);
            dk_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                dsa, qb_, dk_acc[a][nt], 0, 0, 0);
          }
        }
      }
    }
  }

  // ---- store dK/dV (exclusive ownership: plain stores, no zero-init) ----
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow_l = t0 + a * 16 + lhi * 4 + r;
      if (krow_l >= Lk) continue;
      float* dkp = dk + ((long)(kbeg + krow_l) * Hk + hk) * D;
      float* dvp = dv + ((long)(kbeg + krow_l) * Hk + hk) * D;
#pragma unroll
      for (int nt = 0; nt < NDT; ++nt) {
        dkp[wid * (D / 4) + nt * 16 + l15] = dk_acc[a][nt][r];
        dvp[wid * (D / 4) + nt * 16 + l15] = dv_acc[a][nt][r];
      }
    }
}

// ------------------------------------------------------------------- dQ
template <int D>
__global__ __launch_bounds__(256) void attn_bwd_dq_kernel(
    float* __restrict__ dq,          // (total_q, Hq, D) fp32 (fully written)
    const bf16_t* __restrict__ q,
    const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout,
    const float* __restrict__ lse,
    const float* __restrict__ delta,
    const int* __restrict__ cu_q, const int* __restrict__ cu_k,
    const int* __restrict__ tile_seq, const int* __restrict__ tile_q0,
    int Hq, int Hk, float scale, int causal,
    long ldq, long ldk, long ldv) {
  __shared__ bf16_t KsSwz[KB3 * D];          // B-frags of S
  __shared__ bf16_t VsSwz[KB3 * D];          // B-frags of dP
  __shared__ bf16_t KsRowT[D * KB3];         // [d][key] for dQ += dS K
  __shared__ bf16_t DSq[QB3 * KB3];          // [q][key] (wave-private rows)

  const int tile = blockIdx.x;
  const int hq = blockIdx.y;
  const int G = Hq / Hk;
  const int hk = hq / G;
  const int seq = tile_seq[tile];
  const int qt0 = tile_q0[tile];             // local first q row of tile
  const int qbeg = cu_q[seq], qend = cu_q[seq + 1];
  const int kbeg = cu_k[seq], kend = cu_k[seq + 1];
  const int Lq = qend - qbeg, Lk = kend - kbeg;
  const int qk_off = Lk - Lq;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;
  constexpr int KS = D / 32;
  constexpr int NDQ = D / 16;                // 16-wide d-tiles of dQ

  // ---- wave's 16 q rows: A-frags of Q / dO, lse/delta ------------------
  const int qrow_l = qt0 + wid * 16 + l15;
  const bool live = qrow_l < Lq;
  const long gq = (long)(qbeg + (live ? qrow_l : 0));
  bf16x8_t qa[KS], da[KS];
#pragma unroll
  for (int ks = 0; ks < KS; ++ks) {
    if (live) {
      qa[ks] = *reinterpret_cast<const bf16x8_t*>(
          q + gq * ldq + (long)hq * D + ks * 32 + lhi * 8);
      da[ks] = *reinterpret_cast<const bf16x8_t*>(
          dout + (gq * Hq + hq) * D + ks * 32 + lhi * 8);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qa[ks][j] = (__bf16)0.f;
        da[ks][j] = (__bf16)0.f;
      }
    }
  }

  f32x4_t dq_acc[NDQ];
#pragma unroll
  for (int nt = 0; nt < NDQ; ++nt) dq_acc[nt] = {0.f, 0.f, 0.f, 0.f};

  // kv range: causal stops after the last key row qt0+63 can see
  int kv_last = Lk;
  if (causal) {
    const int vis = qk_off + qt0 + QB3;      // exclusive
    kv_last = vis < Lk ? vis : Lk;
  }
  if (kv_last < 0) kv_last = 0;

  for (int t0 = 0; t0 < kv_last; t0 += KB3) {
    __syncthreads();   // prior iteration's K/V reads done
    // ---- cooperative stage of K/V (swizzled) + K^T --------------------
    {
      const int tid = threadIdx.x;
#pragma unroll
      for (int it = 0; it < KB3 * D / (256 * 8); ++it) {
        const int flat = (tid + it * 256) * 8;
        const int row = flat / D;
        const int col = flat % D;
        const int krow = t0 + row;
        bf16x8 k8, v8;
        if (krow < Lk) {
          k8 = *reinterpret_cast<const bf16x8*>(
              k + (long)(kbeg + krow) * ldk + (long)hk * D + col);
          v8 = *reinterpret_cast<const bf16x8*>(
              v + (long)(kbeg + krow) * ldv + (long)hk * D + col);
        } else {
          for (int j = 0; j < 8; ++j)
            k8.v[j] = f2bf(0.f), v8.v[j] = f2bf(0.f);
        }
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(KsSwz) + kswz3<D>(row, col * 2)) = k8;
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(VsSwz) + kswz3<D>(row, col * 2)) = v8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          KsRowT[trid3(col + j, row, KB3)] = k8.v[j];
      }
    }
    __syncthreads();

    // ---- S, dP over wave's 16 q rows x 64 keys -------------------------
    f32x4_t sc[4], dpc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      sc[nt] = {0.f, 0.f, 0.f, 0.f};
      dpc[nt] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int krow = nt * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(KsSwz) +
            kswz3<D>(krow, (ks * 32 + lhi * 8) * 2));
        const bf16x8_t vb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(VsSwz) +
            kswz3<D>(krow, (ks * 32 + lhi * 8) * 2));
        sc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qa[ks], kb, sc[nt], 0, 0, 0);
        dpc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            da[ks], vb, dpc[nt], 0, 0, 0);
      }
    }

    // ---- dS -> wave-private LDS rows (no barrier needed; interior
    // tiles skip the mask — PMC 8.9:1 VALU:MFMA) ----------------------
    const bool interior =
        (t0 + KB3 <= Lk) && (qt0 + wid * 16 + 16 <= Lq) &&
        (!causal || t0 + KB3 - 1 <= qk_off + qt0 + wid * 16);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow_c = qt0 + wid * 16 + lhi * 4 + r;
      const bool qdead = qrow_c >= Lq;
      const long gqc = (long)(qbeg + (qdead ? 0 : qrow_c)) * Hq + hq;
      const float lse_r = qdead ? 0.f : lse[gqc];
      const float del_r = qdead ? 0.f : delta[gqc];
      const int qpos = qk_off + qrow_c;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        float p;
        if (interior) {
          p = __expf(sc[nt][r] * scale - lse_r);
        } else {
          const int kpos = t0 + nt * 16 + l15;
          const bool dead = qdead || (kpos >= Lk) ||
                            (causal && kpos > qpos);
          p = dead ? 0.f : __expf(sc[nt][r] * scale - lse_r);
        }
        const float ds = p * (dpc[nt][r] - del_r) * scale;
        DSq[trid3(wid * 16 + lhi * 4 + r, nt * 16 + l15, KB3)] = f2bf(ds);
      }
    }

    // ---- dQ += dS K  (A: DSq rows, B: KsRowT) --------------------------
#pragma unroll
    for (int kq = 0; kq < 2; ++kq) {         // k = keys, 2 x 32
      const bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
          DSq + trid3_vec(wid * 16 + l15, kq * 4 + lhi, KB3));
#pragma unroll
      for (int nt = 0; nt < NDQ; ++nt) {
        const int d = nt * 16 + l15;
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            KsRowT + trid3_vec(d, kq * 4 + lhi, KB3));
        dq_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, kb, dq_acc[nt], 0, 0, 0);
      }
    }
  }

  // ---- store dQ (exclusive ownership) -----------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow_c = qt0 + wid * 16 + lhi * 4 + r;
    if (qrow_c >= Lq) continue;
    float* dqp = dq + ((long)(qbeg + qrow_c) * Hq + hq) * D;
#pragma unroll
    for (int nt = 0; nt < NDQ; ++nt) dqp[nt * 16 + l15] = dq_acc[nt][r];
  }
}

// ---------------------------------------------------------------- launcher
void varlen_attention_backward_v3(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,  // fp32 (empty ok)
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor dout, torch::Tensor lse,
    torch::Tensor cu_seqlens_q, torch::Tensor cu_seqlens_k,
    torch::Tensor t64_seq, torch::Tensor t64_q0,
    double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  auto packed = [](const torch::Tensor& t) {
    return t.stride(2) == 1 && t.stride(1) == t.size(2);
  };
  TORCH_CHECK(packed(q) && packed(k) && packed(v));
  TORCH_CHECK(out.is_contiguous() && dout.is_contiguous());
  TORCH_CHECK(lse.is_contiguous() && lse.dtype() == torch::kFloat32);
  TORCH_CHECK(dq.dtype() == torch::kFloat32 && dq.is_contiguous());
  const int Hq = q.size(1), D = q.size(2), Hk = k.size(1);
  TORCH_CHECK(D == 128 || D == 64, "backward v3 supports head_dim 64/128");
  TORCH_CHECK(Hq % Hk == 0);
  const long tq = q.size(0);
  auto stream = at::hip::getCurrentHIPStream();

  auto delta = torch::empty({tq, (long)Hq}, lse.options());
  const long rows = tq * Hq;
  const int ntiles = t64_seq.size(0);
  const dim3 blk(256);
  if (D == 128) {
    attn_bwd_delta_kernel3<128><<<dim3((rows + 3) / 4), blk, 0, stream>>>(
        delta.data_ptr<float>(), (const bf16_t*)dout.data_ptr(),
        (const bf16_t*)out.data_ptr(), rows);
    attn_bwd_dkv_kernel<128><<<dim3(ntiles, Hk), blk, 0, stream>>>(
        dk.data_ptr<float>(), dv.data_ptr<float>(),
        (const bf16_t*)q.data_ptr(), (const bf16_t*)k.data_ptr(),
        (const bf16_t*)v.data_ptr(), (const bf16_t*)dout.data_ptr(),
        lse.data_ptr<float>(), delta.data_ptr<float>(),
        cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
        t64_seq.data_ptr<int>(), t64_q0.data_ptr<int>(), Hq, Hk,
        (float)scale, causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
    attn_bwd_dq_kernel<128><<<dim3(ntiles, Hq), blk, 0, stream>>>(
        dq.data_ptr<float>(),
        (const bf16_t*)q.data_ptr(), (const bf16_t*)k.data_ptr(),
        (const bf16_t*)v.data_ptr(), (const bf16_t*)dout.data_ptr(),
        lse.data_ptr<float>(), delta.data_ptr<float>(),
        cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
        t64_seq.data_ptr<int>(), t64_q0.data_ptr<int>(), Hq, Hk,
        (float)scale, causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
  } else {
    attn_bwd_delta_kernel3<64><<<dim3((rows + 3) / 4), blk, 0, stream>>>(
        delta.data_ptr<float>(), (const bf16_t*)dout.data_ptr(),
        (const bf16_t*)out.data_ptr(), rows);
    attn_bwd_dkv_kernel<64><<<dim3(ntiles, Hk), blk, 0, stream>>>(
        dk.data_ptr<float>(), dv.data_ptr<float>(),
        (const bf16_t*)q.data_ptr(), (const bf16_t*)k.data_ptr(),
        (const bf16_t*)v.data_ptr(), (const bf16_t*)dout.data_ptr(),
        lse.data_ptr<float>(), delta.data_ptr<float>(),
        cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
        t64_seq.data_ptr<int>(), t64_q0.data_ptr<int>(), Hq, Hk,
        (float)scale, causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
    attn_bwd_dq_kernel<64><<<dim3(ntiles, Hq), blk, 0, stream>>>(
        dq.data_ptr<float>(),
        (const bf16_t*)q.data_ptr(), (const bf16_t*)k.data_ptr(),
        (const bf16_t*)v.data_ptr(), (const bf16_t*)dout.data_ptr(),
        lse.data_ptr<float>(), delta.data_ptr<float>(),
        cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
        t64_seq.data_ptr<int>(), t64_q0.data_ptr<int>(), Hq, Hk,
        (float)scale, causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
  }
  HIP_CHECK_KERNEL();
}
