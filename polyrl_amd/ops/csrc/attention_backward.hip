// Varlen causal flash-attention BACKWARD (bf16, head_dim 128, MFMA).
//
// Reference capability: the training-side flash-attn backward the reference
// reaches through the flash-attn package (SURVEY.md §2.2.2 row 'flash-attn')
// — rebuilt CDNA4-first to pair with attention_prefill.hip's forward.
//
// Design (KV-stationary, FA2-style):
//  * grid = (kv_tile of 32 keys, q-head).  GQA folds into atomics: blocks of
//    the G q-heads sharing a kv-head atomically accumulate dK/dV (fp32).
//  * 4 waves per block; wave w owns q-blocks w, w+4, ... of 32 rows each.
//  * per q-block: S = QK^T (MFMA), P = exp(S - lse) (lse from forward, no
//    re-reduction), dP = dO V^T (MFMA), dS = P (dP - delta) scale with
//    delta = rowsum(dO * O) precomputed by delta_kernel, then
//    dQ += dS K (atomic fp32), dV += P^T dO, dK += dS^T Q (wave-private
//    accumulators, atomically flushed once per block).
//  * K and V are staged twice in LDS: XOR-swizzled for the B-fragments of
//    the QK^T / dO V^T products (guide §6 G4: row-major [32][128] bf16 is a
//    16-way bank conflict on ds_read_b128) and row-major for the
//    d-major B-fragments of dS K.  Q / dO / P^T / dS^T live in wave-private
//    LDS tiles, so only the K/V restage needs a block barrier.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define KB 32              // keys per block tile
#define QB 32              // q rows per wave iteration
#define HEAD_DIM 128
#define NWAVE 4

DEV_INLINE int kswz(int row, int byte_col) {
  return row * (HEAD_DIM * 2) + (byte_col ^ ((row & 7) << 4));
}

// ------------------------------------------------------- delta preprocess
// delta[t, h] = sum_d dO[t,h,d] * O[t,h,d]   (fp32)
__global__ void attn_bwd_delta_kernel(float* __restrict__ delta,
                                      const bf16_t* __restrict__ dout,
                                      const bf16_t* __restrict__ out,
                                      long rows /* total_q * Hq */) {
  const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const bf16x8* dp = reinterpret_cast<const bf16x8*>(dout + row * HEAD_DIM);
  const bf16x8* op = reinterpret_cast<const bf16x8*>(out + row * HEAD_DIM);
  float acc = 0.f;
  if (lane < HEAD_DIM / 8) {
    bf16x8 d8 = dp[lane], o8 = op[lane];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += bf2f(d8.v[j]) * bf2f(o8.v[j]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
  if (lane == 0) delta[row] = acc;
}

// ------------------------------------------------------------- backward
__global__ __launch_bounds__(256) void attn_bwd_kernel(
    float* __restrict__ dq,          // (total_q, Hq, 128) fp32, zeroed
    float* __restrict__ dk,          // (total_k, Hk, 128) fp32, zeroed
    float* __restrict__ dv,          // (total_k, Hk, 128) fp32, zeroed
    const bf16_t* __restrict__ q,    // packed (strided) inputs
    const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout,  // (total_q, Hq, 128) contiguous
    const float* __restrict__ lse,    // (total_q, Hq)
    const float* __restrict__ delta,  // (total_q, Hq)
    const int* __restrict__ cu_q, const int* __restrict__ cu_k,
    const int* __restrict__ tile_seq, const int* __restrict__ tile_k0,
    int Hq, int Hk, float scale, int causal,
    long ldq, long ldk, long ldv) {
  __shared__ bf16_t KsSwz[KB * HEAD_DIM];
  __shared__ bf16_t VsSwz[KB * HEAD_DIM];
  __shared__ bf16_t KsRow[KB * HEAD_DIM];
  __shared__ bf16_t Qs[NWAVE][QB * HEAD_DIM];
  __shared__ bf16_t DOs[NWAVE][QB * HEAD_DIM];
  __shared__ bf16_t Pt[NWAVE][KB * QB];    // P^T  (key-major)
  __shared__ bf16_t DSt[NWAVE][KB * QB];   // dS^T (key-major)

  const int tile = blockIdx.x;
  const int hq = blockIdx.y;
  const int G = Hq / Hk;
  const int hk = hq / G;
  const int seq = tile_seq[tile];
  const int t0 = tile_k0[tile];            // local first key of this tile
  const int qbeg = cu_q[seq], qend = cu_q[seq + 1];
  const int kbeg = cu_k[seq], kend = cu_k[seq + 1];
  const int Lq = qend - qbeg, Lk = kend - kbeg;
  const int qk_off = Lk - Lq;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;               // 0..3

  // ---- stage K/V tile: swizzled + row-major K, swizzled V ---------------
  {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int flat = (tid + it * 256) * 8;
      const int row = flat / HEAD_DIM;
      const int col = flat % HEAD_DIM;
      const int krow = t0 + row;
      bf16x8 k8, v8;
      if (krow < Lk) {
        k8 = *reinterpret_cast<const bf16x8*>(
            k + (long)(kbeg + krow) * ldk + (long)hk * HEAD_DIM + col);
        v8 = *reinterpret_cast<const bf16x8*>(
            v + (long)(kbeg + krow) * ldv + (long)hk * HEAD_DIM + col);
      } else {
        for (int j = 0; j < 8; ++j) k8.v[j] = f2bf(0.f), v8.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(KsSwz) + kswz(row, col * 2)) = k8;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(VsSwz) + kswz(row, col * 2)) = v8;
      *reinterpret_cast<bf16x8*>(KsRow + row * HEAD_DIM + col) = k8;
    }
  }
  __syncthreads();

  // dK/dV accumulators: 2 key sub-tiles x 8 d-tiles, C rows = key lhi*4+r
  f32x4_t dv_acc[2][8], dk_acc[2][8];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      dv_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
      dk_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
    }

  // causal: first q row that can see key t0 is (t0 - qk_off)
  int q_start = causal ? ((t0 - qk_off) > 0 ? (t0 - qk_off) : 0) : 0;
  q_start -= q_start % QB;

  for (int qb = q_start + wid * QB; qb < Lq; qb += NWAVE * QB) {
    // ---- stage this wave's Q and dO tiles (row-major, wave-private) ----
    bf16_t* qs = Qs[wid];
    bf16_t* dos = DOs[wid];
#pragma unroll
    for (int it = 0; it < 8; ++it) {           // 64 lanes x 8 iters x 8 el
      const int flat = (lane + it * 64) * 8;
      const int row = flat / HEAD_DIM;
      const int col = flat % HEAD_DIM;
      const int qrow = qb + row;
      bf16x8 q8, d8;
      if (qrow < Lq) {
        q8 = *reinterpret_cast<const bf16x8*>(
            q + (long)(qbeg + qrow) * ldq + (long)hq * HEAD_DIM + col);
        d8 = *reinterpret_cast<const bf16x8*>(
            dout + ((long)(qbeg + qrow) * Hq + hq) * HEAD_DIM + col);
      } else {
        for (int j = 0; j < 8; ++j) q8.v[j] = f2bf(0.f), d8.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(qs + row * HEAD_DIM + col) = q8;
      *reinterpret_cast<bf16x8*>(dos + row * HEAD_DIM + col) = d8;
    }
    // wave-private tiles: ds_write -> ds_read ordered by lgkmcnt, no barrier

    // ---- A-fragments of Q and dO: row = l15 (+16), k = lhi*8.. ----------
    bf16x8_t qa[2][4], da[2][4];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = half * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        qa[half][ks] = *reinterpret_cast<const bf16x8_t*>(
            qs + row * HEAD_DIM + ks * 32 + lhi * 8);
        da[half][ks] = *reinterpret_cast<const bf16x8_t*>(
            dos + row * HEAD_DIM + ks * 32 + lhi * 8);
      }
    }

    // ---- S = Q K^T, dP = dO V^T  (C: row=qrow half*16+lhi*4+r, col=key l15)
    f32x4_t sc[2][2], dpc[2][2];   // [q half][key sub-tile]
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        sc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
        dpc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int krow = nt * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(KsSwz) + kswz(krow, (ks * 32 + lhi * 8) * 2));
        const bf16x8_t vb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(VsSwz) + kswz(krow, (ks * 32 + lhi * 8) * 2));
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
          sc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qa[qi][ks], kb, sc[qi][nt], 0, 0, 0);
          dpc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              da[qi][ks], vb, dpc[qi][nt], 0, 0, 0);
        }
      }
    }

    // ---- P = exp(S*scale - lse);  dS = P * (dP - delta) * scale ----------
    float pv[2][2][4], dsv[2][2][4];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_l = qb + qi * 16 + lhi * 4 + r;
        const bool qdead = qrow_l >= Lq;
        const long gq = (long)(qbeg + (qdead ? 0 : qrow_l)) * Hq + hq;
        const float lse_r = qdead ? 0.f : lse[gq];
        const float del_r = qdead ? 0.f : delta[gq];
        const int qpos = qk_off + qrow_l;
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          const int kpos = t0 + nt * 16 + l15;
          const bool dead = qdead || (kpos >= Lk) ||
                            (causal && kpos > qpos);
          const float s = sc[qi][nt][r] * scale;
          const float p = dead ? 0.f : __expf(s - lse_r);
          pv[qi][nt][r] = p;
          dsv[qi][nt][r] = p * (dpc[qi][nt][r] - del_r) * scale;
        }
      }

    // ---- dQ += dS K  (A = dS via wave LDS, B = row-major K) --------------
    bf16_t* pt = Pt[wid];
    bf16_t* dst = DSt[wid];
    // write TRANSPOSED: pt[key * QB + qrow]  (A-frag for dV/dK)
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow = qi * 16 + lhi * 4 + r;
          const int key = nt * 16 + l15;
          pt[key * QB + qrow] = f2bf(pv[qi][nt][r]);
          dst[key * QB + qrow] = f2bf(dsv[qi][nt][r]);
        }
    // dS A-fragments (row = qrow l15, k = key lhi*8+j over all 32 keys in
    // ONE fragment — same shape as the forward's PV step): dst is
    // key-major, so the k-run (lhi*8..lhi*8+7 keys) is stride-QB; read as
    // scalars, row base per lane.
    f32x4_t dqc[2][8];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) dqc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
      bf16x8_t dsa;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dsa[j] = *reinterpret_cast<const __bf16*>(
            dst + (lhi * 8 + j) * QB + qi * 16 + l15);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        bf16x8_t kb;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          kb[j] = *reinterpret_cast<const __bf16*>(
              KsRow + (lhi * 8 + j) * HEAD_DIM + nt * 16 + l15);
        dqc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, kb, dqc[qi][nt], 0, 0, 0);
      }
    }
    // atomically add dQ (C: row = qi*16 + lhi*4 + r, col d = nt*16 + l15)
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_l = qb + qi * 16 + lhi * 4 + r;
        if (qrow_l >= Lq) continue;
        float* dqp = dq + ((long)(qbeg + qrow_l) * Hq + hq) * HEAD_DIM;
#pragma unroll
        for (int nt = 0; nt < 8; ++nt)
          atomicAdd(&dqp[nt * 16 + l15], dqc[qi][nt][r]);
      }

    // ---- dV += P^T dO ; dK += dS^T Q  (A from Pt/DSt, B from Qs/DOs) -----
    // A row = key (a*16 + l15), k = q row lhi*8+j (one 32-deep fragment);
    // Pt/DSt are key-major so the k-run is contiguous.
#pragma unroll
    for (int a = 0; a < 2; ++a) {            // key sub-tile (rows of dK/dV)
      const int keyrow = a * 16 + l15;
      const bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
          pt + keyrow * QB + lhi * 8);
      const bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
          dst + keyrow * QB + lhi * 8);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        bf16x8_t dob, qb_;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dob[j] = *reinterpret_cast<const __bf16*>(
              dos + (lhi * 8 + j) * HEAD_DIM + nt * 16 + l15);
          qb_[j] = *reinterpret_cast<const __bf16*>(
              qs + (lhi * 8 + j) * HEAD_DIM + nt * 16 + l15);
        }
        dv_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, dob, dv_acc[a][nt], 0, 0, 0);
        dk_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, qb_, dk_acc[a][nt], 0, 0, 0);
      }
    }
  }

  // ---- flush dK/dV (atomics: waves + GQA heads + q-tiles all merge) ------
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow_l = t0 + a * 16 + lhi * 4 + r;
      if (krow_l >= Lk) continue;
      float* dkp = dk + ((long)(kbeg + krow_l) * Hk + hk) * HEAD_DIM;
      float* dvp = dv + ((long)(kbeg + krow_l) * Hk + hk) * HEAD_DIM;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        atomicAdd(&dkp[nt * 16 + l15], dk_acc[a][nt][r]);
        atomicAdd(&dvp[nt * 16 + l15], dv_acc[a][nt][r]);
      }
    }
}

// ---------------------------------------------------------- backward v2
// Same math as attn_bwd_kernel, restructured LDS traffic:
//  * TRANSPOSED images KsRowT / QsT / DOsT ([d][x], 8-x-block XOR swizzle)
//    turn every B-fragment read of the dQ / dV / dK MFMAs into ONE
//    bf16x8 vector read (v1 did 8 scalar ds_read_u16 per fragment),
//  * Q / dO A-fragments load straight from global (L2-resident re-reads
//    across kv-tiles) and the transposed images are staged from those
//    SAME registers — no second pass over HBM, no row-major LDS copies,
//  * Pt / DSt get the same swizzle (their vector reads were bank-heavy).
// trid(d, x): element (d, x) of a [128][X] image, whole-8-block XOR keeps
// vector reads contiguous and 16-B aligned.
DEV_INLINE int trid(int d, int x, int X) {
  return d * X + (x ^ ((d & 3) << 3));
}
DEV_INLINE int trid_vec(int d, int xblock, int X) {  // x = xblock*8 .. +7
  return d * X + ((xblock * 8) ^ ((d & 3) << 3));
}

__global__ __launch_bounds__(256) void attn_bwd_kernel_v2(
    float* __restrict__ dq, float* __restrict__ dk, float* __restrict__ dv,
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    const int* __restrict__ cu_q, const int* __restrict__ cu_k,
    const int* __restrict__ tile_seq, const int* __restrict__ tile_k0,
    int Hq, int Hk, float scale, int causal,
    long ldq, long ldk, long ldv) {
  __shared__ bf16_t KsSwz[KB * HEAD_DIM];
  __shared__ bf16_t VsSwz[KB * HEAD_DIM];
  __shared__ bf16_t KsRowT[HEAD_DIM * KB];         // [d][key], swizzled
  __shared__ bf16_t QsT[NWAVE][HEAD_DIM * QB];     // [d][qrow], swizzled
  __shared__ bf16_t DOsT[NWAVE][HEAD_DIM * QB];
  __shared__ bf16_t Pt[NWAVE][KB * QB];            // [key][qrow], swizzled
  __shared__ bf16_t DSt[NWAVE][KB * QB];

  const int tile = blockIdx.x;
  const int hq = blockIdx.y;
  const int G = Hq / Hk;
  const int hk = hq / G;
  const int seq = tile_seq[tile];
  const int t0 = tile_k0[tile];
  const int qbeg = cu_q[seq], qend = cu_q[seq + 1];
  const int kbeg = cu_k[seq], kend = cu_k[seq + 1];
  const int Lq = qend - qbeg, Lk = kend - kbeg;
  const int qk_off = Lk - Lq;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;

  // ---- stage K/V: swizzled images + transposed K ------------------------
  {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int flat = (tid + it * 256) * 8;
      const int row = flat / HEAD_DIM;
      const int col = flat % HEAD_DIM;
      const int krow = t0 + row;
      bf16x8 k8, v8;
      if (krow < Lk) {
        k8 = *reinterpret_cast<const bf16x8*>(
            k + (long)(kbeg + krow) * ldk + (long)hk * HEAD_DIM + col);
        v8 = *reinterpret_cast<const bf16x8*>(
            v + (long)(kbeg + krow) * ldv + (long)hk * HEAD_DIM + col);
      } else {
        for (int j = 0; j < 8; ++j) k8.v[j] = f2bf(0.f), v8.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(KsSwz) + kswz(row, col * 2)) = k8;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(VsSwz) + kswz(row, col * 2)) = v8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        KsRowT[trid(col + j, row, KB)] = k8.v[j];
    }
  }
  __syncthreads();

  f32x4_t dv_acc[2][8], dk_acc[2][8];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      dv_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
      dk_acc[a][nt] = {0.f, 0.f, 0.f, 0.f};
    }

  int q_start = causal ? ((t0 - qk_off) > 0 ? (t0 - qk_off) : 0) : 0;
  q_start -= q_start % QB;

  for (int qb = q_start + wid * QB; qb < Lq; qb += NWAVE * QB) {
    // ---- A-fragments of Q / dO straight from global; stage transposed --
    bf16_t* qst = QsT[wid];
    bf16_t* dost = DOsT[wid];
    bf16x8_t qa[2][4], da[2][4];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int qrow_l = qb + half * 16 + l15;
      const bool live = qrow_l < Lq;
      const long gq = (long)(qbeg + (live ? qrow_l : 0));
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        if (live) {
          qa[half][ks] = *reinterpret_cast<const bf16x8_t*>(
              q + gq * ldq + (long)hq * HEAD_DIM + ks * 32 + lhi * 8);
          da[half][ks] = *reinterpret_cast<const bf16x8_t*>(
              dout + (gq * Hq + hq) * HEAD_DIM + ks * 32 + lhi * 8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qa[half][ks][j] = (__bf16)0.f;
            da[half][ks][j] = (__bf16)0.f;
          }
        }
        const int qrow = half * 16 + l15;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = ks * 32 + lhi * 8 + j;
          qst[trid(d, qrow, QB)] = (bf16_t)qa[half][ks][j];
          dost[trid(d, qrow, QB)] = (bf16_t)da[half][ks][j];
        }
      }
    }

    // ---- S = Q K^T, dP = dO V^T ---------------------------------------
    f32x4_t sc[2][2], dpc[2][2];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        sc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
        dpc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int krow = nt * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(KsSwz) + kswz(krow, (ks * 32 + lhi * 8) * 2));
        const bf16x8_t vb = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(VsSwz) + kswz(krow, (ks * 32 + lhi * 8) * 2));
#pragma unroll
        for (int qi = 0; qi < 2; ++qi) {
          sc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qa[qi][ks], kb, sc[qi][nt], 0, 0, 0);
          dpc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              da[qi][ks], vb, dpc[qi][nt], 0, 0, 0);
        }
      }
    }

    // ---- P, dS ---------------------------------------------------------
    float pv[2][2][4], dsv[2][2][4];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_l = qb + qi * 16 + lhi * 4 + r;
        const bool qdead = qrow_l >= Lq;
        const long gq = (long)(qbeg + (qdead ? 0 : qrow_l)) * Hq + hq;
        const float lse_r = qdead ? 0.f : lse[gq];
        const float del_r = qdead ? 0.f : delta[gq];
        const int qpos = qk_off + qrow_l;
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          const int kpos = t0 + nt * 16 + l15;
          const bool dead = qdead || (kpos >= Lk) ||
                            (causal && kpos > qpos);
          const float s = sc[qi][nt][r] * scale;
          const float p = dead ? 0.f : __expf(s - lse_r);
          pv[qi][nt][r] = p;
          dsv[qi][nt][r] = p * (dpc[qi][nt][r] - del_r) * scale;
        }
      }

    // ---- transposed P / dS (swizzled) ----------------------------------
    bf16_t* pt = Pt[wid];
    bf16_t* dst = DSt[wid];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow = qi * 16 + lhi * 4 + r;
          const int key = nt * 16 + l15;
          pt[trid(key, qrow, QB)] = f2bf(pv[qi][nt][r]);
          dst[trid(key, qrow, QB)] = f2bf(dsv[qi][nt][r]);
        }

    // ---- dQ += dS K  (A: swizzled scalar gather; B: KsRowT vector) -----
    f32x4_t dqc[2][8];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) dqc[qi][nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
      bf16x8_t dsa;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dsa[j] = *reinterpret_cast<const __bf16*>(
            dst + trid(lhi * 8 + j, qi * 16 + l15, QB));
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        const int d = nt * 16 + l15;
        const bf16x8_t kb = *reinterpret_cast<const bf16x8_t*>(
            KsRowT + trid_vec(d, lhi, KB));
        dqc[qi][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, kb, dqc[qi][nt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_l = qb + qi * 16 + lhi * 4 + r;
        if (qrow_l >= Lq) continue;
        float* dqp = dq + ((long)(qbeg + qrow_l) * Hq + hq) * HEAD_DIM;
#pragma unroll
        for (int nt = 0; nt < 8; ++nt)
          atomicAdd(&dqp[nt * 16 + l15], dqc[qi][nt][r]);
      }

    // ---- dV += P^T dO ; dK += dS^T Q  (all vector reads) ---------------
#pragma unroll
    for (int a = 0; a < 2; ++a) {
      const int keyrow = a * 16 + l15;
      const bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
          pt + trid_vec(keyrow, lhi, QB));
      const bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
          dst + trid_vec(keyrow, lhi, QB));
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        const int d = nt * 16 + l15;
        const bf16x8_t dob = *reinterpret_cast<const bf16x8_t*>(
            dost + trid_vec(d, lhi, QB));
        const bf16x8_t qb_ = *reinterpret_cast<const bf16x8_t*>(
            qst + trid_vec(d, lhi, QB));
        dv_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, dob, dv_acc[a][nt], 0, 0, 0);
        dk_acc[a][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, qb_, dk_acc[a][nt], 0, 0, 0);
      }
    }
  }

  // ---- flush dK/dV ------------------------------------------------------
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow_l = t0 + a * 16 + lhi * 4 + r;
      if (krow_l >= Lk) continue;
      float* dkp = dk + ((long)(kbeg + krow_l) * Hk + hk) * HEAD_DIM;
      float* dvp = dv + ((long)(kbeg + krow_l) * Hk + hk) * HEAD_DIM;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        atomicAdd(&dkp[nt * 16 + l15], dk_acc[a][nt][r]);
        atomicAdd(&dvp[nt * 16 + l15], dv_acc[a][nt][r]);
      }
    }
}

void varlen_attention_backward(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,  // fp32, zeroed
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor dout, torch::Tensor lse,
    torch::Tensor cu_seqlens_q, torch::Tensor cu_seqlens_k,
    torch::Tensor tile_seq, torch::Tensor tile_k0,
    double scale, bool causal, bool use_v2) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  auto packed = [](const torch::Tensor& t) {
    return t.stride(2) == 1 && t.stride(1) == t.size(2);
  };
  TORCH_CHECK(packed(q) && packed(k) && packed(v));
  TORCH_CHECK(out.is_contiguous() && dout.is_contiguous());
  TORCH_CHECK(lse.is_contiguous() && lse.dtype() == torch::kFloat32);
  TORCH_CHECK(dq.dtype() == torch::kFloat32 && dq.is_contiguous());
  const int Hq = q.size(1), D = q.size(2), Hk = k.size(1);
  TORCH_CHECK(D == 128, "backward supports head_dim 128");
  TORCH_CHECK(Hq % Hk == 0);
  const long tq = q.size(0);
  auto stream = at::hip::getCurrentHIPStream();

  auto delta = torch::empty({tq, (long)Hq}, lse.options());
  {
    const long rows = tq * Hq;
    attn_bwd_delta_kernel<<<dim3((rows + 3) / 4), dim3(256), 0, stream>>>(
        delta.data_ptr<float>(), (const bf16_t*)dout.data_ptr(),
        (const bf16_t*)out.data_ptr(), rows);
  }
  const int ntiles = tile_seq.size(0);
  auto kern = use_v2 ? attn_bwd_kernel_v2 : attn_bwd_kernel;
  kern<<<dim3(ntiles, Hq), dim3(256), 0, stream>>>(
      dq.data_ptr<float>(), dk.data_ptr<float>(), dv.data_ptr<float>(),
      (const bf16_t*)q.data_ptr(), (const bf16_t*)k.data_ptr(),
      (const bf16_t*)v.data_ptr(), (const bf16_t*)dout.data_ptr(),
      lse.data_ptr<float>(), delta.data_ptr<float>(),
      cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
      tile_seq.data_ptr<int>(), tile_k0.data_ptr<int>(), Hq, Hk,
      (float)scale, causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
  HIP_CHECK_KERNEL();
}
