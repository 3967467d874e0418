// Varlen causal prefill attention, bf16, head_dim 64/128, GQA — MFMA forward.
//
// Reference capability: chunked-prefill flash attention (SURVEY.md §2.4.3
// row 2).  MI355X-first structure (flash-style online softmax, no S matrix
// materialized):
//   * grid = (q_tiles, Hq); a workgroup = 4 waves = 64 query rows of one
//     (sequence, q-head); wave w owns 16 rows.
//   * per KV tile (KVBLK=32 keys): K and V staged to LDS — K XOR-swizzled
//     (guide G4: row-major [32][128] bf16 would be a 16-way bank conflict on
//     ds_read_b128), V row-major.
//   * QK^T and P·V on v_mfma_f32_16x16x32_bf16.  Q fragments are
//     register-resident for the whole kernel.
//   * online softmax per row: running (m, l), O rescale by exp(m_old-m_new);
//     row stats via 16-lane shfl_xor reduce (C-fragment rows live in 16-lane
//     groups).
//   * P changes fragment layout (C-layout -> A-layout) through a wave-private
//     LDS tile (correctness-first; a permlane path can replace it later).
// Chunked-prefill semantics: query block aligned to the END of the keys
// (q row i at absolute kv position (Lk - Lq) + i), matching ops/ref.py.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define KVBLK 64
#define QROWS_PER_WAVE 16
#define NWAVE 4
#define QTILE (QROWS_PER_WAVE * NWAVE)  // 64

// K-tile swizzle: byte_col ^= (row & 7) << 4  (16-B granularity, preserves
// ds_read_b128 alignment; spreads a 16-lane group over 8 slots)
template <int D>
DEV_INLINE int kswz(int row, int byte_col) {
  return row * (D * 2) + (byte_col ^ ((row & 7) << 4));
}

template <int HEAD_DIM>
__global__ __launch_bounds__(256) void prefill_attn_kernel(
    bf16_t* __restrict__ out,        // (total_q, Hq, D)
    float* __restrict__ lse,         // (total_q, Hq) log-sum-exp, or null
    const bf16_t* __restrict__ q,    // (total_q, Hq, D)
    const bf16_t* __restrict__ k,    // (total_k, Hk, D)
    const bf16_t* __restrict__ v,    // (total_k, Hk, D)
    const int* __restrict__ cu_q,    // (B+1,)
    const int* __restrict__ cu_k,
    const int* __restrict__ tile_seq,  // (ntiles,)
    const int* __restrict__ tile_q0,   // (ntiles,) local first q row
    int Hq, int Hk, float scale, int causal, long ldq, long ldk, long ldv) {
  __shared__ bf16_t Ks[KVBLK * HEAD_DIM];          // swizzled
  __shared__ bf16_t Vs[KVBLK * HEAD_DIM];          // row-major
  __shared__ bf16_t Ps[NWAVE][QROWS_PER_WAVE * KVBLK];

  const int tile = blockIdx.x;
  const int hq = blockIdx.y;
  const int hk = hq / (Hq / Hk);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];                    // local
  const int qbeg = cu_q[seq], qend = cu_q[seq + 1];
  const int kbeg = cu_k[seq], kend = cu_k[seq + 1];
  const int Lq = qend - qbeg, Lk = kend - kbeg;
  const int qk_off = Lk - Lq;                      // chunked-prefill alignment

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int lhi = lane >> 4;                       // 0..3
  constexpr int KS = HEAD_DIM / 32;                // A/B k-steps over d
  constexpr int NDT = HEAD_DIM / 16;               // 16-wide d-tiles of O

  // this wave's q rows: local q0 + wid*16 + (row index 0..15)
  const int wq0 = q0 + wid * QROWS_PER_WAVE;

  // ---- Q fragments (A-layout): row i = l15, k-elem = lhi*8 + j ----
  // 4 k-steps cover d = 0..127.  Rows past Lq load row 0 (masked later).
  bf16x8_t qfrag[KS];
  {
    const int qrow_l = wq0 + l15;
    const long grow = (long)qbeg + ((qrow_l < Lq) ? qrow_l : 0);
    const bf16_t* qp = q + grow * ldq + (long)hq * HEAD_DIM + lhi * 8;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32);
  }

  // online-softmax state: rows (lhi*4 + r) for r=0..3 (C-layout rows)
  float m_run[4], l_run[4];
  f32x4_t o_acc[NDT];  // d-tiles of 16; reg r = row lhi*4+r
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int nt = 0; nt < NDT; ++nt) o_acc[nt] = {0.f, 0.f, 0.f, 0.f};

  // causal kv extent for this block
  const int max_qpos = qk_off + min(q0 + QTILE - 1, Lq - 1);
  const int kv_end = causal ? min(Lk, max_qpos + 1) : Lk;

  for (int t0 = 0; t0 < kv_end; t0 += KVBLK) {
    // ---------------- stage K (swizzled) and V (row-major) ----------------
    // 256 threads x 2 iters x 8 elems = 4096 elems = 32x128
    {
      const int tid = threadIdx.x;
#pragma unroll
      for (int it = 0; it < KVBLK * HEAD_DIM / (256 * 8); ++it) {
        const int flat = (tid + it * 256) * 8;
        const int row = flat / HEAD_DIM;
        const int col = flat % HEAD_DIM;
        const int krow = t0 + row;
        bf16x8 kv8, vv8;
        if (krow < Lk) {
          kv8 = *reinterpret_cast<const bf16x8*>(
              k + (long)(kbeg + krow) * ldk + (long)hk * HEAD_DIM + col);
          vv8 = *reinterpret_cast<const bf16x8*>(
              v + (long)(kbeg + krow) * ldv + (long)hk * HEAD_DIM + col);
        } else {
          for (int j = 0; j < 8; ++j) kv8.v[j] = f2bf(0.f), vv8.v[j] = f2bf(0.f);
        }
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(Ks) + kswz<HEAD_DIM>(row, col * 2)) = kv8;
        *reinterpret_cast<bf16x8*>(Vs + row * HEAD_DIM + col) = vv8;
      }
    }
    __syncthreads();

    // ---------------- QK^T: S[16 q][KVBLK keys] = Q @ K^T -----------------
    // B-frag (d x key): lane reads K[key = nt*16 + l15][d = ks*32 + lhi*8 ..]
    constexpr int NKT = KVBLK / 16;
    f32x4_t c[NKT];
#pragma unroll
    for (int nt = 0; nt < NKT; ++nt) c[nt] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < NKT; ++nt) {
      const int krow = nt * 16 + l15;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(Ks) +
            kswz<HEAD_DIM>(krow, (ks * 32 + lhi * 8) * 2));
        c[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bf, c[nt],
                                                        0, 0, 0);
      }
    }

    // ---------------- scale + causal mask ---------------------------------
    // C layout: row = lhi*4 + r, col = l15.  Interior tiles (every (q, k)
    // pair live: tile fully below the causal diagonal of the wave's
    // earliest row, all rows/keys in range) skip the per-element mask —
    // the kernels are VALU-bound (PMC: 16:1 VALU:MFMA), not memory-bound.
    float s[NKT][4];
    const bool interior = (t0 + KVBLK <= Lk) && (wq0 + 16 <= Lq) &&
                          (!causal || t0 + KVBLK - 1 <= qk_off + wq0);
    if (interior) {
#pragma unroll
      for (int nt = 0; nt < NKT; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) s[nt][r] = c[nt][r] * scale;
    } else {
#pragma unroll
      for (int nt = 0; nt < NKT; ++nt) {
        const int kpos = t0 + nt * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow_l = wq0 + lhi * 4 + r;
          const int qpos = qk_off + qrow_l;
          const bool dead = (kpos >= Lk) || (qrow_l >= Lq) ||
                            (causal && kpos > qpos);
          s[nt][r] = dead ? -1e30f : c[nt][r] * scale;
        }
      }
    }

    // ---------------- online softmax (per C-row) ---------------------------
    float p[NKT][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float rm = s[0][r];
#pragma unroll
      for (int nt = 1; nt < NKT; ++nt) rm = fmaxf(rm, s[nt][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) rm = fmaxf(rm, __shfl_xor(rm, off, 64));
      const float mn = fmaxf(m_run[r], rm);
      alpha[r] = __expf(m_run[r] - mn);
      m_run[r] = mn;
      float rs = 0.f;
#pragma unroll
      for (int nt = 0; nt < NKT; ++nt) {
        p[nt][r] = __expf(s[nt][r] - mn);
        rs += p[nt][r];
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) rs += __shfl_xor(rs, off, 64);
      l_run[r] = l_run[r] * alpha[r] + rs;
    }
    // rescale O
#pragma unroll
    for (int nt = 0; nt < NDT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[nt][r] *= alpha[r];

    // ---------------- P: C-layout -> A-layout through wave-private LDS ----
    bf16_t* ps = Ps[wid];
#pragma unroll
    for (int nt = 0; nt < NKT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        ps[(lhi * 4 + r) * KVBLK + nt * 16 + l15] = f2bf(p[nt][r]);
    // wave-private tile: the compiler orders ds_write -> ds_read by lgkmcnt

    // ---------------- P @ V ------------------------------------------------
    // k runs over the KVBLK keys in 32-deep steps (A-frag k = kq*32+lhi*8)
#pragma unroll
    for (int kq = 0; kq < KVBLK / 32; ++kq) {
      const bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
          ps + l15 * KVBLK + kq * 32 + lhi * 8);
#pragma unroll
      for (int nt = 0; nt < NDT; ++nt) {
        bf16x8_t bv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          bv[j] = *reinterpret_cast<const __bf16*>(
              Vs + (kq * 32 + lhi * 8 + j) * HEAD_DIM + nt * 16 + l15);
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bv,
                                                            o_acc[nt],
                                                            0, 0, 0);
      }
    }
    __syncthreads();  // all waves done with Ks/Vs before restage
  }

  // ---------------- epilogue: out = O / l; lse = m + log l ----------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow_l = wq0 + lhi * 4 + r;
    if (qrow_l >= Lq) continue;
    const float inv = (l_run[r] > 0.f) ? 1.0f / l_run[r] : 0.f;
    bf16_t* op = out + (((long)qbeg + qrow_l) * Hq + hq) * HEAD_DIM;
#pragma unroll
    for (int nt = 0; nt < NDT; ++nt)
      op[nt * 16 + l15] = f2bf(o_acc[nt][r] * inv);
    if (lse != nullptr && l15 == 0)
      lse[((long)qbeg + qrow_l) * Hq + hq] =
          (l_run[r] > 0.f) ? (m_run[r] + __logf(l_run[r])) : -1e30f;
  }
}

void varlen_prefill_attention(torch::Tensor out, torch::Tensor q,
                              torch::Tensor k, torch::Tensor v,
                              torch::Tensor cu_seqlens_q,
                              torch::Tensor cu_seqlens_k,
                              torch::Tensor tile_seq, torch::Tensor tile_q0,
                              double scale, bool causal,
                              torch::Tensor lse /* undefined or (tq,Hq) f32 */) {
  const bool want_lse = lse.defined() && lse.numel() > 0;
  if (want_lse) {
    TORCH_CHECK(lse.dtype() == torch::kFloat32 && lse.is_contiguous());
    TORCH_CHECK(lse.size(0) == q.size(0) && lse.size(1) == q.size(1));
  }
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  auto packed = [](const torch::Tensor& t) {
    return t.stride(2) == 1 && t.stride(1) == t.size(2);
  };
  TORCH_CHECK(packed(q) && packed(k) && packed(v),
              "q/k/v heads and dims must be packed (token stride free)");
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(cu_seqlens_q.dtype() == torch::kInt32);
  TORCH_CHECK(tile_seq.dtype() == torch::kInt32 && tile_q0.dtype() == torch::kInt32);
  const int Hq = q.size(1), D = q.size(2), Hk = k.size(1);
  TORCH_CHECK(D == 128 || D == 64, "prefill kernel supports head_dim 64/128");
  TORCH_CHECK(Hq % Hk == 0);
  const int ntiles = tile_seq.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  auto kern = (D == 128) ? prefill_attn_kernel<128> : prefill_attn_kernel<64>;
  kern<<<dim3(ntiles, Hq), dim3(256), 0, stream>>>(
      (bf16_t*)out.data_ptr(),
      want_lse ? lse.data_ptr<float>() : nullptr,
      (const bf16_t*)q.data_ptr(),
      (const bf16_t*)k.data_ptr(), (const bf16_t*)v.data_ptr(),
      cu_seqlens_q.data_ptr<int>(), cu_seqlens_k.data_ptr<int>(),
      tile_seq.data_ptr<int>(), tile_q0.data_ptr<int>(), Hq, Hk, (float)scale,
      causal ? 1 : 0, q.stride(0), k.stride(0), v.stride(0));
  HIP_CHECK_KERNEL();
}
