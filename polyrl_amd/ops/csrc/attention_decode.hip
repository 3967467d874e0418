// Paged decode attention (continuous-batching, one new token per sequence).
//
// Reference capability: the paged/continuous-batching decode attention the
// reference reaches through its inference engine.
//
// MI355X-first design: the op is HBM-bound (each step reads the whole KV
// history once), so the kernel is built around streaming K/V at 16 B/lane:
//   * grid = (B * Hk) workgroups (batch x kv-head); 4 waves each.
//   * a wave's 64 lanes split into GROUPS = 64/(D/8) lane-groups; each group
//     owns one key per iteration, lane j of a group loads K[t, hk, 8j:8j+8]
//     (one bf16x8 = 16 B -> coalesced 1 KiB per wave instruction).
//   * q(x G heads) is register-resident; scores via an in-group shfl_xor
//     reduce; online softmax state (m, l, o[8]) per (group, head) in VGPRs.
//   * partial states merge wave-internally by shfl (groups are lane slices),
//     then across the 4 waves through LDS.
// GQA group size G (= Hq/Hk) and head_dim D are compile-time (guide rule 20:
// runtime-indexed register arrays spill to scratch), dispatched below.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

template <int D, int G>
__global__ __launch_bounds__(256) void paged_decode_attn_kernel(
    bf16_t* __restrict__ out,              // (B, Hq, D)
    float* __restrict__ partial,           // (B, Hq, S, D+2) when nsplit>1
    const bf16_t* __restrict__ q,          // (B, Hq, D)
    const bf16_t* __restrict__ k_cache,    // (num_pages, page_size, Hk, D)
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ page_table,    // (B, max_pages)
    const int* __restrict__ context_lens,  // (B,)
    int Hq, int Hk, int page_size, int max_pages, float scale, long ldq,
    int nsplit) {
  constexpr int GL = D / 8;        // lanes per key group
  constexpr int KPW = 64 / GL;     // keys per wave per iteration
  constexpr int NW = 4;            // waves per block
  // flash-decoding context split (PMC-measured: at B*Hk < ~1024 blocks
  // the kernel is occupancy/latency-bound — 1 wave/SIMD cannot hide the
  // online-update chain; splitting the context across nsplit blocks per
  // (b, hk) restores the wave supply, a tiny merge kernel combines)
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
  const int grp = lane / GL;       // key-group within wave
  const int gl = lane % GL;        // lane within group
  const int d0 = gl * 8;           // this lane's d-slice

  // q registers: G heads x 8 elems (this lane's slice), pre-scaled
  float qr[G][8];
#pragma unroll
  for (int h = 0; h < G; ++h) {
    const bf16x8 qv = *reinterpret_cast<const bf16x8*>(
        q + ((long)b * ldq + (long)(hk * G + h) * D + d0));
#pragma unroll
    for (int j = 0; j < 8; ++j) qr[h][j] = bf2f(qv.v[j]) * scale;
  }

  // online-softmax state per head.  m starts at a large negative FINITE
  // value: merging two empty partials with -inf init would compute
  // exp(-inf - -inf) = NaN; with -1e30 the empty state contributes
  // exp(0)*0 = 0 and any real score immediately dominates the max.
  float m[G], l[G], o[G][8];
#pragma unroll
  for (int h = 0; h < G; ++h) {
    m[h] = -1e30f;
    l[h] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[h][j] = 0.f;
  }

  const int kv_stride = Hk * D;  // row stride inside a page slot
  const int* ptab = page_table + (long)b * max_pages;

  // stream keys: token t = cbeg + iter*NW*KPW + wid*KPW + grp
  for (int t0 = cbeg + wid * KPW + grp; t0 < cend; t0 += NW * KPW) {
    const int page = ptab[t0 / page_size];
    const long slot = (long)page * page_size + (t0 % page_size);
    const bf16_t* kp = k_cache + slot * kv_stride + hk * D + d0;
    const bf16_t* vp = v_cache + slot * kv_stride + hk * D + d0;
    const bf16x8 kv = *reinterpret_cast<const bf16x8*>(kp);
    const bf16x8 vv = *reinterpret_cast<const bf16x8*>(vp);
    float kf[8], vf[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      kf[j] = bf2f(kv.v[j]);
      vf[j] = bf2f(vv.v[j]);
    }
#pragma unroll
    for (int h = 0; h < G; ++h) {
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) s += qr[h][j] * kf[j];
      // reduce over the GL lanes of this group
#pragma unroll
      for (int off = GL / 2; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
      // online update (score s is group-uniform)
      const float mn = fmaxf(m[h], s);
      const float alpha = __expf(m[h] - mn);
      const float pt = __expf(s - mn);
      l[h] = l[h] * alpha + pt;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[h][j] = o[h][j] * alpha + pt * vf[j];
      m[h] = mn;
    }
  }

  // ---- merge the KPW groups within each wave (shfl across lane slices) ----
#pragma unroll
  for (int h = 0; h < G; ++h) {
#pragma unroll
    for (int step = 1; step < KPW; step <<= 1) {
      const int src = lane ^ (step * GL);
      const float om = __shfl(m[h], src, 64);
      const float ol = __shfl(l[h], src, 64);
      float ov[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = __shfl(o[h][j], src, 64);
      const float mn = fmaxf(m[h], om);
      const float a = __expf(m[h] - mn), bta = __expf(om - mn);
      l[h] = l[h] * a + ol * bta;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[h][j] = o[h][j] * a + ov[j] * bta;
      m[h] = mn;
    }
  }

  // ---- merge the 4 waves through LDS (group 0 of each wave holds state) ---
  // LDS layout: [NW][G][GL][10] floats: 8 o + m + l  (D=128,G<=8: 20.5 KiB)
  __shared__ float mrg[NW][G][GL][10];
  if (grp == 0) {
#pragma unroll
    for (int h = 0; h < G; ++h) {
#pragma unroll
      for (int j = 0; j < 8; ++j) mrg[wid][h][gl][j] = o[h][j];
      mrg[wid][h][gl][8] = m[h];
      mrg[wid][h][gl][9] = l[h];
    }
  }
  __syncthreads();
  // wave 0, group 0 merges and writes
  if (wid == 0 && grp == 0) {
#pragma unroll
    for (int h = 0; h < G; ++h) {
      float gm = -INFINITY;
#pragma unroll
      for (int w = 0; w < NW; ++w) gm = fmaxf(gm, mrg[w][h][gl][8]);
      float gl_sum = 0.f, go[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) go[j] = 0.f;
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float a = __expf(mrg[w][h][gl][8] - gm);
        gl_sum += mrg[w][h][gl][9] * a;
#pragma unroll
        for (int j = 0; j < 8; ++j) go[j] += mrg[w][h][gl][j] * a;
      }
      if (nsplit == 1) {
        const float inv = (gl_sum > 0.f) ? 1.0f / gl_sum : 0.f;
        bf16x8 ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) ov.v[j] = f2bf(go[j] * inv);
        *reinterpret_cast<bf16x8*>(
            out + (((long)b * Hq + hk * G + h) * D + d0)) = ov;
      } else {
        float* pp = partial +
            ((((long)b * Hq + hk * G + h) * nsplit) + sp) * (D + 2);
#pragma unroll
        for (int j = 0; j < 8; ++j) pp[d0 + j] = go[j];
        if (gl == 0) {
          pp[D] = gm;
          pp[D + 1] = gl_sum;
        }
      }
    }
  }
}

// combine the nsplit context partials of one (b, hq): sum-exp merge
template <int D>
__global__ void paged_decode_merge_kernel(
    bf16_t* __restrict__ out,              // (B, Hq, D)
    const float* __restrict__ partial,     // (B*Hq, S, D+2)
    int nsplit) {
  constexpr int EPL = D / 64;              // d-elems per lane (64 threads)
  const long row = blockIdx.x;             // b*Hq + hq
  const int lane = threadIdx.x;
  const float* base = partial + row * (long)nsplit * (D + 2);
  float gm = -1e30f;
  for (int sp = 0; sp < nsplit; ++sp)
    gm = fmaxf(gm, base[sp * (D + 2) + D]);
  float gl_sum = 0.f;
  float acc[EPL];
#pragma unroll
  for (int j = 0; j < EPL; ++j) acc[j] = 0.f;
  for (int sp = 0; sp < nsplit; ++sp) {
    const float* pp = base + sp * (D + 2);
    const float a = __expf(pp[D] - gm);
    gl_sum += pp[D + 1] * a;
#pragma unroll
    for (int j = 0; j < EPL; ++j)
      acc[j] += pp[lane + j * 64] * a;
  }
  const float inv = (gl_sum > 0.f) ? 1.0f / gl_sum : 0.f;
#pragma unroll
  for (int j = 0; j < EPL; ++j)
    out[row * D + lane + j * 64] = f2bf(acc[j] * inv);
}

// EXPERIMENTAL round-3 candidate (default OFF, opt-in POLYRL_DECODE_MFMA=1
// for on-box A/B): MFMA S/PV decode — moves the per-key dot products and
// P·V off the VALU (PMC: decode is exactly VALU-throughput-bound at
// B*Hk>=1024, profiles/PROFILES.md).  Textual include: -fno-gpu-rdc builds
// cannot launch kernels defined in another TU.  Falls back to the
// production VALU kernel for (D, G) combos the prototype doesn't cover.
#include "experimental/attention_decode_mfma.hip"

static bool use_mfma_decode() {
  static const bool v = []() {
    const char* e = getenv("POLYRL_DECODE_MFMA");
    return e && e[0] == '1';
  }();
  return v;
}

template <int D>
static void dispatch_g(torch::Tensor& out, const torch::Tensor& q,
                       const torch::Tensor& k_cache, const torch::Tensor& v_cache,
                       const torch::Tensor& page_table,
                       const torch::Tensor& context_lens, int Hq, int Hk,
                       int page_size, int max_pages, float scale,
                       hipStream_t stream) {
  const int G = Hq / Hk;
  const long B = q.size(0);
  // flash-decoding context split: pick nsplit from the grid size alone
  // (L-independent => hipGraph-replay-safe).  Target >= 1024 blocks so
  // the latency-bound online-update chain has wave supply; PMC measured
  // B=32 x Hk=8 (256 blocks) at 7x the KV-read bound from occupancy.
  int nsplit = 1;
  const long base_blocks = B * Hk;
  if (base_blocks < 1024) {
    nsplit = (int)((1024 + base_blocks - 1) / base_blocks);
    if (nsplit > 16) nsplit = 16;
  }
  float* ppart = nullptr;
  if (nsplit > 1) {
    // Thread-local grow-only workspace: a fresh torch::empty every call
    // would allocate DURING hipGraph capture (the engine's graphed decode
    // wraps this op) and can deadlock/fail the capture.  The warmup pass
    // that precedes every capture (rollout/engine.py::_decode_graphed)
    // sizes this buffer on the same thread, so the captured call reuses a
    // stable pointer and replays are correct (partials are written then
    // merged within one launch sequence).
    static thread_local torch::Tensor ws;
    const long need = B * Hq * (long)nsplit * (D + 2);
    if (!ws.defined() || ws.numel() < need ||
        ws.device() != q.device()) {
      ws = torch::empty({need}, q.options().dtype(torch::kFloat32));
    }
    ppart = ws.data_ptr<float>();
  }
  const dim3 grid(B * Hk * nsplit), block(256);
  auto args = [&](auto kern) {
    kern<<<grid, block, 0, stream>>>(
        (bf16_t*)out.data_ptr(), ppart, (const bf16_t*)q.data_ptr(),
        (const bf16_t*)k_cache.data_ptr(), (const bf16_t*)v_cache.data_ptr(),
        page_table.data_ptr<int>(), context_lens.data_ptr<int>(), Hq, Hk,
        page_size, max_pages, scale, q.stride(0), nsplit);
    if (nsplit > 1)
      paged_decode_merge_kernel<D><<<dim3(B * Hq), dim3(64), 0, stream>>>(
          (bf16_t*)out.data_ptr(), ppart, nsplit);
  };
  if (use_mfma_decode()) {
    // prototype coverage: D=128 G in {1,4,8}; D=64 G=4 (llama/qwen shapes)
    if constexpr (D == 128) {
      if (G == 1) { args(paged_decode_attn_mfma_kernel<128, 1>); return; }
      if (G == 4) { args(paged_decode_attn_mfma_kernel<128, 4>); return; }
      if (G == 8) { args(paged_decode_attn_mfma_kernel<128, 8>); return; }
    } else if constexpr (D == 64) {
      if (G == 4) { args(paged_decode_attn_mfma_kernel<64, 4>); return; }
    }
    // uncovered combo: fall through to the production VALU kernel
  }
  switch (G) {
    case 1: args(paged_decode_attn_kernel<D, 1>); break;
    case 2: args(paged_decode_attn_kernel<D, 2>); break;
    case 3: args(paged_decode_attn_kernel<D, 3>); break;
    case 4: args(paged_decode_attn_kernel<D, 4>); break;
    case 5: args(paged_decode_attn_kernel<D, 5>); break;
    case 6: args(paged_decode_attn_kernel<D, 6>); break;
    case 7: args(paged_decode_attn_kernel<D, 7>); break;
    case 8: args(paged_decode_attn_kernel<D, 8>); break;
    default:
      TORCH_CHECK(false, "GQA group size ", G, " unsupported (1..8)");
  }
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor page_table,
                            torch::Tensor context_lens, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q heads/dims must be packed (token stride free)");
  TORCH_CHECK(page_table.dtype() == torch::kInt32 && page_table.is_contiguous());
  TORCH_CHECK(context_lens.dtype() == torch::kInt32);
  const int Hq = q.size(1), D = q.size(2);
  const int Hk = k_cache.size(2);
  const int page_size = k_cache.size(1);
  const int max_pages = page_table.size(1);
  TORCH_CHECK(Hq % Hk == 0);
  auto stream = at::hip::getCurrentHIPStream();
  if (D == 128) {
    dispatch_g<128>(out, q, k_cache, v_cache, page_table, context_lens, Hq,
                    Hk, page_size, max_pages, (float)scale, stream);
  } else if (D == 64) {
    dispatch_g<64>(out, q, k_cache, v_cache, page_table, context_lens, Hq, Hk,
                   page_size, max_pages, (float)scale, stream);
  } else {
    TORCH_CHECK(false, "head_dim ", D, " unsupported (64 or 128)");
  }
  HIP_CHECK_KERNEL();
}
