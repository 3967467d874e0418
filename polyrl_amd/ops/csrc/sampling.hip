// Fused sampling: temperature + top-k + top-p filter + categorical draw +
// sampled-token logprob capture, one workgroup per row.
//
// Reference capability: the engine-side sampling + per-token logprob kernels
// (SURVEY.md §2.4.3).  Design (MI355X-first, no sort):
//  * top-k threshold = k-th largest logit, found by LDS-histogram radix
//    select over the monotone uint32 encoding of the logit (temperature
//    doesn't change order, so select on raw logits).
//  * top-p threshold found by the same radix walk, accumulating per-bin
//    sum(exp(x/T - m/T)) mass instead of counts.
//  * the draw is Gumbel-argmax over the kept set: argmax(x/T + G_i), exact
//    categorical sampling in one streaming pass, counter-based RNG
//    (splitmix64 keyed by (seed, row, i)) so replay is deterministic.
//  * reported logprob is under the raw (unfiltered, T=1) softmax, matching
//    what rollout clients consume as output_token_logprobs.
// bf16 logits have 16 payload bits -> 2 radix rounds are exact; fp32 runs 4.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

DEV_INLINE uint32_t enc_f32(float x) {
  uint32_t b = __float_as_uint(x);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}

DEV_INLINE uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

// uniform in (0,1]
DEV_INLINE float rng_uniform(uint64_t seed, uint64_t idx) {
  uint64_t r = splitmix64(seed ^ (idx * 0xd1342543de82ef95ull + 0x2545f4914f6cdd1dull));
  return ((r >> 40) + 1.0f) * (1.0f / 16777216.0f);  // 24-bit mantissa
}

struct ArgMax {
  float v;
  int i;
};

template <int NW>
DEV_INLINE ArgMax block_argmax(float v, int idx, float* lds_v, int* lds_i) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(v, off, 64);
    int oi = __shfl_xor(idx, off, 64);
    if (ov > v || (ov == v && oi < idx)) { v = ov; idx = oi; }
  }
  if (lane == 0) { lds_v[wid] = v; lds_i[wid] = idx; }
  __syncthreads();
  // re-broadcast through LDS: a shfl would only reach wave 0
  if (threadIdx.x == 0) {
    float rv = lds_v[0];
    int ri = lds_i[0];
#pragma unroll
    for (int w = 1; w < NW; ++w) {
      if (lds_v[w] > rv || (lds_v[w] == rv && lds_i[w] < ri)) {
        rv = lds_v[w];
        ri = lds_i[w];
      }
    }
    lds_v[0] = rv;
    lds_i[0] = ri;
  }
  __syncthreads();
  ArgMax out;
  out.v = lds_v[0];
  out.i = lds_i[0];
  return out;
}

// radix walk: find enc-threshold tau such that the kept set {enc >= tau}
// satisfies the constraint (count >= k for top-k, mass >= target for top-p).
// COUNT mode: WANT_MASS=false.  hist built over elements matching prefix.
template <typename T, bool WANT_MASS>
DEV_INLINE uint32_t radix_select(const T* __restrict__ x, int V, int rounds,
                                 float inv_temp, float m_scaled,
                                 float need_count, float target_mass,
                                 uint32_t* h_cnt, float* h_mass) {
  uint32_t prefix = 0;  // high bits decided so far (left-aligned)
  float committed_cnt = 0.f, committed_mass = 0.f;
  for (int r = 0; r < rounds; ++r) {
    const int shift = 24 - 8 * r;
    // zero histogram
    for (int b = threadIdx.x; b < 256; b += blockDim.x) {
      h_cnt[b] = 0;
      if (WANT_MASS) h_mass[b] = 0.f;
    }
    __syncthreads();
    const uint32_t pmask = (r == 0) ? 0u : (0xFFFFFFFFu << (shift + 8));
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      float xi = (float)x[i];
      uint32_t e = enc_f32(xi);
      if ((e & pmask) != prefix) continue;
      uint32_t bin = (e >> shift) & 0xFFu;
      atomicAdd(&h_cnt[bin], 1u);
      if (WANT_MASS) atomicAdd(&h_mass[bin], __expf(xi * inv_temp - m_scaled));
    }
    __syncthreads();
    // single-thread walk (256 bins) then broadcast via LDS
    __shared__ uint32_t sel_bin;
    __shared__ float sel_cnt, sel_mass;
    if (threadIdx.x == 0) {
      float cum_c = committed_cnt, cum_m = committed_mass;
      int b = 255;
      for (; b >= 0; --b) {
        cum_c += (float)h_cnt[b];
        if (WANT_MASS) cum_m += h_mass[b];
        bool ok = WANT_MASS ? (cum_m >= target_mass) : (cum_c >= need_count);
        if (ok) break;
      }
      if (b < 0) b = 0;  // numerical slack: keep everything in lowest bin
      sel_bin = (uint32_t)b;
      sel_cnt = cum_c - (float)h_cnt[b];
      sel_mass = WANT_MASS ? (cum_m - h_mass[b]) : 0.f;
    }
    __syncthreads();
    prefix |= (sel_bin << shift);
    committed_cnt = sel_cnt;
    committed_mass = sel_mass;
    __syncthreads();
  }
  // tau = lower edge of the final selected bin
  return prefix;
}

// Fast path: no top-k / top-p filter (the common RL rollout setting).
// ONE streaming pass computes simultaneously, 8 elements per lane per step
// (vectorized bf16x8/float4 loads, G13):
//   * online max m + sumexp s (for the reported raw logprob),
//   * raw argmax (greedy),
//   * Gumbel-argmax z = x/T + G_i with the winner's logit tracked inline.
// 1024 threads/block, one block per row: at decode batch >= 256 the whole
// chip is busy; a second grid dim is unnecessary because decode sampling
// runs alongside nothing.
template <typename T, int VEC>
__global__ __launch_bounds__(1024) void sample_fast_kernel(
    int64_t* __restrict__ out_tokens, float* __restrict__ out_logprobs,
    const T* __restrict__ logits, const float* __restrict__ temperature,
    uint64_t seed, const long long* __restrict__ seed_ptr, int V) {
  if (seed_ptr != nullptr) seed = (uint64_t)*seed_ptr;
  __shared__ float red_f[16];
  __shared__ int red_i[16];
  const long row = blockIdx.x;
  const T* x = logits + row * V;
  const float temp = temperature[row];
  const bool greedy = (temp == 0.f);
  const float inv_temp = greedy ? 1.f : (1.0f / temp);
  const uint64_t rowkey = seed + (uint64_t)row * 0x9e3779b97f4a7c15ull;

  float m = -INFINITY, s = 0.f;       // online raw softmax state
  float rmax = -INFINITY;             // raw argmax (greedy pick)
  int rarg = 0;
  float best = -INFINITY;             // gumbel winner
  int best_i = 0;
  float best_logit = 0.f;

  const int nvec = V / VEC;
  using vec_t = typename std::conditional<sizeof(T) == 2, bf16x8, float4>::type;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    vec_t pack = reinterpret_cast<const vec_t*>(x)[i];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float xi;
      if constexpr (sizeof(T) == 2) xi = bf2f(((const bf16x8&)pack).v[j]);
      else xi = ((const float*)&pack)[j];
      const int idx = i * VEC + j;
      if (xi > rmax) { rmax = xi; rarg = idx; }
      if (xi > m) { s *= __expf(m - xi); m = xi; }
      s += __expf(xi - m);
      if (!greedy) {
        float u = rng_uniform(rowkey, (uint64_t)idx);
        float z = __fmaf_rn(xi, inv_temp, -__logf(-__logf(u)));
        if (z > best) { best = z; best_i = idx; best_logit = xi; }
      }
    }
  }
  for (int idx = nvec * VEC + threadIdx.x; idx < V; idx += blockDim.x) {
    float xi = (float)x[idx];
    if (xi > rmax) { rmax = xi; rarg = idx; }
    if (xi > m) { s *= __expf(m - xi); m = xi; }
    s += __expf(xi - m);
    if (!greedy) {
      float u = rng_uniform(rowkey, (uint64_t)idx);
      float z = __fmaf_rn(xi, inv_temp, -__logf(-__logf(u)));
      if (z > best) { best = z; best_i = idx; best_logit = xi; }
    }
  }

  // block reductions (16 waves)
  ArgMax am = block_argmax<16>(rmax, rarg, red_f, red_i);
  __syncthreads();
  const float gm = am.v;
  s *= (m == -INFINITY) ? 0.f : __expf(m - gm);
  float gs = block_reduce_sum<16>(s, red_f);
  __syncthreads();
  const float lz_raw = gm + __logf(gs);

  if (greedy) {
    if (threadIdx.x == 0) {
      out_tokens[row] = am.i;
      out_logprobs[row] = (float)x[am.i] - lz_raw;
    }
    return;
  }
  // winner's raw logit rides along through a second argmax reduce keyed on z
  ArgMax pick = block_argmax<16>(best, best_i, red_f, red_i);
  if (threadIdx.x == 0) {
    out_tokens[row] = pick.i;
    out_logprobs[row] = (float)x[pick.i] - lz_raw;
  }
  (void)best_logit;
}

template <typename T>
__global__ void sample_kernel(int64_t* __restrict__ out_tokens,
                              float* __restrict__ out_logprobs,
                              const T* __restrict__ logits,
                              const float* __restrict__ temperature,
                              const int* __restrict__ top_k,
                              const float* __restrict__ top_p,
                              uint64_t seed, int V, int rounds) {
  __shared__ float red_f[8];
  __shared__ int red_i[8];
  __shared__ uint32_t h_cnt[256];
  __shared__ float h_mass[256];

  const long row = blockIdx.x;
  const T* x = logits + row * V;
  const float temp = temperature[row];
  const int k = top_k[row];
  const float p = top_p[row];

  // pass 1: raw max + argmax + raw sumexp (for logprob report / greedy)
  float m = -INFINITY, s = 0.f;
  float local_max = -INFINITY;
  int local_arg = 0;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    float xi = (float)x[i];
    if (xi > local_max) { local_max = xi; local_arg = i; }
    if (xi > m) {
      s *= __expf(m - xi);
      m = xi;
    }
    s += __expf(xi - m);
  }
  ArgMax am = block_argmax<4>(local_max, local_arg, red_f, red_i);
  __syncthreads();
  float gm = am.v;
  s *= (m == -INFINITY) ? 0.f : __expf(m - gm);
  float gs = block_reduce_sum<4>(s, red_f);
  __syncthreads();
  const float lz_raw = gm + __logf(gs);

  if (temp == 0.f) {  // greedy
    if (threadIdx.x == 0) {
      out_tokens[row] = am.i;
      out_logprobs[row] = (float)x[am.i] - lz_raw;
    }
    return;
  }

  const float inv_temp = 1.0f / temp;
  // thresholds (enc-space); default: keep everything
  uint32_t tau = 0;
  if (k > 0 && k < V) {
    uint32_t tk = radix_select<T, false>(x, V, rounds, inv_temp, 0.f,
                                         (float)k, 0.f, h_cnt, h_mass);
    tau = max(tau, tk);
    __syncthreads();
  }
  if (p < 1.0f) {
    // total scaled mass
    float st = 0.f;
    const float m_scaled = gm * inv_temp;
    for (int i = threadIdx.x; i < V; i += blockDim.x)
      st += __expf((float)x[i] * inv_temp - m_scaled);
    float gst = block_reduce_sum<4>(st, red_f);
    __syncthreads();
    uint32_t tp = radix_select<T, true>(x, V, rounds, inv_temp, m_scaled,
                                        0.f, p * gst, h_cnt, h_mass);
    tau = max(tau, tp);
    __syncthreads();
  }

  // Gumbel-argmax over kept set
  float best = -INFINITY;
  int best_i = 0;
  const uint64_t rowkey = seed + (uint64_t)row * 0x9e3779b97f4a7c15ull;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    float xi = (float)x[i];
    if (enc_f32(xi) < tau) continue;
    float u = rng_uniform(rowkey, (uint64_t)i);
    float g = -__logf(-__logf(u));
    float z = xi * inv_temp + g;
    if (z > best) { best = z; best_i = i; }
  }
  ArgMax pick = block_argmax<4>(best, best_i, red_f, red_i);
  if (threadIdx.x == 0) {
    out_tokens[row] = pick.i;
    out_logprobs[row] = (float)x[pick.i] - lz_raw;
  }
}

void sample(torch::Tensor out_tokens, torch::Tensor out_logprobs,
            torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor top_k, torch::Tensor top_p, int64_t seed,
            bool no_filter, torch::Tensor seed_dev /* optional (1,) i64 */) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(out_tokens.dtype() == torch::kInt64);
  TORCH_CHECK(out_logprobs.dtype() == torch::kFloat32);
  TORCH_CHECK(temperature.dtype() == torch::kFloat32);
  TORCH_CHECK(top_k.dtype() == torch::kInt32);
  TORCH_CHECK(top_p.dtype() == torch::kFloat32);
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  if (no_filter) {  // fused single-pass path (no top-k/top-p), G13-vectorized
    const long long* sp = seed_dev.defined() && seed_dev.numel() > 0
                              ? (const long long*)seed_dev.data_ptr<int64_t>()
                              : nullptr;
    if (logits.dtype() == torch::kBFloat16) {
      sample_fast_kernel<bf16_t, 8><<<dim3(N), dim3(1024), 0, stream>>>(
          out_tokens.data_ptr<int64_t>(), out_logprobs.data_ptr<float>(),
          (const bf16_t*)logits.data_ptr(), temperature.data_ptr<float>(),
          (uint64_t)seed, sp, V);
    } else {
      sample_fast_kernel<float, 4><<<dim3(N), dim3(1024), 0, stream>>>(
          out_tokens.data_ptr<int64_t>(), out_logprobs.data_ptr<float>(),
          (const float*)logits.data_ptr(), temperature.data_ptr<float>(),
          (uint64_t)seed, sp, V);
    }
    HIP_CHECK_KERNEL();
    return;
  }
  if (logits.dtype() == torch::kBFloat16) {
    sample_kernel<bf16_t><<<dim3(N), dim3(256), 0, stream>>>(
        out_tokens.data_ptr<int64_t>(), out_logprobs.data_ptr<float>(),
        (const bf16_t*)logits.data_ptr(), temperature.data_ptr<float>(),
        top_k.data_ptr<int>(), top_p.data_ptr<float>(), (uint64_t)seed, V,
        /*rounds=*/2);
  } else if (logits.dtype() == torch::kFloat32) {
    sample_kernel<float><<<dim3(N), dim3(256), 0, stream>>>(
        out_tokens.data_ptr<int64_t>(), out_logprobs.data_ptr<float>(),
        (const float*)logits.data_ptr(), temperature.data_ptr<float>(),
        top_k.data_ptr<int>(), top_p.data_ptr<float>(), (uint64_t)seed, V,
        /*rounds=*/4);
  } else {
    TORCH_CHECK(false, "sample: dtype must be bf16 or fp32");
  }
  HIP_CHECK_KERNEL();
}
