// Fused per-token log-prob gather (+ entropy) over the vocab dim.
//
// Replaces the logprobs_from_logits / entropy_from_logits pair (reference
// capability: SURVEY.md §2.4.3 'logprobs_from_logits + entropy', consumed by
// compute_log_prob and the rollout logprob capture).  One streaming pass:
// online max + sum-exp + weighted-sum for entropy, then lp = x[label] - LZ.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// one block (4 waves) per row; V large (32k..160k)
template <typename T, bool WANT_ENTROPY>
__global__ void gather_logprobs_kernel(float* __restrict__ out_lp,
                                       float* __restrict__ out_ent,
                                       const T* __restrict__ logits,
                                       const int64_t* __restrict__ labels,
                                       int V) {
  __shared__ float red[8];
  const long row = blockIdx.x;
  const T* x = logits + row * V;

  // online (m, s, t): s = sum exp(x-m); t = sum exp(x-m)*x
  float m = -INFINITY, s = 0.f, t = 0.f;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    float xi = (float)x[i];
    if (xi > m) {
      float scale = __expf(m - xi);
      s *= scale;
      if (WANT_ENTROPY) t *= scale;
      m = xi;
    }
    float e = __expf(xi - m);
    s += e;
    if (WANT_ENTROPY) t += e * xi;
  }
  // merge thread-local streams across the block: rescale to global max
  float gm = block_reduce_max<4>(m, red);
  __syncthreads();
  float scale = (m == -INFINITY) ? 0.f : __expf(m - gm);
  s *= scale;
  float gs = block_reduce_sum<4>(s, red);
  __syncthreads();
  float gt = 0.f;
  if (WANT_ENTROPY) {
    t *= scale;
    gt = block_reduce_sum<4>(t, red);
  }
  if (threadIdx.x == 0) {
    float lz = gm + __logf(gs);
    out_lp[row] = (float)x[labels[row]] - lz;
    if (WANT_ENTROPY) out_ent[row] = lz - gt / gs;
  }
}

// ---------------------------------------------------- training backward
// d lp / d logits = onehot(label) - softmax(logits); chain with dlp.
// One streaming pass per row using the forward-saved logsumexp.
__global__ void gather_logprobs_bwd_kernel(
    bf16_t* __restrict__ dlogits,
    const bf16_t* __restrict__ logits,
    const int64_t* __restrict__ labels,
    const float* __restrict__ lse,
    const float* __restrict__ dlp,
    int V) {
  const long row = blockIdx.x;
  const bf16_t* x = logits + row * V;
  bf16_t* dx = dlogits + row * V;
  const float g = dlp[row];
  const float l = lse[row];
  const long lab = labels[row];
  const bf16x8* xp = reinterpret_cast<const bf16x8*>(x);
  bf16x8* dp = reinterpret_cast<bf16x8*>(dx);
  const int n8 = V / 8;
  for (int i = threadIdx.x; i < n8; i += blockDim.x) {
    bf16x8 v = xp[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = f2bf(-g * __expf(bf2f(v.v[j]) - l));
    dp[i] = o;
  }
  // tail when V % 8 != 0
  for (int i = n8 * 8 + threadIdx.x; i < V; i += blockDim.x)
    dx[i] = f2bf(-g * __expf(bf2f(x[i]) - l));
  // add the onehot term after every softmax write landed
  __syncthreads();
  if (threadIdx.x == 0)
    dx[lab] = f2bf(bf2f(dx[lab]) + g);
}

template <typename T>
static void launch_gather(torch::Tensor out_lp, torch::Tensor out_ent,
                          torch::Tensor logits, torch::Tensor labels,
                          bool want_entropy) {
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  if (want_entropy) {
    gather_logprobs_kernel<T, true><<<dim3(N), dim3(256), 0, stream>>>(
        out_lp.data_ptr<float>(), out_ent.data_ptr<float>(),
        (const T*)logits.data_ptr(), labels.data_ptr<int64_t>(), V);
  } else {
    gather_logprobs_kernel<T, false><<<dim3(N), dim3(256), 0, stream>>>(
        out_lp.data_ptr<float>(), nullptr, (const T*)logits.data_ptr(),
        labels.data_ptr<int64_t>(), V);
  }
  HIP_CHECK_KERNEL();
}

// training fwd: lp + logsumexp (for the hand-written backward)
__global__ void gather_logprobs_lse_kernel(
    float* __restrict__ out_lp, float* __restrict__ out_lse,
    const bf16_t* __restrict__ logits, const int64_t* __restrict__ labels,
    int V) {
  __shared__ float red[8];
  const long row = blockIdx.x;
  const bf16_t* x = logits + row * V;
  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    float xi = bf2f(x[i]);
    if (xi > m) {
      s *= __expf(m - xi);
      m = xi;
    }
    s += __expf(xi - m);
  }
  float gm = block_reduce_max<4>(m, red);
  __syncthreads();
  s *= (m == -INFINITY) ? 0.f : __expf(m - gm);
  float gs = block_reduce_sum<4>(s, red);
  if (threadIdx.x == 0) {
    float lz = gm + __logf(gs);
    out_lp[row] = bf2f(x[labels[row]]) - lz;
    out_lse[row] = lz;
  }
}

void gather_logprobs_train_fwd(torch::Tensor out_lp, torch::Tensor out_lse,
                               torch::Tensor logits, torch::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 &&
              logits.is_contiguous() &&
              logits.dtype() == torch::kBFloat16);
  TORCH_CHECK(labels.dtype() == torch::kInt64);
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto stream = at::hip::getCurrentHIPStream();
  gather_logprobs_lse_kernel<<<dim3(N), dim3(256), 0, stream>>>(
      out_lp.data_ptr<float>(), out_lse.data_ptr<float>(),
      (const bf16_t*)logits.data_ptr(), labels.data_ptr<int64_t>(), V);
  HIP_CHECK_KERNEL();
}

torch::Tensor gather_logprobs_train_bwd(torch::Tensor logits,
                                        torch::Tensor labels,
                                        torch::Tensor lse,
                                        torch::Tensor dlp) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() &&
              logits.dtype() == torch::kBFloat16);
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStream();
  gather_logprobs_bwd_kernel<<<dim3(N), dim3(256), 0, stream>>>(
      (bf16_t*)dlogits.data_ptr(), (const bf16_t*)logits.data_ptr(),
      labels.data_ptr<int64_t>(), lse.data_ptr<float>(),
      dlp.contiguous().data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  return dlogits;
}

void gather_logprobs(torch::Tensor out_lp, torch::Tensor out_ent,
                     torch::Tensor logits, torch::Tensor labels,
                     bool want_entropy) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.dtype() == torch::kInt64);
  TORCH_CHECK(out_lp.dtype() == torch::kFloat32);
  if (logits.dtype() == torch::kBFloat16) {
    launch_gather<bf16_t>(out_lp, out_ent, logits, labels, want_entropy);
  } else if (logits.dtype() == torch::kFloat32) {
    launch_gather<float>(out_lp, out_ent, logits, labels, want_entropy);
  } else {
    TORCH_CHECK(false, "gather_logprobs: dtype must be bf16 or fp32");
  }
}
