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
