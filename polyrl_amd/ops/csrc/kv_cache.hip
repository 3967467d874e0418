// Paged-KV-cache scatter: write each new token's K/V rows into its page slot.
//
// Layout (see polyrl_amd/rollout/kv_cache.py): k_cache / v_cache are
// (num_pages, page_size, Hk, D) bf16; slot_mapping[i] = page * page_size +
// offset for token i of the flattened new-token batch.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

__global__ void kv_append_kernel(bf16_t* __restrict__ k_cache,
                                 bf16_t* __restrict__ v_cache,
                                 const bf16_t* __restrict__ k,
                                 const bf16_t* __restrict__ v,
                                 const int* __restrict__ slot_mapping,
                                 int HkD /* Hk*D */, long ldk, long ldv) {
  const int tok = blockIdx.x;
  const long slot = slot_mapping[tok];
  if (slot < 0) return;  // padding token
  const int nvec = HkD / 8;
  const bf16x8* ks = reinterpret_cast<const bf16x8*>(k + (long)tok * ldk);
  const bf16x8* vs = reinterpret_cast<const bf16x8*>(v + (long)tok * ldv);
  bf16x8* kd = reinterpret_cast<bf16x8*>(k_cache + slot * HkD);
  bf16x8* vd = reinterpret_cast<bf16x8*>(v_cache + slot * HkD);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    kd[i] = ks[i];
    vd[i] = vs[i];
  }
}

void kv_cache_append(torch::Tensor k_cache, torch::Tensor v_cache,
                     torch::Tensor k, torch::Tensor v,
                     torch::Tensor slot_mapping) {
  // k/v: (N, Hk, D) — token stride may exceed Hk*D (views into fused qkv),
  // heads/dims packed.
  TORCH_CHECK(k_cache.is_cuda() && k_cache.dtype() == torch::kBFloat16);
  TORCH_CHECK(k.stride(-1) == 1 && v.stride(-1) == 1);
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt32);
  const int N = k.size(0);
  if (N == 0) return;
  const int HkD = (int)(k.size(1) * k.size(2));
  TORCH_CHECK(k.dim() == 3 && k.stride(1) == k.size(2), "heads must be packed");
  TORCH_CHECK(HkD % 8 == 0);
  auto stream = at::hip::getCurrentHIPStream();
  int threads = std::min(256, HkD / 8);
  kv_append_kernel<<<dim3(N), dim3(threads), 0, stream>>>(
      (bf16_t*)k_cache.data_ptr(), (bf16_t*)v_cache.data_ptr(),
      (const bf16_t*)k.data_ptr(), (const bf16_t*)v.data_ptr(),
      slot_mapping.data_ptr<int>(), HkD, k.stride(0), v.stride(0));
  HIP_CHECK_KERNEL();
}
