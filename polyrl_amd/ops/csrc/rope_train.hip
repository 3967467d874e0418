// Trainer-path NEOX RoPE, forward AND backward, bf16 (fp32 math).
//
// The autograd path's rotation (4 muls + 2 adds + torch.cat over the two
// D/2 halves, per q and per k, per layer) is ~14 elementwise passes; this
// is one pass each way.  Rotation backward is rotation by -theta:
//   fwd: y1 =  a*cos - b*sin ; y2 = b*cos + a*sin   (a = x[:D/2], b = x[D/2:])
//   bwd: da =  dy1*cos + dy2*sin ; db = -dy1*sin + dy2*cos
//
// x: (T, H, D) bf16 contiguous; cos/sin: (T, D/2) fp32 (per-token tables,
// models/llama.py RotaryCache).  Reference capability: the rotary apply
// inside flash-attn/HF models (SURVEY.md §2.4.3 RoPE row).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// one 64-lane wave per (token, head) row; lane i handles pair (i, i+D/2).
// D/2 <= 64 covers D in {64, 128}: lane i < D/2 active.
template <bool BWD>
__global__ __launch_bounds__(256) void rope_train_kernel(
    bf16_t* __restrict__ out,
    const bf16_t* __restrict__ x,
    const float* __restrict__ cos_t,   // (T, D/2)
    const float* __restrict__ sin_t,
    long rows /* T*H */, int H, int D) {
  const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const int half = D / 2;
  if (lane >= half) return;
  const long t = row / H;
  const float c = cos_t[t * half + lane];
  const float s = sin_t[t * half + lane];
  const bf16_t* xp = x + row * D;
  bf16_t* op = out + row * D;
  const float a = bf2f(xp[lane]);
  const float b = bf2f(xp[lane + half]);
  if (BWD) {
    op[lane] = f2bf(a * c + b * s);
    op[lane + half] = f2bf(-a * s + b * c);
  } else {
    op[lane] = f2bf(a * c - b * s);
    op[lane + half] = f2bf(b * c + a * s);
  }
}

torch::Tensor rope_train_apply(torch::Tensor x, torch::Tensor cos_t,
                               torch::Tensor sin_t, bool backward) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 &&
              x.is_contiguous());
  TORCH_CHECK(x.dim() == 3);
  TORCH_CHECK(cos_t.is_contiguous() && cos_t.dtype() == torch::kFloat32);
  const long T = x.size(0);
  const int H = x.size(1), D = x.size(2);
  TORCH_CHECK(D == 64 || D == 128, "rope_train supports head_dim 64/128");
  TORCH_CHECK(cos_t.size(0) == T && cos_t.size(1) == D / 2);
  auto out = torch::empty_like(x);
  const long rows = T * H;
  auto stream = at::hip::getCurrentHIPStream();
  auto kern = backward ? rope_train_kernel<true> : rope_train_kernel<false>;
  kern<<<dim3((rows + 3) / 4), dim3(256), 0, stream>>>(
      (bf16_t*)out.data_ptr(), (const bf16_t*)x.data_ptr(),
      cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), rows, H, D);
  HIP_CHECK_KERNEL();
  return out;
}
