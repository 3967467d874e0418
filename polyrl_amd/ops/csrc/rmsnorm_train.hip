// Trainer-path RMSNorm (+ optional fused residual add), forward AND
// backward, bf16 with fp32 internals.
//
// Why: the autograd path's eager norm (x.float() -> F.rms_norm -> .to(bf16)
// -> *w) materializes an fp32 copy of every (T, 4096) activation and runs
// ~6 elementwise passes per call; rocprof r01 attributed ~1.9 s of a
// 2-step window to eager elementwise (profiles/PROFILES.md item 4).  This
// fuses each norm to one read/write pass (fwd) + two passes (bwd).
//
// Math (fp32 accumulation):
//   h = x + res                  (res optional)
//   r = rsqrt(mean(h^2) + eps)   (saved per row for backward)
//   y = (h * r) * w              (cast to bf16 after the product)
// Backward:
//   s    = sum_j dy_j * w_j * h_j
//   dh   = r * dy * w - h * r^3/H * s     (same dh flows to x and res)
//   dw_j = sum_t dy[t,j] * h[t,j] * r[t]  (column-reduction kernel)
//
// Reference capability: the fused RMSNorm the reference stack reaches via
// flash-attn/apex inside verl models (SURVEY.md §2.4.3 RMSNorm row) —
// here with a hand-written backward as well.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// ------------------------------------------------------------- forward
// block = 256 threads, one row per block; H % 8 == 0
__global__ __launch_bounds__(256) void rmsnorm_train_fwd_kernel(
    bf16_t* __restrict__ y,          // (T, H)
    bf16_t* __restrict__ h_out,      // (T, H) x+res (== x when res null)
    float* __restrict__ rstd,        // (T,)
    const bf16_t* __restrict__ x,
    const bf16_t* __restrict__ res,  // nullable
    const bf16_t* __restrict__ w,
    int H, float eps) {
  const long row = blockIdx.x;
  const int tid = threadIdx.x;
  const bf16x8* xp = reinterpret_cast<const bf16x8*>(x + row * H);
  const bf16x8* rp = res ? reinterpret_cast<const bf16x8*>(res + row * H)
                         : nullptr;
  bf16x8* hp = reinterpret_cast<bf16x8*>(h_out + row * H);
  const int n8 = H / 8;
  float ssq = 0.f;
  for (int i = tid; i < n8; i += 256) {
    bf16x8 v = xp[i];
    if (rp) {
      bf16x8 r8 = rp[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = f2bf(bf2f(v.v[j]) + bf2f(r8.v[j]));
    }
    hp[i] = v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      ssq += f * f;
    }
  }
  __shared__ float lds[4];
  ssq = block_reduce_sum<4>(ssq, lds);
  const float r = rsqrtf(ssq / H + eps);
  if (tid == 0) rstd[row] = r;
  const bf16x8* wp = reinterpret_cast<const bf16x8*>(w);
  bf16x8* yp = reinterpret_cast<bf16x8*>(y + row * H);
  for (int i = tid; i < n8; i += 256) {
    bf16x8 v = hp[i], w8 = wp[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = f2bf(bf2f(v.v[j]) * r * bf2f(w8.v[j]));
    yp[i] = o;
  }
}

// ------------------------------------------------------------ backward dh
__global__ __launch_bounds__(256) void rmsnorm_train_bwd_dx_kernel(
    bf16_t* __restrict__ dx,         // (T, H)
    const bf16_t* __restrict__ dy,
    const bf16_t* __restrict__ h,
    const bf16_t* __restrict__ w,
    const float* __restrict__ rstd,
    int H) {
  const long row = blockIdx.x;
  const int tid = threadIdx.x;
  const bf16x8* dyp = reinterpret_cast<const bf16x8*>(dy + row * H);
  const bf16x8* hp = reinterpret_cast<const bf16x8*>(h + row * H);
  const bf16x8* wp = reinterpret_cast<const bf16x8*>(w);
  bf16x8* dxp = reinterpret_cast<bf16x8*>(dx + row * H);
  const int n8 = H / 8;
  const float r = rstd[row];
  float s = 0.f;
  for (int i = tid; i < n8; i += 256) {
    bf16x8 d8 = dyp[i], h8 = hp[i], w8 = wp[i];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      s += bf2f(d8.v[j]) * bf2f(w8.v[j]) * bf2f(h8.v[j]);
  }
  __shared__ float lds[4];
  s = block_reduce_sum<4>(s, lds);
  const float k = r * r * r / H * s;
  for (int i = tid; i < n8; i += 256) {
    bf16x8 d8 = dyp[i], h8 = hp[i], w8 = wp[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = f2bf(r * bf2f(d8.v[j]) * bf2f(w8.v[j]) - bf2f(h8.v[j]) * k);
    dxp[i] = o;
  }
}

// ------------------------------------------------------------ backward dw
// column reduction: thread covers fixed columns, strides over rows
// (consecutive threads read consecutive columns -> coalesced)
__global__ __launch_bounds__(256) void rmsnorm_train_bwd_dw_kernel(
    float* __restrict__ dw,          // (H,) fp32, zeroed
    const bf16_t* __restrict__ dy,
    const bf16_t* __restrict__ h,
    const float* __restrict__ rstd,
    long T, int H, int rows_per_block) {
  const int col = blockIdx.x * 256 + threadIdx.x;
  if (col >= H) return;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min(r0 + rows_per_block, T);
  float acc = 0.f;
  for (long t = r0; t < r1; ++t)
    acc += bf2f(dy[t * H + col]) * bf2f(h[t * H + col]) * rstd[t];
  if (gridDim.y == 1)
    dw[col] = acc;
  else
    atomicAdd(&dw[col], acc);
}

// ---------------------------------------------------------------- launchers
std::vector<torch::Tensor> rmsnorm_train_fwd(torch::Tensor x,
                                             torch::Tensor res,  // undef ok
                                             torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 &&
              x.is_contiguous());
  const long T = x.numel() / x.size(-1);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0);
  const bool has_res = res.defined() && res.numel() > 0;
  if (has_res) TORCH_CHECK(res.is_contiguous() && res.sizes() == x.sizes());
  auto y = torch::empty_like(x);
  auto h = torch::empty_like(x);
  auto rstd = torch::empty({T}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  rmsnorm_train_fwd_kernel<<<dim3(T), dim3(256), 0, stream>>>(
      (bf16_t*)y.data_ptr(), (bf16_t*)h.data_ptr(), rstd.data_ptr<float>(),
      (const bf16_t*)x.data_ptr(),
      has_res ? (const bf16_t*)res.data_ptr() : nullptr,
      (const bf16_t*)w.data_ptr(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {y, h, rstd};
}

std::vector<torch::Tensor> rmsnorm_train_bwd(torch::Tensor dy,
                                             torch::Tensor h,
                                             torch::Tensor w,
                                             torch::Tensor rstd) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
  auto dyc = dy.contiguous();
  const long T = h.numel() / h.size(-1);
  const int H = h.size(-1);
  auto dx = torch::empty_like(h);
  // dw zero-filled: gridDim.y partial sums accumulate with atomics
  auto dw = torch::zeros({(long)H}, h.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  rmsnorm_train_bwd_dx_kernel<<<dim3(T), dim3(256), 0, stream>>>(
      (bf16_t*)dx.data_ptr(), (const bf16_t*)dyc.data_ptr(),
      (const bf16_t*)h.data_ptr(), (const bf16_t*)w.data_ptr(),
      rstd.data_ptr<float>(), H);
  // dw: split rows across gridDim.y so the grid reaches ~256 blocks
  int xblocks = (H + 255) / 256;
  int rows_per_block = (int)((T + 255) / 256) * 16;   // ~256/xblocks rows
  rows_per_block = rows_per_block < 256 ? 256 : rows_per_block;
  long yblocks = (T + rows_per_block - 1) / rows_per_block;
  rmsnorm_train_bwd_dw_kernel<<<dim3(xblocks, yblocks), dim3(256),
                                0, stream>>>(
      dw.data_ptr<float>(), (const bf16_t*)dyc.data_ptr(),
      (const bf16_t*)h.data_ptr(), rstd.data_ptr<float>(), T, H,
      rows_per_block);
  HIP_CHECK_KERNEL();
  return {dx, dw};
}
