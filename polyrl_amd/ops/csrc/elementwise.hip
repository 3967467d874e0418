// Memory-bound decoder-layer kernels: RMSNorm, fused add+RMSNorm, SiLU-mul,
// RoPE (NEOX rotate-half, table-driven).
//
// All bf16 I/O is vectorized as bf16x8 (16 B/lane, guide G13); accumulation
// in fp32.  Reference semantics: polyrl_amd/ops/ref.py.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// ------------------------------------------------------------------ RMSNorm
// one block per row; D multiple of 8; block 256 threads.
template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(bf16_t* __restrict__ out,
                               bf16_t* __restrict__ residual,  // inout if FUSED_ADD
                               const bf16_t* __restrict__ x,
                               const bf16_t* __restrict__ weight,
                               int D, float eps) {
  __shared__ float red[8];
  const int row = blockIdx.x;
  const bf16_t* xr = x + (long)row * D;
  bf16_t* rr = FUSED_ADD ? residual + (long)row * D : nullptr;
  bf16_t* outr = out + (long)row * D;

  const int nvec = D / 8;
  float ss = 0.f;
  // pass 1: (optionally add residual and write it back), accumulate sum sq
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(xr)[i];
    if (FUSED_ADD) {
      bf16x8 r = reinterpret_cast<const bf16x8*>(rr)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = f2bf(bf2f(v.v[j]) + bf2f(r.v[j]));
      reinterpret_cast<bf16x8*>(rr)[i] = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v.v[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum<4>(ss, red);
  const float inv = rsqrtf(ss / D + eps);
  __syncthreads();  // red reuse barrier (block_reduce wrote LDS)

  // pass 2: normalize*weight (re-read the written residual if fused)
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = FUSED_ADD ? reinterpret_cast<const bf16x8*>(rr)[i]
                         : reinterpret_cast<const bf16x8*>(xr)[i];
    bf16x8 w = reinterpret_cast<const bf16x8*>(weight)[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = f2bf(bf2f(v.v[j]) * inv * bf2f(w.v[j]));
    reinterpret_cast<bf16x8*>(outr)[i] = o;
  }
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "hidden dim must be a multiple of 8");
  const long rows = x.numel() / D;
  auto stream = at::hip::getCurrentHIPStream();
  rmsnorm_kernel<false><<<dim3(rows), dim3(256), 0, stream>>>(
      (bf16_t*)out.data_ptr(), nullptr, (const bf16_t*)x.data_ptr(),
      (const bf16_t*)weight.data_ptr(), D, (float)eps);
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm(torch::Tensor out, torch::Tensor residual,
                       torch::Tensor x, torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous() && out.is_contiguous());
  const int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0);
  const long rows = x.numel() / D;
  auto stream = at::hip::getCurrentHIPStream();
  rmsnorm_kernel<true><<<dim3(rows), dim3(256), 0, stream>>>(
      (bf16_t*)out.data_ptr(), (bf16_t*)residual.data_ptr(),
      (const bf16_t*)x.data_ptr(), (const bf16_t*)weight.data_ptr(), D,
      (float)eps);
  HIP_CHECK_KERNEL();
}

// ----------------------------------------------------------------- SiLU-mul
// gate/up may be strided row views into a fused (N, 2I) gate_up projection
// (row stride ldg/ldu elements, inner dim contiguous) — avoids the
// .contiguous() copies in the engine forward.
__global__ void silu_mul_kernel(bf16_t* __restrict__ out,
                                const bf16_t* __restrict__ gate,
                                const bf16_t* __restrict__ up, long rows,
                                int cvec /* cols/8 */, long ldg, long ldu) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const long nvec = rows * cvec;
  for (; i < nvec; i += stride) {
    const long r = i / cvec;
    const long c = i % cvec;
    bf16x8 g = *reinterpret_cast<const bf16x8*>(gate + r * ldg + c * 8);
    bf16x8 u = *reinterpret_cast<const bf16x8*>(up + r * ldu + c * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      float s = gf / (1.f + __expf(-gf));
      o.v[j] = f2bf(s * bf2f(u.v[j]));
    }
    reinterpret_cast<bf16x8*>(out)[i] = o;
  }
}

// training backward of silu(gate)*up: one pass, contiguous operands
// dgate = dy * up * (sig + g*sig*(1-sig)); dup = dy * silu(g)
__global__ void silu_mul_bwd_kernel(bf16_t* __restrict__ dgate,
                                    bf16_t* __restrict__ dup,
                                    const bf16_t* __restrict__ dy,
                                    const bf16_t* __restrict__ gate,
                                    const bf16_t* __restrict__ up,
                                    long nvec) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) {
    bf16x8 d = reinterpret_cast<const bf16x8*>(dy)[i];
    bf16x8 g = reinterpret_cast<const bf16x8*>(gate)[i];
    bf16x8 u = reinterpret_cast<const bf16x8*>(up)[i];
    bf16x8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(g.v[j]);
      const float df = bf2f(d.v[j]);
      const float sig = 1.f / (1.f + __expf(-gf));
      const float silu = gf * sig;
      og.v[j] = f2bf(df * bf2f(u.v[j]) * (sig + gf * sig * (1.f - sig)));
      ou.v[j] = f2bf(df * silu);
    }
    reinterpret_cast<bf16x8*>(dgate)[i] = og;
    reinterpret_cast<bf16x8*>(dup)[i] = ou;
  }
}

void silu_mul_bwd(torch::Tensor dgate, torch::Tensor dup, torch::Tensor dy,
                  torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16);
  TORCH_CHECK(dy.is_contiguous() && gate.is_contiguous() &&
              up.is_contiguous() && dgate.is_contiguous() &&
              dup.is_contiguous());
  TORCH_CHECK(gate.numel() % 8 == 0);
  const long nvec = gate.numel() / 8;
  const long blocks = std::min<long>((nvec + 255) / 256,
                                     MAX_RESIDENT_BLOCKS);
  auto stream = at::hip::getCurrentHIPStream();
  silu_mul_bwd_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (bf16_t*)dgate.data_ptr(), (bf16_t*)dup.data_ptr(),
      (const bf16_t*)dy.data_ptr(), (const bf16_t*)gate.data_ptr(),
      (const bf16_t*)up.data_ptr(), nvec);
  HIP_CHECK_KERNEL();
}

void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.dtype() == torch::kBFloat16);
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(gate.dim() == 2 && up.dim() == 2);
  TORCH_CHECK(gate.stride(1) == 1 && up.stride(1) == 1,
              "inner dim must be contiguous");
  const long rows = gate.size(0);
  const long cols = gate.size(1);
  TORCH_CHECK(cols % 8 == 0);
  const long nvec = rows * (cols / 8);
  int blocks = (int)std::min<long>((nvec + 255) / 256, MAX_RESIDENT_BLOCKS);
  auto stream = at::hip::getCurrentHIPStream();
  silu_mul_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (bf16_t*)out.data_ptr(), (const bf16_t*)gate.data_ptr(),
      (const bf16_t*)up.data_ptr(), rows, (int)(cols / 8), gate.stride(0),
      up.stride(0));
  HIP_CHECK_KERNEL();
}

// --------------------------------------------------------------------- RoPE
// NEOX rotate-half, in place on q and k. cos/sin tables precomputed on host
// (guide App. B: never on-device trig), fp32, indexed by per-token position.
// q: (N, Hq, D) k: (N, Hk, D) contiguous; cos/sin: (max_pos, D/2).
// One block per token; threads cover (head, d) pairs; d pairs are
// (d, d + D/2). Loads are bf16x4 (8 B) on each half.
__global__ void rope_kernel(bf16_t* __restrict__ q, bf16_t* __restrict__ k,
                            const int* __restrict__ positions,
                            const float* __restrict__ cos_tab,
                            const float* __restrict__ sin_tab, int Hq, int Hk,
                            int D, long ldq, long ldk /* token strides */) {
  const int tok = blockIdx.x;
  const int pos = positions[tok];
  const int half = D / 2;
  const float* c = cos_tab + (long)pos * half;
  const float* s = sin_tab + (long)pos * half;
  const int H = Hq + Hk;
  // each thread handles 4 consecutive rotary pairs of one head
  const int quads_per_head = half / 4;
  for (int t = threadIdx.x; t < H * quads_per_head; t += blockDim.x) {
    const int h = t / quads_per_head;
    const int qd = (t % quads_per_head) * 4;
    bf16_t* base = (h < Hq) ? q + (long)tok * ldq + (long)h * D
                            : k + (long)tok * ldk + (long)(h - Hq) * D;
    bf16x4 x1 = *reinterpret_cast<bf16x4*>(base + qd);
    bf16x4 x2 = *reinterpret_cast<bf16x4*>(base + half + qd);
    float4 cv = *reinterpret_cast<const float4*>(c + qd);
    float4 sv = *reinterpret_cast<const float4*>(s + qd);
    const float cc[4] = {cv.x, cv.y, cv.z, cv.w};
    const float ss[4] = {sv.x, sv.y, sv.z, sv.w};
    bf16x4 o1, o2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = bf2f(x1.v[j]), b = bf2f(x2.v[j]);
      o1.v[j] = f2bf(a * cc[j] - b * ss[j]);
      o2.v[j] = f2bf(b * cc[j] + a * ss[j]);
    }
    *reinterpret_cast<bf16x4*>(base + qd) = o1;
    *reinterpret_cast<bf16x4*>(base + half + qd) = o2;
  }
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_tab, torch::Tensor sin_tab) {
  // q: (N, Hq, D), k: (N, Hk, D); token stride may exceed Hq*D (views into a
  // fused qkv projection), heads/dims must be contiguous.
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1);
  TORCH_CHECK(q.stride(1) == q.size(2) && k.stride(1) == k.size(2),
              "head dim must be packed");
  TORCH_CHECK(positions.dtype() == torch::kInt32);
  const int N = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hk = k.size(1);
  TORCH_CHECK((D / 2) % 4 == 0, "head_dim/2 must be a multiple of 4");
  auto stream = at::hip::getCurrentHIPStream();
  rope_kernel<<<dim3(N), dim3(256), 0, stream>>>(
      (bf16_t*)q.data_ptr(), (bf16_t*)k.data_ptr(),
      positions.data_ptr<int>(), cos_tab.data_ptr<float>(),
      sin_tab.data_ptr<float>(), Hq, Hk, D, q.stride(0), k.stride(0));
  HIP_CHECK_KERNEL();
}
