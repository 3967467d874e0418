// torch binding for the experimental GEMM (JIT-loaded by
// profiles/microbench_gemm.py; NOT part of setup.py).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef __hip_bfloat16 bf16;
extern "C" __global__ void gemm_bf16_nt(const bf16*, const bf16*, bf16*,
                                        int, int, int);
extern "C" __global__ void gemm_decode_bf16(const bf16*, const bf16*,
                                            float*, int, int);

torch::Tensor gemm_nt(torch::Tensor A, torch::Tensor W) {
  TORCH_CHECK(A.is_cuda() && W.is_cuda());
  TORCH_CHECK(A.dtype() == torch::kBFloat16 && W.dtype() == torch::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && W.is_contiguous());
  const int M = A.size(0), K = A.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == K);
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "prototype needs M%256==N%256==K%64==0");
  auto C = torch::empty({M, N}, A.options());
  const int lds_bytes = 4 * 256 * 64 * 2;   // 128 KiB (2 bufs x A+W)
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute((const void*)gemm_bf16_nt,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              lds_bytes);
    attr_set = true;
  }
  const int grid = (M / 256) * (N / 256);
  hipLaunchKernelGGL(gemm_bf16_nt, dim3(grid), dim3(512), lds_bytes,
                     c10::hip::getCurrentHIPStream().stream(),
                     (const bf16*)A.data_ptr(), (const bf16*)W.data_ptr(),
                     (bf16*)C.data_ptr(), M, N, K);
  return C;
}

torch::Tensor gemm_decode(torch::Tensor x, torch::Tensor W, int ksplit) {
  TORCH_CHECK(x.is_cuda() && W.is_cuda());
  TORCH_CHECK(x.dtype() == torch::kBFloat16 && W.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && W.is_contiguous());
  const int M = x.size(0), K = x.size(1), N = W.size(0);
  TORCH_CHECK(M == 256 && W.size(1) == K);
  TORCH_CHECK(N % 64 == 0 && K % (64 * ksplit) == 0);
  auto out = ksplit == 1
      ? torch::empty({M, N}, x.options().dtype(torch::kFloat32))
      : torch::zeros({M, N}, x.options().dtype(torch::kFloat32));
  const int lds_bytes = 2 * (256 * 64 + 64 * 64) * 2;   // 80 KiB
  static bool attr2 = false;
  if (!attr2) {
    (void)hipFuncSetAttribute((const void*)gemm_decode_bf16,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              lds_bytes);
    attr2 = true;
  }
  dim3 grid(N / 64, ksplit);
  hipLaunchKernelGGL(gemm_decode_bf16, grid, dim3(512), lds_bytes,
                     c10::hip::getCurrentHIPStream().stream(),
                     (const bf16*)x.data_ptr(), (const bf16*)W.data_ptr(),
                     out.data_ptr<float>(), N, K);
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_nt", &gemm_nt);
  m.def("gemm_decode", &gemm_decode);
}
