// MFMA fragment-layout self-test: a single-tile 16x16x32-per-step GEMM
// C(MxN) = A(MxK) @ B(KxN) with B supplied TRANSPOSED (NxK row-major).
// Used by tests/test_gpu_kernels.py against torch.matmul on asymmetric
// inputs (transpose-detecting, guide §5.4 rule 16).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

__global__ void mfma_gemm_bt_kernel(const bf16_t* __restrict__ A,
                                    const bf16_t* __restrict__ Bt,
                                    float* __restrict__ C, int M, int N,
                                    int K) {
  // grid: (M/16, N/16) tiles, one wave per tile
  const int m0 = blockIdx.x * 16;
  const int n0 = blockIdx.y * 16;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k = 0; k < K; k += 32) {
    bf16x8 a = frag_a_rowmajor(A, m0, K, k);
    bf16x8 b = frag_bt_rowmajor(Bt, n0, K, k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  store_cd_rowmajor(C, acc, m0, N, n0);
}

torch::Tensor mfma_gemm_bt(torch::Tensor A, torch::Tensor Bt) {
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda());
  TORCH_CHECK(A.dtype() == torch::kBFloat16 && Bt.dtype() == torch::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && Bt.is_contiguous());
  int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K && M % 16 == 0 && N % 16 == 0 && K % 32 == 0);
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  dim3 grid(M / 16, N / 16);
  hipLaunchKernelGGL(mfma_gemm_bt_kernel, grid, dim3(64), 0,
                     at::cuda::getCurrentCUDAStream().stream(),
                     (const bf16_t*)A.data_ptr(), (const bf16_t*)Bt.data_ptr(),
                     C.data_ptr<float>(), (int)M, (int)N, (int)K);
  HIP_CHECK(hipGetLastError());
  return C;
}

