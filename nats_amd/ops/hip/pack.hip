// Fused weight-packing kernels.
//
// Every training step re-packs the recurrent weights into the MFMA
// operand layouts (transposed, group-interleaved, zero-padded bf16 —
// see ops/gru.py / ops/cond_gru.py). The torch expression of each pack
// is a 4-8 kernel chain (pad, cat/stack, permute-contiguous, cast);
// these kernels produce each packed buffer in ONE pass reading the fp32
// parameters directly.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

constexpr int JB = 16;

// forward scan operand: [ngrp*3*16][Hpad]; group g of row-block grp holds
// output columns j=grp*16+c of (g0: U[:, :H], g1: U[:, H:], g2: Ux)
// transposed; zero outside.
__global__ void pack_fwd_kernel(const float* __restrict__ U,   // [H][2H]
                                const float* __restrict__ Ux,  // [H][H]
                                bf16_t* __restrict__ out, int H, int Hpad,
                                int nrows) {
  const int row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                  threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row >= nrows) return;
  const int grp = row / (3 * JB);
  const int g = (row / JB) % 3;
  const int j = grp * JB + row % JB;
  bf16_t* orow = out + (long)row * Hpad;
  for (int k = lane; k < Hpad; k += NATS_WAVE) {
    float v = 0.f;
    if (j < H && k < H) {
      v = (g == 0) ? U[(long)k * 2 * H + j]
          : (g == 1) ? U[(long)k * 2 * H + H + j]
                     : Ux[(long)k * H + j];
    }
    orow[k] = (bf16_t)v;
  }
}

// [rows0][wA + wB] concat of A|B, zero-padded to [R][K] (backward
// operand [U | Ux] and the decoder's [W_1 | Wx_1])
__global__ void pack_cat2_kernel(const float* __restrict__ A,
                                 const float* __restrict__ B,
                                 bf16_t* __restrict__ out, int rows0, int wA,
                                 int wB, int R, int K) {
  const int row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                  threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row >= R) return;
  bf16_t* orow = out + (long)row * K;
  for (int k = lane; k < K; k += NATS_WAVE) {
    float v = 0.f;
    if (row < rows0) {
      if (k < wA) v = A[(long)row * wA + k];
      else if (k < wA + wB) v = B[(long)row * wB + (k - wA)];
    }
    orow[k] = (bf16_t)v;
  }
}

// zero-padded (optionally transposed) single matrix: out[r][k] =
// src[r][k] (or src[k][r]) within (rows0, cols0)
__global__ void pack_pad_kernel(const float* __restrict__ src,
                                bf16_t* __restrict__ out, int rows0,
                                int cols0, int R, int K, int transpose) {
  const int row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                  threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row >= R) return;
  bf16_t* orow = out + (long)row * K;
  for (int k = lane; k < K; k += NATS_WAVE) {
    float v = 0.f;
    if (row < rows0 && k < cols0)
      v = transpose ? src[(long)k * rows0 + row] : src[(long)row * cols0 + k];
    orow[k] = (bf16_t)v;
  }
}

// GRU_1 4-group operand: [ngrp*4*16][K1=Hpad+Cpad]; groups r2/u2 read
// (U_1 | W_1), pxa reads (Ux_1 | 0), pxb reads (0 | Wx_1)
// (ops/cond_gru.py pack_gru1_weights)
__global__ void pack_gru1_kernel(const float* __restrict__ U_1,  // [H][2H]
                                 const float* __restrict__ W_1,  // [C][2H]
                                 const float* __restrict__ Ux_1, // [H][H]
                                 const float* __restrict__ Wx_1, // [C][H]
                                 bf16_t* __restrict__ out, int H, int C,
                                 int Hpad, int Cpad, int nrows) {
  const int row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                  threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row >= nrows) return;
  const int grp = row / (4 * JB);
  const int g = (row / JB) % 4;
  const int j = grp * JB + row % JB;
  const int K1 = Hpad + Cpad;
  bf16_t* orow = out + (long)row * K1;
  for (int k = lane; k < K1; k += NATS_WAVE) {
    float v = 0.f;
    if (j < H) {
      if (k < Hpad) {
        if (k < H) {
          if (g == 0) v = U_1[(long)k * 2 * H + j];
          else if (g == 1) v = U_1[(long)k * 2 * H + H + j];
          else if (g == 2) v = Ux_1[(long)k * H + j];
        }
      } else {
        const int c = k - Hpad;
        if (c < C) {
          if (g == 0) v = W_1[(long)c * 2 * H + j];
          else if (g == 1) v = W_1[(long)c * 2 * H + H + j];
          else if (g == 3) v = Wx_1[(long)c * H + j];
        }
      }
    }
    orow[k] = (bf16_t)v;
  }
}

inline int cdiv_p(int a, int b) { return (a + b - 1) / b; }

}  // namespace

torch::Tensor pack_fwd_weights_hip(torch::Tensor U, torch::Tensor Ux) {
  TORCH_CHECK(U.is_cuda() && U.dtype() == torch::kFloat32);
  auto Uc = U.contiguous();
  auto Uxc = Ux.contiguous();
  const int H = Ux.size(1);
  const int ngrp = cdiv_p(H, JB);
  const int Hpad = cdiv_p(H, 32) * 32;
  const int nrows = ngrp * 3 * JB;
  auto out = torch::empty({nrows, Hpad},
                          U.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(pack_fwd_kernel, dim3(cdiv_p(nrows, 4)), dim3(256), 0,
                     stream, Uc.data_ptr<float>(), Uxc.data_ptr<float>(),
                     (bf16_t*)out.data_ptr(), H, Hpad, nrows);
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor pack_cat2_hip(torch::Tensor A, torch::Tensor B, long R,
                            long K) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  const int rows0 = A.size(0);
  auto out = torch::empty({R, K}, A.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(pack_cat2_kernel, dim3(cdiv_p((int)R, 4)), dim3(256), 0,
                     stream, Ac.data_ptr<float>(), Bc.data_ptr<float>(),
                     (bf16_t*)out.data_ptr(), rows0, (int)A.size(1),
                     (int)B.size(1), (int)R, (int)K);
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor pack_pad_hip(torch::Tensor src, long R, long K,
                           bool transpose) {
  TORCH_CHECK(src.is_cuda() && src.dtype() == torch::kFloat32);
  auto Sc = src.contiguous();
  const int rows0 = transpose ? src.size(1) : src.size(0);
  const int cols0 = transpose ? src.size(0) : src.size(1);
  auto out = torch::empty({R, K}, src.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(pack_pad_kernel, dim3(cdiv_p((int)R, 4)), dim3(256), 0,
                     stream, Sc.data_ptr<float>(), (bf16_t*)out.data_ptr(),
                     rows0, cols0, (int)R, (int)K, transpose ? 1 : 0);
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor pack_gru1_weights_hip(torch::Tensor U_1, torch::Tensor W_1,
                                    torch::Tensor Ux_1, torch::Tensor Wx_1,
                                    long Hpad, long Cpad) {
  TORCH_CHECK(U_1.is_cuda() && U_1.dtype() == torch::kFloat32);
  auto a = U_1.contiguous();
  auto b = W_1.contiguous();
  auto c = Ux_1.contiguous();
  auto d = Wx_1.contiguous();
  const int H = Ux_1.size(0);
  const int C = W_1.size(0);
  const int ngrp = cdiv_p(H, JB);
  const int nrows = ngrp * 4 * JB;
  auto out = torch::empty({nrows, Hpad + Cpad},
                          U_1.options().dtype(torch::kBFloat16));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(pack_gru1_kernel, dim3(cdiv_p(nrows, 4)), dim3(256), 0,
                     stream, a.data_ptr<float>(), b.data_ptr<float>(),
                     c.data_ptr<float>(), d.data_ptr<float>(),
                     (bf16_t*)out.data_ptr(), H, C, (int)Hpad, (int)Cpad,
                     nrows);
  HIP_CHECK(hipGetLastError());
  return out;
}
