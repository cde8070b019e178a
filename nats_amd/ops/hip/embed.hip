// Embedding gather / scatter-add (reference op sites
// nats.py:700-701 `Wemb[x.flatten()].reshape(...)` and 730-734 target
// embedding + shift-right-with-zero-BOS row).
//
// The decoder's shift is FUSED into the gather (shift_rows = B writes the
// first B output rows as zeros and reads ids offset by one timestep) —
// the torch path spent two extra kernels (zeros_like + slice copy) per
// forward on it. Backward is a row-parallel atomic scatter-add into
// dWemb. These ops are ~0.1% of the CNN/DM step (profiles/
// cnn_kernel_stats_final3.csv: indexing_backward 16 us/step) — built for
// inventory completeness and launch-count reduction, not throughput.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

// one wave per output row; E-strided float4-ish copy (E is small: 100)
__global__ void embed_gather_kernel(const float* __restrict__ Wemb,
                                    const long* __restrict__ ids,
                                    float* __restrict__ out, long N, int E,
                                    int shift_rows, long V) {
  const long row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                   threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row >= N) return;
  float* orow = out + row * E;
  if (row < shift_rows) {
    for (int e = lane; e < E; e += NATS_WAVE) orow[e] = 0.f;
    return;
  }
  long id = ids[row - shift_rows];
  if (id < 0) id = 0;  // BOS sentinel (-1) in the one-step path
  const float* src = Wemb + (long)id * E;
  for (int e = lane; e < E; e += NATS_WAVE) orow[e] = src[e];
}

__global__ void embed_scatter_kernel(const float* __restrict__ dout,
                                     const long* __restrict__ ids,
                                     float* __restrict__ dW, long N, int E,
                                     int shift_rows) {
  const long row = blockIdx.x * (blockDim.x / NATS_WAVE) +
                   threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  if (row < shift_rows || row >= N) return;  // zero-BOS rows carry no grad
  long id = ids[row - shift_rows];
  if (id < 0) id = 0;
  const float* drow = dout + row * E;
  float* wrow = dW + (long)id * E;
  for (int e = lane; e < E; e += NATS_WAVE) atomicAdd(wrow + e, drow[e]);
}

}  // namespace

torch::Tensor embed_gather(torch::Tensor Wemb, torch::Tensor ids,
                           long shift_rows) {
  TORCH_CHECK(Wemb.is_cuda() && Wemb.dtype() == torch::kFloat32 &&
              Wemb.is_contiguous());
  TORCH_CHECK(ids.dtype() == torch::kInt64 && ids.is_contiguous());
  const int E = Wemb.size(1);
  const long n_ids = ids.numel();
  // with a shift the LAST timestep's ids are not consumed (reference
  // shift drops emb[-1]): output rows = n_ids (shift replaces, not adds)
  const long N = n_ids;
  auto out_sizes = ids.sizes().vec();
  out_sizes.push_back(E);
  auto out = torch::empty(out_sizes, Wemb.options());
  const int waves_per_block = 4;
  const long nblocks = (N + waves_per_block - 1) / waves_per_block;
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(embed_gather_kernel, dim3((unsigned)nblocks),
                     dim3(waves_per_block * NATS_WAVE), 0, stream,
                     Wemb.data_ptr<float>(), ids.data_ptr<long>(),
                     out.data_ptr<float>(), N, E, (int)shift_rows,
                     Wemb.size(0));
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor embed_scatter_add(torch::Tensor dout, torch::Tensor ids,
                                long V, long shift_rows) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous());
  auto dout_f = dout.to(torch::kFloat32);
  const int E = dout.size(-1);
  const long N = dout.numel() / E;
  auto dW = torch::zeros({V, E}, dout_f.options());
  const int waves_per_block = 4;
  const long nblocks = (N + waves_per_block - 1) / waves_per_block;
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  hipLaunchKernelGGL(embed_scatter_kernel, dim3((unsigned)nblocks),
                     dim3(waves_per_block * NATS_WAVE), 0, stream,
                     dout_f.data_ptr<float>(), ids.data_ptr<long>(),
                     dW.data_ptr<float>(), N, E, (int)shift_rows);
  HIP_CHECK(hipGetLastError());
  return dW;
}
