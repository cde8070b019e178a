#include "hip/hip_runtime.h"
// Fused large-vocabulary softmax + cross-entropy, forward and backward
// (kernel rows K18-K19 in SURVEY §2.4; semantics = softmax +
// categorical_crossentropy + mask-reduce, nats.py:763-770 — the masking
// stays in python, this op returns per-position NLL).
//
// Forward: one workgroup per row; a single online pass computes the row
// max and sum(exp) (never materialising the probability matrix), then
// nll = logsumexp - (logit[target] - max). Saves (max, log_sum_exp) for
// backward. bf16 logits are read vectorised (bf16x8, G13).
//
// Backward: dlogit[n, v] = (exp(logit - max) / sum - onehot[target]) *
// dnll[n], recomputed from the saved row stats in one streaming pass.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <int BLOCK>
__global__ void softmax_ce_fwd_kernel(const bf16_t* __restrict__ logits,
                                      const int64_t* __restrict__ targets,
                                      float* __restrict__ nll,
                                      float* __restrict__ stats,  // [N][2]
                                      int64_t N, int64_t V) {
  const int64_t n = blockIdx.x;
  if (n >= N) return;
  const bf16_t* row = logits + n * V;

  // online max + sum(exp(x - max)) per thread, then block reduce
  float m = -INFINITY, s = 0.f;
  const int64_t V8 = V & ~(int64_t)7;
  for (int64_t v = threadIdx.x * 8; v < V8; v += (int64_t)BLOCK * 8) {
    bf16x8 x8 = *(const bf16x8*)(row + v);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float x = (float)x8[i];
      if (x > m) {
        s *= __expf(m - x);
        m = x;
      }
      s += __expf(x - m);
    }
  }
  for (int64_t v = V8 + threadIdx.x; v < V; v += BLOCK) {
    const float x = (float)row[v];
    if (x > m) {
      s *= __expf(m - x);
      m = x;
    }
    s += __expf(x - m);
  }

  // wave reduce (max, sum) then LDS reduce across waves.
  // Guards: a thread that saw no elements has (m=-inf, s=0); merging two
  // such pairs must not evaluate exp(-inf - -inf) = NaN.
  __shared__ float sm[BLOCK / NATS_WAVE], ss[BLOCK / NATS_WAVE];
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1) {
    const float om = __shfl_down(m, off);
    const float os = __shfl_down(s, off);
    if (om > m) {
      s = (s == 0.f) ? os : (s * __expf(m - om) + os);
      m = om;
    } else if (os != 0.f) {
      s += os * __expf(om - m);
    }
  }
  const int wave = threadIdx.x / NATS_WAVE;
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0) {
    sm[wave] = m;
    ss[wave] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = sm[0], S = ss[0];
#pragma unroll
    for (int w = 1; w < BLOCK / NATS_WAVE; ++w) {
      if (sm[w] > M) {
        S = (S == 0.f) ? ss[w] : (S * __expf(M - sm[w]) + ss[w]);
        M = sm[w];
      } else if (ss[w] != 0.f) {
        S += ss[w] * __expf(sm[w] - M);
      }
    }
    const float lse = M + __logf(S);
    nll[n] = lse - (float)row[targets[n]];
    stats[n * 2] = M;
    stats[n * 2 + 1] = lse;
  }
}

template <int BLOCK>
__global__ void softmax_ce_bwd_kernel(const bf16_t* __restrict__ logits,
                                      const int64_t* __restrict__ targets,
                                      const float* __restrict__ stats,
                                      const float* __restrict__ dnll,
                                      bf16_t* __restrict__ dlogits, int64_t N,
                                      int64_t V) {
  const int64_t n = blockIdx.x;
  if (n >= N) return;
  const bf16_t* row = logits + n * V;
  bf16_t* drow = dlogits + n * V;
  const float lse = stats[n * 2 + 1];
  const float g = dnll[n];
  const int64_t tgt = targets[n];
  const int64_t V8 = V & ~(int64_t)7;
  for (int64_t v = threadIdx.x * 8; v < V8; v += (int64_t)BLOCK * 8) {
    bf16x8 x8 = *(const bf16x8*)(row + v);
    bf16x8 d8;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float p = __expf((float)x8[i] - lse);
      if (v + i == tgt) p -= 1.f;
      d8[i] = (bf16_t)(p * g);
    }
    *(bf16x8*)(drow + v) = d8;
  }
  for (int64_t v = V8 + threadIdx.x; v < V; v += BLOCK) {
    float p = __expf((float)row[v] - lse);
    if (v == tgt) p -= 1.f;
    drow[v] = (bf16_t)(p * g);
  }
}

}  // namespace

std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor targets) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(logits.dtype() == torch::kBFloat16);
  TORCH_CHECK(targets.dtype() == torch::kInt64 && targets.is_contiguous());
  const int64_t N = logits.size(0), V = logits.size(1);
  auto nll = torch::empty({N}, logits.options().dtype(torch::kFloat32));
  auto stats = torch::empty({N, 2}, logits.options().dtype(torch::kFloat32));
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL((softmax_ce_fwd_kernel<BLOCK>), dim3(N), dim3(BLOCK), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream(),
                     (const bf16_t*)logits.data_ptr(),
                     targets.data_ptr<int64_t>(), nll.data_ptr<float>(),
                     stats.data_ptr<float>(), N, V);
  HIP_CHECK(hipGetLastError());
  return {nll, stats};
}

torch::Tensor softmax_ce_bwd(torch::Tensor logits, torch::Tensor targets,
                             torch::Tensor stats, torch::Tensor dnll) {
  const int64_t N = logits.size(0), V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL((softmax_ce_bwd_kernel<BLOCK>), dim3(N), dim3(BLOCK), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream(),
                     (const bf16_t*)logits.data_ptr(),
                     targets.data_ptr<int64_t>(), stats.data_ptr<float>(),
                     dnll.contiguous().data_ptr<float>(),
                     (bf16_t*)dlogits.data_ptr(), N, V);
  HIP_CHECK(hipGetLastError());
  return dlogits;
}
