// Fused decode-time distraction rerank (north star: "the distraction
// terms — KL divergence over the attention-weight history and cosine
// distance over the context/state history — ... are fused HIP
// reductions"). One kernel computes, for every live hypothesis, the
// three penalties of nats.py:981-999 in a single pass over its history:
//   -kl * min_t KL(alpha_t || alpha_cur)        (scipy entropy conv.)
//   +cf * max_t (1 - cos(ctx_t, ctx_cur))
//   +sf * max_t (1 - cos(state_t, state_cur))
// grid (live_k); each wave owns one history step; block-level min/max
// combine in LDS.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cfloat>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NW = BLOCK / NATS_WAVE;

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int o = NATS_WAVE / 2; o > 0; o >>= 1) v += __shfl_down(v, o);
  return v;
}

__global__ __launch_bounds__(BLOCK) void rerank_penalties_kernel(
    const float* __restrict__ hist_a,  // [n][k][Ts]
    const float* __restrict__ hist_c,  // [n][k][C]
    const float* __restrict__ hist_s,  // [n][k][H]
    const float* __restrict__ cur_a,   // [k][Ts]
    const float* __restrict__ cur_c,   // [k][C]
    const float* __restrict__ cur_s,   // [k][H]
    float* __restrict__ pen,           // [k]
    float kl_f, float ctx_f, float state_f, int n, int k, int Ts, int C,
    int H) {
  __shared__ float red_kl[NW], red_cd[NW], red_sd[NW];
  const int hyp = blockIdx.x;
  const int wave = threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);

  float kl_min = FLT_MAX, cd_max = -FLT_MAX, sd_max = -FLT_MAX;
  for (int t = wave; t < n; t += NW) {
    // KL(p_t || q): scipy normalises both distributions first
    const float* p = hist_a + ((long)t * k + hyp) * Ts;
    const float* q = cur_a + (long)hyp * Ts;
    float psum = 0.f, qsum = 0.f;
    for (int s = lane; s < Ts; s += NATS_WAVE) {
      psum += p[s];
      qsum += q[s];
    }
    psum = wave_sum(psum);
    qsum = wave_sum(qsum);
    psum = __shfl(psum, 0);
    qsum = __shfl(qsum, 0);
    float kl = 0.f;
    for (int s = lane; s < Ts; s += NATS_WAVE) {
      const float pv = p[s] / psum;
      const float qv = q[s] / qsum;
      if (pv > 0.f) kl += pv * (__logf(pv) - __logf(qv));
    }
    kl = wave_sum(kl);

    const float* hc = hist_c + ((long)t * k + hyp) * C;
    const float* cc = cur_c + (long)hyp * C;
    float dot_c = 0.f, n1 = 0.f, n2 = 0.f;
    for (int s = lane; s < C; s += NATS_WAVE) {
      dot_c += hc[s] * cc[s];
      n1 += hc[s] * hc[s];
      n2 += cc[s] * cc[s];
    }
    dot_c = wave_sum(dot_c);
    n1 = wave_sum(n1);
    n2 = wave_sum(n2);
    const float den_c = sqrtf(n1) * sqrtf(n2);
    const float cd = (den_c == 0.f) ? 0.f : (1.f - dot_c / den_c);

    const float* hs = hist_s + ((long)t * k + hyp) * H;
    const float* cs = cur_s + (long)hyp * H;
    float dot_s = 0.f, m1 = 0.f, m2 = 0.f;
    for (int s = lane; s < H; s += NATS_WAVE) {
      dot_s += hs[s] * cs[s];
      m1 += hs[s] * hs[s];
      m2 += cs[s] * cs[s];
    }
    dot_s = wave_sum(dot_s);
    m1 = wave_sum(m1);
    m2 = wave_sum(m2);
    const float den_s = sqrtf(m1) * sqrtf(m2);
    const float sd = (den_s == 0.f) ? 0.f : (1.f - dot_s / den_s);

    if (lane == 0) {
      kl_min = fminf(kl_min, kl);
      cd_max = fmaxf(cd_max, cd);
      sd_max = fmaxf(sd_max, sd);
    }
  }
  if (lane == 0) {
    red_kl[wave] = kl_min;
    red_cd[wave] = cd_max;
    red_sd[wave] = sd_max;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a = FLT_MAX, b = -FLT_MAX, c = -FLT_MAX;
    for (int w = 0; w < NW; ++w) {
      a = fminf(a, red_kl[w]);
      b = fmaxf(b, red_cd[w]);
      c = fmaxf(c, red_sd[w]);
    }
    pen[hyp] = (n > 0) ? (-kl_f * a + ctx_f * b + state_f * c) : 0.f;
  }
}

}  // namespace

torch::Tensor rerank_penalties(torch::Tensor hist_a, torch::Tensor hist_c,
                               torch::Tensor hist_s, torch::Tensor cur_a,
                               torch::Tensor cur_c, torch::Tensor cur_s,
                               double kl_f, double ctx_f, double state_f) {
  TORCH_CHECK(hist_a.is_cuda() && hist_a.dtype() == torch::kFloat32);
  const int n = hist_a.size(0), k = hist_a.size(1), Ts = hist_a.size(2);
  const int C = hist_c.size(2), H = hist_s.size(2);
  auto pen = torch::empty({k}, hist_a.options());
  hipLaunchKernelGGL(rerank_penalties_kernel, dim3(k), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream().stream(),
                     hist_a.contiguous().data_ptr<float>(),
                     hist_c.contiguous().data_ptr<float>(),
                     hist_s.contiguous().data_ptr<float>(),
                     cur_a.contiguous().data_ptr<float>(),
                     cur_c.contiguous().data_ptr<float>(),
                     cur_s.contiguous().data_ptr<float>(),
                     pen.data_ptr<float>(), (float)kl_f, (float)ctx_f,
                     (float)state_f, n, k, Ts, C, H);
  HIP_CHECK(hipGetLastError());
  return pen;
}
