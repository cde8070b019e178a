// Fused multi-tensor optimizer kernels:
// global-norm gradient clip + the reference's adadelta update
// (nats.py:1145-1173, rho/eps preserved) in two launches over a chunk
// table — replaces ~40 per-tensor op sequences per step.
//
// Chunk table: the python wrapper flattens the parameter list into
// fixed-size chunks; per chunk a (tensor_idx, elem_offset) pair plus
// per-tensor base pointers (p, g, rg2, ru2 — all fp32).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

namespace {

constexpr int CHUNK = 1 << 16;  // elements per chunk
constexpr int BLOCK = 256;

__global__ void fused_gradnorm_kernel(const float* const* __restrict__ gs,
                                      const long* __restrict__ sizes,
                                      const int* __restrict__ chunk_tensor,
                                      const long* __restrict__ chunk_off,
                                      float* __restrict__ g2_out) {
  __shared__ float red[BLOCK / NATS_WAVE];
  const int ci = blockIdx.x;
  const int ti = chunk_tensor[ci];
  const long off = chunk_off[ci];
  const float* g = gs[ti] + off;
  const long rem = sizes[ti] - off;
  const long n = rem < (long)CHUNK ? rem : (long)CHUNK;
  float acc = 0.f;
  const long n4 = n & ~(long)3;
  for (long i = threadIdx.x * 4; i < n4; i += (long)BLOCK * 4) {
    const float4 v = *(const float4*)(g + i);
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  for (long i = n4 + threadIdx.x; i < n; i += BLOCK) acc += g[i] * g[i];
#pragma unroll
  for (int o = NATS_WAVE / 2; o > 0; o >>= 1) acc += __shfl_down(acc, o);
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0)
    red[threadIdx.x / NATS_WAVE] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < BLOCK / NATS_WAVE; ++w) s += red[w];
    atomicAdd(g2_out, s);
  }
}

// clip scale computed on-device (no host sync): scale = clip_c/sqrt(g2)
// if g2 > clip_c^2 else 1 (nats.py:1344-1356); then the adadelta update.
__global__ void fused_adadelta_kernel(
    float* const* __restrict__ ps, float* const* __restrict__ gs,
    float* const* __restrict__ rg2s, float* const* __restrict__ ru2s,
    const long* __restrict__ sizes, const int* __restrict__ chunk_tensor,
    const long* __restrict__ chunk_off, const float* __restrict__ g2_in,
    float clip_c, float rho, float eps) {
  const int ci = blockIdx.x;
  const int ti = chunk_tensor[ci];
  const long off = chunk_off[ci];
  float* p = ps[ti] + off;
  float* g = gs[ti] + off;
  float* rg2 = rg2s[ti] + off;
  float* ru2 = ru2s[ti] + off;
  const long rem = sizes[ti] - off;
  const long n = rem < (long)CHUNK ? rem : (long)CHUNK;
  float scale = 1.f;
  if (clip_c > 0.f) {
    const float g2 = *g2_in;
    if (g2 > clip_c * clip_c) scale = clip_c / sqrtf(g2);
  }
  // float4 accesses: the scalar version measured 436 us/step for the
  // 27.5M-param model (~2 TB/s) — 8 scalar 4B streams per element left
  // too few lines in flight; 16B vectors recover the roofline
  const long n4 = n & ~(long)3;
  for (long i = (long)threadIdx.x * 4; i < n4; i += (long)BLOCK * 4) {
    float4 gv4 = *(float4*)(g + i);
    float4 r24 = *(float4*)(rg2 + i);
    float4 ru4 = *(float4*)(ru2 + i);
    float4 pv4 = *(float4*)(p + i);
    float* gv = &gv4.x;
    float* r2 = &r24.x;
    float* ru = &ru4.x;
    float* pv = &pv4.x;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      const float gc = gv[k] * scale;
      const float r2n = rho * r2[k] + (1.f - rho) * gc * gc;
      const float ud = -sqrtf(ru[k] + eps) / sqrtf(r2n + eps) * gc;
      ru[k] = rho * ru[k] + (1.f - rho) * ud * ud;
      r2[k] = r2n;
      pv[k] += ud;
      gv[k] = gc;  // leave the clipped gradient visible (parity w/ ref)
    }
    *(float4*)(g + i) = gv4;
    *(float4*)(rg2 + i) = r24;
    *(float4*)(ru2 + i) = ru4;
    *(float4*)(p + i) = pv4;
  }
  for (long i = n4 + threadIdx.x; i < n; i += BLOCK) {
    const float gv = g[i] * scale;
    const float r2 = rho * rg2[i] + (1.f - rho) * gv * gv;
    rg2[i] = r2;
    const float ud = -sqrtf(ru2[i] + eps) / sqrtf(r2 + eps) * gv;
    ru2[i] = rho * ru2[i] + (1.f - rho) * ud * ud;
    p[i] += ud;
    g[i] = gv;
  }
}

}  // namespace

// tensors: [p..., g..., rg2..., ru2...] + device-side tables built by the
// wrapper: ptrs (4,n_tensors) int64, sizes (n_tensors) int64,
// chunk_tensor (n_chunks) int32, chunk_off (n_chunks) int64,
// g2 (1) fp32 zeroed.
void fused_adadelta_step(torch::Tensor ptrs, torch::Tensor sizes,
                         torch::Tensor chunk_tensor, torch::Tensor chunk_off,
                         torch::Tensor g2, double clip_c, double rho,
                         double eps) {
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  const int n_chunks = chunk_tensor.size(0);
  const int nt = sizes.size(0);
  auto pp = (float* const*)ptrs.data_ptr<int64_t>();
  hipLaunchKernelGGL(fused_gradnorm_kernel, dim3(n_chunks), dim3(BLOCK), 0,
                     stream, (const float* const*)(pp + nt),
                     sizes.data_ptr<int64_t>(),
                     chunk_tensor.data_ptr<int>(),
                     chunk_off.data_ptr<int64_t>(), g2.data_ptr<float>());
  hipLaunchKernelGGL(fused_adadelta_kernel, dim3(n_chunks), dim3(BLOCK), 0,
                     stream, pp, (float* const*)(pp + nt),
                     (float* const*)(pp + 2 * nt),
                     (float* const*)(pp + 3 * nt), sizes.data_ptr<int64_t>(),
                     chunk_tensor.data_ptr<int>(),
                     chunk_off.data_ptr<int64_t>(), g2.data_ptr<float>(),
                     (float)clip_c, (float)rho, (float)eps);
  HIP_CHECK(hipGetLastError());
}
