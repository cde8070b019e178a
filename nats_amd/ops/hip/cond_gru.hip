// Fused conditional-GRU decoder (training scan + one-step) for gfx950.
//
// Implements the reference's gru_cond_layer step (nats.py:498-572; kernel
// rows K10-K16 in SURVEY §2.4) as five fused kernels per timestep driven
// by a C++ time loop (no python in the scan):
//   1. GRU_2          — reuses nats_gru_step_fwd (identical cell math)
//   2. pstate GEMM    — h1 @ W_att (small MFMA GEMM)
//   3. attention      — e-scores + distraction-over-weights + masked
//                       softmax + acc_alpha update, one WG per batch row
//   4. context        — weighted sum over source + distraction gate over
//                       content vectors + acc_ctx update
//   5. GRU_1          — 4-output-group MFMA ([h1|ctx] packed operand) +
//                       gates/mask pointwise
// The backward pass is the exact reverse chain with the running-sum
// accumulators' (acc_ctx/acc_alpha) gradients threaded backwards
// (SURVEY §7 "hard parts" (ii)); weight gradients that factor over time
// (dU_1, dW_1, dW_att, dWc_att, dU_con, ...) are computed as single
// time-batched GEMMs in python from the per-step buffers saved here.
//
// Occupancy design: at dim 1000 an output-tile grid is only ngrpH=63
// workgroups (a quarter of the 256 CUs), so every per-step GEMM splits
// its K dimension across extra grid dimensions and the elementwise
// reductions chunk their serial axis, with fp32 partials summed by the
// NEXT kernel in the chain rather than an extra pass (gru1/gru2 split-K
// -> pointwise combine; pstate split-K -> escore/softmax inline sums;
// escore/scatter A-chunks -> e_buf/daccA atomics; dh_carry split-K ->
// next step's GRU_1 pointwise sums the halves). Scratch that is
// atomically accumulated each step (e_buf, ctxpre_f32) is re-zeroed by
// its consumer in the same pass, so the steady-state loop launches no
// memsets.
//
// Round-2 fusions (each GPU-measured; negatives reverted in history):
// the dctx passthrough + dot-buffer zeroing merged into the GRU_1
// backward pointwise; the distraction-gate backward plus the
// dU_con/dW_con column reductions live in the dctx GEMM's epilogue;
// dh1 += dpstate @ W_att converts fp32 A-fragments in-register; the
// e-score kernel stages the combined pstate + attention vectors in LDS.
// Ladder and profiles in profiles/README.md.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>

#include "common.h"
#include "gru_kernels.h"

namespace {

constexpr int JB = 16;

// ---------------- small GEMM: pstate = h1 @ W_att ----------------
// A = [32][ldK] bf16 (zero-padded), Bt = [Npad][ldK] bf16. Split-K over
// grid.z into Cpart [KS][32][N] f32 partials (the unsplit version was 14
// waves on the whole chip, 11.3us of pure load latency): consumers
// (cond_attn_escore inline, cond_attn_softmax for the saved pstate_all)
// sum the KS partials.
__global__ void cond_small_gemm_bt(const bf16_t* __restrict__ A,
                                   const bf16_t* __restrict__ Bt,
                                   float* __restrict__ Cpart, int B, int N,
                                   int ldK, int Kpad) {
  const int m0 = blockIdx.x * 16;
  const int n0 = blockIdx.y * 16;
  const int KS = gridDim.z;
  const int kchunk = ((Kpad / KS + 31) / 32) * 32;
  const int kbeg = min(Kpad, (int)blockIdx.z * kchunk);
  const int kend = min(Kpad, kbeg + kchunk);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  // NB: Bt's row stride is Kpad (the packed weight width), NOT ldK.
  NATS_MFMA_KLOOP(acc, A, m0, ldK, Bt, n0, Kpad, kbeg, kend);
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  const int col = n0 + (lane & 15);
  const int rbase = m0 + (lane >> 4) * 4;
  if (col >= N) return;
  float* C = Cpart + (long)blockIdx.z * 32 * N;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = rbase + i;
    if (row < B) C[(long)row * N + col] = acc[i];
  }
}

// ---------------- attention e-scores (s-parallel) ----------------
// grid (B, ceil(Ts/256), ACH): thread = one source position s, grid.z
// chunks the A (attention-dim) loop; chunks atomicAdd into e_buf, which
// arrives ZEROED (cond_attn_softmax re-zeroes it after its last read, so
// no per-step memset). pstate arrives as [KS][32][A] split-K partials.
__global__ __launch_bounds__(256) void cond_attn_escore(
    const float* __restrict__ pctx,      // [Ts][B][A]
    const float* __restrict__ ps_part,   // [KS][32][A] pstate partials
    int PS_KS,
    const float* __restrict__ accA,      // [B][Ts] (pre-update)
    const float* __restrict__ Dwei,      // [A]
    const float* __restrict__ Uatt,      // [A]
    const float* __restrict__ catt_p,    // [1]
    float* __restrict__ e_buf,           // [Ts][B]
    int B, int Ts, int A) {
  // combined pstate + Dwei/Uatt staged in LDS once per block: the inner
  // i-loop otherwise issues PS_KS+2 extra L2 loads per element for every
  // one of this block's 256 s-positions
  extern __shared__ float sm_e[];  // [A] ps, [A] Dwei, [A] Uatt
  float* sm_ps = sm_e;
  float* sm_dw = sm_e + A;
  float* sm_ua = sm_e + 2 * A;
  const int b = blockIdx.x;
  for (int i = threadIdx.x; i < A; i += blockDim.x) {
    float ps = 0.f;
    for (int k = 0; k < PS_KS; ++k) ps += ps_part[((long)k * 32 + b) * A + i];
    sm_ps[i] = ps;
    sm_dw[i] = Dwei[i];
    sm_ua[i] = Uatt[i];
  }
  __syncthreads();
  const int s = blockIdx.y * blockDim.x + threadIdx.x;
  if (s >= Ts) return;
  const int ACH = gridDim.z;
  const int chunkA = ((A + ACH - 1) / ACH + 3) & ~3;
  const int ibeg = blockIdx.z * chunkA;
  const int iend = min(A, ibeg + chunkA);
  if (ibeg >= iend) return;
  const float accAu = accA[(long)b * Ts + s];
  const float* prow = pctx + ((long)s * B + b) * A;
  float e = (blockIdx.z == 0) ? catt_p[0] : 0.f;
  for (int i = ibeg; i < iend; ++i) {
    e += tanhf(prow[i] + sm_ps[i] + accAu * sm_dw[i]) * sm_ua[i];
  }
  if (ACH == 1) {
    e_buf[(long)s * B + b] += e;  // single chunk: still additive (zeroed)
  } else {
    atomicAdd(e_buf + (long)s * B + b, e);
  }
}

// ---------------- softmax finish + acc_alpha update (one WG per b) ----
__global__ __launch_bounds__(1024) void cond_attn_softmax(
    float* __restrict__ accA,            // [B][Ts] in/out
    float* __restrict__ accA_used_t,     // [B][Ts] out (pre-update copy)
    const float* __restrict__ ctx_mask,  // [Ts][B] or null
    const float* __restrict__ mask_t,    // [B] or null
    float* __restrict__ e_buf,           // [Ts][B]; re-zeroed for t+1
    float* __restrict__ alphas_t,        // [B][Ts] out
    const float* __restrict__ ps_part,   // [KS][32][A] pstate partials
    int PS_KS,
    float* __restrict__ pstate_t,        // [B][A] combined (for backward)
    int A,
    int B, int Ts) {
  { // combine the split-K pstate partials once (backward reads pstate_all)
    const int b0 = blockIdx.x;
    for (int i = threadIdx.x; i < A; i += blockDim.x) {
      float ps = 0.f;
      for (int k = 0; k < PS_KS; ++k)
        ps += ps_part[((long)k * 32 + b0) * A + i];
      pstate_t[(long)b0 * A + i] = ps;
    }
  }
  const int b = blockIdx.x;
  __shared__ float red[256 / NATS_WAVE];
  __shared__ float bcast;

  // masked positions must not influence the max: they are zeroed AFTER
  // exp, but a padded row's e shifting M changes rounding for the real
  // rows — decode results would then depend on how far the source was
  // padded (the graph-decode length bucketing pads to 64)
  float lmax = -INFINITY;
  for (int s = threadIdx.x; s < Ts; s += blockDim.x) {
    if (ctx_mask != nullptr && ctx_mask[(long)s * B + b] == 0.f) continue;
    lmax = fmaxf(lmax, e_buf[(long)s * B + b]);
  }
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1)
    lmax = fmaxf(lmax, __shfl_down(lmax, off));
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0)
    red[threadIdx.x / NATS_WAVE] = lmax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = red[0];
    for (int w = 1; w < (int)blockDim.x / NATS_WAVE; ++w)
      M = fmaxf(M, red[w]);
    bcast = M;
  }
  __syncthreads();
  // all-masked rows cannot occur (every sequence has >= 1 real token),
  // but guard the degenerate -inf max anyway
  const float M = isfinite(bcast) ? bcast : 0.f;

  float lsum = 0.f;
  for (int s = threadIdx.x; s < Ts; s += blockDim.x) {
    float a = __expf(e_buf[(long)s * B + b] - M);
    e_buf[(long)s * B + b] = 0.f;  // ready for the next step's atomics
    if (ctx_mask != nullptr) a *= ctx_mask[(long)s * B + b];
    alphas_t[(long)b * Ts + s] = a;  // unnormalised, fixed below
    lsum += a;
  }
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1)
    lsum += __shfl_down(lsum, off);
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0)
    red[threadIdx.x / NATS_WAVE] = lsum;
  __syncthreads();
  if (threadIdx.x == 0) {
    float S = 0.f;
    for (int w = 0; w < (int)blockDim.x / NATS_WAVE; ++w) S += red[w];
    bcast = S;
  }
  __syncthreads();
  const float inv = 1.f / bcast;

  const float mm = (mask_t != nullptr) ? mask_t[b] : 1.f;
  for (int s = threadIdx.x; s < Ts; s += blockDim.x) {
    const float al = alphas_t[(long)b * Ts + s] * inv;
    alphas_t[(long)b * Ts + s] = al;
    accA_used_t[(long)b * Ts + s] = accA[(long)b * Ts + s];
    accA[(long)b * Ts + s] += mm * al;
  }
}

// ---------------- weighted context: s-chunked partial sums ----------
// grid (B, ceil(C/256), SCH); f32 atomicAdd into ctxpre_f32 (zeroed per
// step). Unrolled by 4 over s for memory-level parallelism.
__global__ void cond_attn_ctx_partial(
    const bf16_t* __restrict__ ctx_bf,   // [Ts][B][C]
    const float* __restrict__ alphas_t,  // [B][Ts]
    float* __restrict__ ctxpre_f32,      // [B][C] (accumulated)
    int B, int Ts, int C, int SCH) {
  const int b = blockIdx.x;
  const int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int chunk = (Ts + SCH - 1) / SCH;
  const int sbeg = blockIdx.z * chunk;
  const int send = min(Ts, sbeg + chunk);
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int s = sbeg;
  for (; s + 4 <= send; s += 4) {
    s0 += (float)ctx_bf[((long)(s + 0) * B + b) * C + c] *
          alphas_t[(long)b * Ts + s + 0];
    s1 += (float)ctx_bf[((long)(s + 1) * B + b) * C + c] *
          alphas_t[(long)b * Ts + s + 1];
    s2 += (float)ctx_bf[((long)(s + 2) * B + b) * C + c] *
          alphas_t[(long)b * Ts + s + 2];
    s3 += (float)ctx_bf[((long)(s + 3) * B + b) * C + c] *
          alphas_t[(long)b * Ts + s + 3];
  }
  for (; s < send; ++s)
    s0 += (float)ctx_bf[((long)s * B + b) * C + c] *
          alphas_t[(long)b * Ts + s];
  atomicAdd(&ctxpre_f32[(long)b * C + c], s0 + s1 + s2 + s3);
}

// ---------------- distraction gate + acc_ctx update ----------------
__global__ void cond_attn_gate_fwd(
    float* __restrict__ ctxpre_f32,  // [B][C]; re-zeroed for t+1's atomics
    const float* __restrict__ Ucon, const float* __restrict__ Wcon,
    float* __restrict__ accC,            // [B][C] in/out
    bf16_t* __restrict__ accC_used_t,    // [B][C]
    bf16_t* __restrict__ ctxpre_t,       // [B][C] (saved pre-gate sum)
    float* __restrict__ ctxs_t,          // [B][C] (gated output)
    bf16_t* __restrict__ hc_bf,          // [32][K1] GRU_1 operand
    int ctx_off, int ldK1, const float* __restrict__ mask_t, int B, int C) {
  const int b = blockIdx.x;
  const int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float sum = ctxpre_f32[(long)b * C + c];
  ctxpre_f32[(long)b * C + c] = 0.f;  // next step's ctx_partial atomics
  ctxpre_t[(long)b * C + c] = (bf16_t)sum;
  const float accCv = accC[(long)b * C + c];
  accC_used_t[(long)b * C + c] = (bf16_t)accCv;
  // distraction over input content vectors (nats.py:545-546)
  const float g = tanhf(Ucon[c] * sum + accCv * Wcon[c]);
  ctxs_t[(long)b * C + c] = g;
  hc_bf[(long)b * ldK1 + ctx_off + c] = (bf16_t)g;
  const float mm = (mask_t != nullptr) ? mask_t[b] : 1.f;
  accC[(long)b * C + c] = accCv + mm * g;
}

// ---------------- GRU_1 forward (fused, small-K path) ----------------
// 16 waves: wave w -> (m = w/8, g = (w%8)/2, ks = w%2) with output groups
// g0 = r2 (K = h1|ctx), g1 = u2 (K = h1|ctx), g2 = pxa (h1@Ux_1),
// g3 = pxb (ctx@Wx_1) — K-selectivity comes from zero blocks in W1pk.
// K is split across wave pairs (K1 is ~3H — the serial chain at 8 waves
// measured 34 us/launch, latency-bound).
__global__ __launch_bounds__(1024) void cond_gru1_step_fwd(
    const bf16_t* __restrict__ hc_bf,  // [32][K1] = [h1 | ctx_t] bf16
    const float* __restrict__ h1_t,    // [B][H] fp32
    const bf16_t* __restrict__ W1pk,   // [ngrp*4*16][K1]
    const float* __restrict__ b1,      // [2H]
    const float* __restrict__ bx1,     // [H]
    const float* __restrict__ mask_t,  // [B] or null
    float* __restrict__ h2_t,          // [B][H] out
    bf16_t* __restrict__ h2bf_out,     // [32][Hpad] out
    int ld_h2bf,
    bf16_t* __restrict__ saved1_t,     // [B][4H] (r2,u2,pxa,hbar)
    int B, int H, int K1) {
  __shared__ float pre[4][2][32][JB + 1];

  const int wg = blockIdx.x;
  const int wave = threadIdx.x / NATS_WAVE;
  const int m = wave / 8;
  const int g = (wave % 8) / 2;
  const int ks = wave % 2;
  const int j0 = wg * JB;
  const int khalf = ((K1 / 2 + 31) / 32) * 32;
  const int kbeg = ks * khalf;
  const int kend = min(K1, (ks + 1) * khalf);

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const bf16_t* brow = W1pk + (long)(wg * 4 + g) * JB * K1;
  NATS_MFMA_KLOOP(acc, hc_bf, 16 * m, K1, brow, 0, K1, kbeg, kend);
  {
    const int lane = threadIdx.x & (NATS_WAVE - 1);
    const int col = lane & 15;
    const int rbase = 16 * m + (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) pre[g][ks][rbase + i][col] = acc[i];
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < B * JB; idx += blockDim.x) {
    const int b = idx / JB;
    const int c = idx % JB;
    const int j = j0 + c;
    if (j >= H) continue;
    const float r2 =
        nats_sigmoid(pre[0][0][b][c] + pre[0][1][b][c] + b1[j]);
    const float u2 =
        nats_sigmoid(pre[1][0][b][c] + pre[1][1][b][c] + b1[H + j]);
    const float pxa = pre[2][0][b][c] + pre[2][1][b][c];
    const float pxb = pre[3][0][b][c] + pre[3][1][b][c];
    const float hbar = tanhf((pxa + bx1[j]) * r2 + pxb);
    const float h1v = h1_t[(long)b * H + j];
    float h2 = u2 * h1v + (1.f - u2) * hbar;
    if (mask_t != nullptr) {
      const float mm = mask_t[b];
      h2 = mm * h2 + (1.f - mm) * h1v;
    }
    h2_t[(long)b * H + j] = h2;
    h2bf_out[(long)b * ld_h2bf + j] = (bf16_t)h2;
    saved1_t[(long)b * 4 * H + j] = (bf16_t)r2;
    saved1_t[(long)b * 4 * H + H + j] = (bf16_t)u2;
    saved1_t[(long)b * 4 * H + 2 * H + j] = (bf16_t)pxa;
    saved1_t[(long)b * 4 * H + 3 * H + j] = (bf16_t)hbar;
  }
}

// ---------------- GRU_1 forward (split-K) ----------------
// The single-kernel variant ran ngrpH (=63 at dim 1000) workgroups — a
// quarter of the chip — at 31.4us/step (profiles/
// cnn_kernel_stats_barrierv2.csv). Split-K: grid (ngrpH, KS) blocks each
// compute a K-quarter of all four gate GEMMs (16 waves: m x group x
// half-split as before) and store fp32 partials; a pointwise kernel sums
// the KS partials and applies the gate nonlinearities (nats.py:529-553).
__global__ __launch_bounds__(1024) void cond_gru1_gemm_splitk(
    const bf16_t* __restrict__ hc_bf,  // [32][K1] = [h1 | ctx_t] bf16
    const bf16_t* __restrict__ W1pk,   // [ngrp*4*16][K1]
    float* __restrict__ part,          // [KS][4][32][Hpad]
    int H, int K1, int Hpad) {
  __shared__ float pre[4][2][32][JB + 1];

  const int wg = blockIdx.x;
  const int ksb = blockIdx.y;
  const int KS = gridDim.y;
  const int wave = threadIdx.x / NATS_WAVE;
  const int m = wave / 8;
  const int g = (wave % 8) / 2;
  const int ks = wave % 2;
  const int j0 = wg * JB;
  const int nchunk = 2 * KS;
  const int kchunk = ((K1 / nchunk + 31) / 32) * 32;
  const int kidx = 2 * ksb + ks;
  const int kbeg = min(K1, kidx * kchunk);
  const int kend = min(K1, (kidx + 1) * kchunk);

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const bf16_t* brow = W1pk + (long)(wg * 4 + g) * JB * K1;
  NATS_MFMA_KLOOP(acc, hc_bf, 16 * m, K1, brow, 0, K1, kbeg, kend);
  {
    const int lane = threadIdx.x & (NATS_WAVE - 1);
    const int col = lane & 15;
    const int rbase = 16 * m + (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) pre[g][ks][rbase + i][col] = acc[i];
  }
  __syncthreads();

  float* dst = part + ((long)ksb * 4 + 0) * 32 * Hpad;
  for (int idx = threadIdx.x; idx < 4 * 32 * JB; idx += blockDim.x) {
    const int gg = idx / (32 * JB);
    const int b = (idx / JB) % 32;
    const int c = idx % JB;
    dst[((long)gg * 32 + b) * Hpad + j0 + c] =
        pre[gg][0][b][c] + pre[gg][1][b][c];
  }
}

__global__ void cond_gru1_step_pointwise(
    const float* __restrict__ part,    // [KS][4][32][Hpad]
    int KS,
    const float* __restrict__ h1_t,    // [B][H] fp32
    const float* __restrict__ b1,      // [2H]
    const float* __restrict__ bx1,     // [H]
    const float* __restrict__ mask_t,  // [B] or null
    float* __restrict__ h2_t,          // [B][H] out
    bf16_t* __restrict__ h2bf_out,     // [32][ld_h2bf] out
    int ld_h2bf,
    bf16_t* __restrict__ saved1_t,     // [B][4H] (r2,u2,pxa,hbar)
    int B, int H, int Hpad) {
  const long total = (long)B * H;
  for (long idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int b = idx / H;
    const int j = idx % H;
    float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
    const long bj = (long)b * Hpad + j;
    for (int k = 0; k < KS; ++k) {
      const float* pk = part + (long)k * 4 * 32 * Hpad;
      p0 += pk[bj];
      p1 += pk[(long)32 * Hpad + bj];
      p2 += pk[(long)2 * 32 * Hpad + bj];
      p3 += pk[(long)3 * 32 * Hpad + bj];
    }
    const float r2 = nats_sigmoid(p0 + b1[j]);
    const float u2 = nats_sigmoid(p1 + b1[H + j]);
    const float pxa = p2;
    const float hbar = tanhf((pxa + bx1[j]) * r2 + p3);
    const float h1v = h1_t[idx];
    float h2 = u2 * h1v + (1.f - u2) * hbar;
    if (mask_t != nullptr) {
      const float mm = mask_t[b];
      h2 = mm * h2 + (1.f - mm) * h1v;
    }
    h2_t[idx] = h2;
    h2bf_out[(long)b * ld_h2bf + j] = (bf16_t)h2;
    saved1_t[(long)b * 4 * H + j] = (bf16_t)r2;
    saved1_t[(long)b * 4 * H + H + j] = (bf16_t)u2;
    saved1_t[(long)b * 4 * H + 2 * H + j] = (bf16_t)pxa;
    saved1_t[(long)b * 4 * H + 3 * H + j] = (bf16_t)hbar;
  }
}

// ---------------- backward kernels ----------------

// Fused per-step backward prologue: the GRU_1 pointwise, the independent
// dctx passthrough (upstream readout grad + acc-chain, formerly its own
// cond_dctx_dir launch), and the per-step re-zero of dot_buf (consumed by
// last step's scatter/reduce before this kernel runs — stream order) all
// share one grid-stride index space.
__global__ void cond_gru1_bwd_pointwise(
    const float* __restrict__ dh_carry,   // [B][H] (split-K half 0)
    const float* __restrict__ dh_carry2,  // [B][H] half 1 or null
    const float* __restrict__ dh2_all_t,  // [B][H] or null
    const bf16_t* __restrict__ saved1_t,  // [B][4H]
    const float* __restrict__ h1_all_t,   // [B][H]
    const float* __restrict__ bx1,        // [H]
    const float* __restrict__ mask_t,     // [B] or null
    bf16_t* __restrict__ dstep1,          // [32][K3H] [dpr2|dpu2|dpxa_lin]
    bf16_t* __restrict__ dstepC,          // [32][K3H] [dpr2|dpu2|dpx2]
    int ldK3,
    float* __restrict__ ddirect_h1,       // [B][H]
    bf16_t* __restrict__ dpre1_t,         // [B][4H]
    const float* __restrict__ dctxs_t,    // [B][C] or null (upstream)
    const float* __restrict__ daccC,      // [B][C] (pre-update, read)
    float* __restrict__ dctx_dir,         // [B][C] out
    int C,
    float* __restrict__ dot_buf,          // [B] (zeroed for this step)
    int B, int H) {
  const long total = (long)B * H;
  const long total_all = total + (long)B * C + B;
  for (long idx2 = blockIdx.x * blockDim.x + threadIdx.x; idx2 < total_all;
       idx2 += (long)gridDim.x * blockDim.x) {
    if (idx2 >= total) {
      const long e = idx2 - total;
      if (e < (long)B * C) {
        const int b = e / C;
        const float mm = (mask_t != nullptr) ? mask_t[b] : 1.f;
        float v = mm * daccC[e];
        if (dctxs_t != nullptr) v += dctxs_t[e];
        dctx_dir[e] = v;
      } else {
        dot_buf[e - (long)B * C] = 0.f;
      }
      continue;
    }
    const long idx = idx2;
    const int b = idx / H;
    const int j = idx % H;
    float dh2 = dh_carry[idx];
    if (dh_carry2 != nullptr) dh2 += dh_carry2[idx];
    if (dh2_all_t != nullptr) dh2 += dh2_all_t[idx];
    const float r2 = (float)saved1_t[(long)b * 4 * H + j];
    const float u2 = (float)saved1_t[(long)b * 4 * H + H + j];
    const float pxa = (float)saved1_t[(long)b * 4 * H + 2 * H + j];
    const float hbar = (float)saved1_t[(long)b * 4 * H + 3 * H + j];
    const float h1v = h1_all_t[idx];
    const float mm = (mask_t != nullptr) ? mask_t[b] : 1.f;
    const float du2 = dh2 * mm * (h1v - hbar);
    const float dhbar = dh2 * mm * (1.f - u2);
    const float dpx2 = dhbar * (1.f - hbar * hbar);
    const float dpxa_lin = dpx2 * r2;
    const float dr2 = dpx2 * (pxa + bx1[j]);
    const float dpr2 = dr2 * r2 * (1.f - r2);
    const float dpu2 = du2 * u2 * (1.f - u2);
    ddirect_h1[idx] = dh2 * (mm * u2 + (1.f - mm));
    dstep1[(long)b * ldK3 + j] = (bf16_t)dpr2;
    dstep1[(long)b * ldK3 + H + j] = (bf16_t)dpu2;
    dstep1[(long)b * ldK3 + 2 * H + j] = (bf16_t)dpxa_lin;
    dstepC[(long)b * ldK3 + j] = (bf16_t)dpr2;
    dstepC[(long)b * ldK3 + H + j] = (bf16_t)dpu2;
    dstepC[(long)b * ldK3 + 2 * H + j] = (bf16_t)dpx2;
    dpre1_t[(long)b * 4 * H + j] = (bf16_t)dpr2;
    dpre1_t[(long)b * 4 * H + H + j] = (bf16_t)dpu2;
    dpre1_t[(long)b * 4 * H + 2 * H + j] = (bf16_t)dpx2;
    dpre1_t[(long)b * 4 * H + 3 * H + j] = (bf16_t)dpxa_lin;
  }
}

// Dual recurrent-GEMM with the distraction-gate backward fused into the
// context side's epilogue. grid.y == 0: dh1 = ddirect_h1 + dstep1 @
// [U_1|Ux_1]^T (plain). grid.y == 1: the former dctx_buf value (dctx_dir
// + dstepC @ [W_1|Wx_1]^T) never touches memory — the gate backward
// (tanh', Ucon/Wcon splits, daccC accumulate; nats.py:545-546 reverse)
// is applied in-register, eliminating the separate cond_gate_bwd pass.
__global__ __launch_bounds__(384) void cond_bwd_gemm_dual_gate(
    const bf16_t* __restrict__ dstep1, const bf16_t* __restrict__ U1cat,
    const float* __restrict__ ddirect_h1, float* __restrict__ dh1_buf,
    int H, const bf16_t* __restrict__ dstepC,
    const bf16_t* __restrict__ W1cat, const float* __restrict__ dctx_dir,
    int C, int Kpad,
    const float* __restrict__ ctxs_t,   // [B][C] gated value
    const float* __restrict__ Ucon, const float* __restrict__ Wcon,
    float* __restrict__ daccC,          // [B][C] in/out
    float* __restrict__ dctxpre_f32,    // [B][C] out
    bf16_t* __restrict__ dctxpre_all_t, // [B][C] out
    const bf16_t* __restrict__ ctxpre_t,    // [B][C] (saved pre-gate sum)
    const bf16_t* __restrict__ accC_used_t, // [B][C] (saved acc state)
    float* __restrict__ gdUcon,         // [C] (+=, one atomic/col/step)
    float* __restrict__ gdWcon,         // [C]
    int B) {
  // dU_con[c] = sum_{t,b} dg*ctxpre and dW_con[c] = sum dg*accC_used
  // fold in here (block-local b-reduction + one atomic per column per
  // step) — the torch expression re-streamed three (T,B,C) buffers
  __shared__ float ucon_s[JB], wcon_s[JB];
  __shared__ float part[3][32][JB + 1];
  const bool ctx_side = (blockIdx.y == 1);
  if (ctx_side && threadIdx.x < JB) {
    ucon_s[threadIdx.x] = 0.f;
    wcon_s[threadIdx.x] = 0.f;
  }
  const int N = ctx_side ? C : H;
  if ((int)blockIdx.x * JB >= N) return;
  const bf16_t* dstep = ctx_side ? dstepC : dstep1;
  const bf16_t* Wt = ctx_side ? W1cat : U1cat;

  const int wg = blockIdx.x;
  const int wave = threadIdx.x / NATS_WAVE;
  const int m = wave / 3;
  const int ks = wave % 3;
  const int i0 = wg * JB;
  const int kchunk = ((Kpad / 3 + 31) / 32) * 32;
  const int kbeg = ks * kchunk;
  const int kend = min(Kpad, (ks + 1) * kchunk);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  NATS_MFMA_KLOOP(acc, dstep, 16 * m, Kpad, Wt, i0, Kpad, kbeg, kend);
  {
    const int lane = threadIdx.x & (NATS_WAVE - 1);
    const int col = lane & 15;
    const int rbase = 16 * m + (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) part[ks][rbase + i][col] = acc[i];
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < B * JB; idx += blockDim.x) {
    const int b = idx / JB;
    const int cc = idx % JB;
    const int i = i0 + cc;
    if (i >= N) continue;
    const long bi = (long)b * N + i;
    const float sum = part[0][b][cc] + part[1][b][cc] + part[2][b][cc];
    if (!ctx_side) {
      dh1_buf[bi] = ddirect_h1[bi] + sum;
    } else {
      const float v = dctx_dir[bi] + sum;  // former dctx_buf value
      const float g = ctxs_t[bi];
      const float dg = v * (1.f - g * g);
      const float dpre = dg * Ucon[i];
      dctxpre_f32[bi] = dpre;
      dctxpre_all_t[bi] = (bf16_t)dpre;
      daccC[bi] += dg * Wcon[i];
      atomicAdd(ucon_s + cc, dg * (float)ctxpre_t[bi]);
      atomicAdd(wcon_s + cc, dg * (float)accC_used_t[bi]);
    }
  }
  if (ctx_side) {
    __syncthreads();
    if (threadIdx.x < JB && i0 + (int)threadIdx.x < N) {
      atomicAdd(gdUcon + i0 + threadIdx.x, ucon_s[threadIdx.x]);
      atomicAdd(gdWcon + i0 + threadIdx.x, wcon_s[threadIdx.x]);
    }
  }
}

// A-fragment built from an UNPADDED fp32 [Rows][Kcols] matrix with
// zero-fill outside (rows >= Rows, k >= Kcols) — lets the dh1 += dpstate
// @ W_att^T GEMM consume the atomically-reduced fp32 dpstate directly,
// without the former cond_dpstate_cast pass.
__device__ __forceinline__ bf16x8 frag_a_f32pad(const float* A, int row0,
                                                int ld, int k0, int Rows,
                                                int Kcols) {
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  const int r = row0 + (lane & 15);
  const int kb = k0 + (lane >> 4) * 8;
  bf16x8 v;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int k = kb + i;
    v[i] = (bf16_t)((r < Rows && k < Kcols) ? A[(long)r * ld + k] : 0.f);
  }
  return v;
}

// dh_{t-1} carry GEMM with grid.y split-K and CONSUMER-side combine:
// the unsplit kernel ran 63 blocks (a quarter of the chip) at 12.6 us;
// round 1's atomic-combine split measured neutral (atomics + re-zero),
// so here each half writes its own [B][H] partial — full coverage, no
// zeroing — and the next step's GRU_1 pointwise sums the two.
__global__ __launch_bounds__(384) void cond_dh_carry_gemm_split(
    const bf16_t* __restrict__ dstep,  // [32][Kpad]
    const bf16_t* __restrict__ Wt,     // [ngrp*16][Kpad]
    const float* __restrict__ ddirect, // [B][H] (added by grid.y == 0)
    float* __restrict__ out,           // [2][B][H] partials
    int B, int H, int Kpad) {
  __shared__ float part[3][32][JB + 1];
  const int wg = blockIdx.x;
  const int ks2 = blockIdx.y;        // K half
  const int wave = threadIdx.x / NATS_WAVE;
  const int m = wave / 3;
  const int ks = wave % 3;
  const int i0 = wg * JB;
  const int khalf = ((Kpad / 2 + 31) / 32) * 32;
  const int hbeg = min(Kpad, ks2 * khalf);
  const int hend = min(Kpad, hbeg + khalf);
  const int kchunk = (((hend - hbeg) / 3 + 31) / 32) * 32;
  const int kbeg = hbeg + ks * kchunk;
  const int kend = min(hend, kbeg + kchunk);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  NATS_MFMA_KLOOP(acc, dstep, 16 * m, Kpad, Wt, i0, Kpad, kbeg, kend);
  {
    const int lane = threadIdx.x & (NATS_WAVE - 1);
    const int col = lane & 15;
    const int rbase = 16 * m + (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) part[ks][rbase + i][col] = acc[i];
  }
  __syncthreads();
  float* o = out + (long)ks2 * B * H;
  for (int idx = threadIdx.x; idx < B * JB; idx += blockDim.x) {
    const int b = idx / JB;
    const int cc = idx % JB;
    const int i = i0 + cc;
    if (i >= H) continue;
    float v = part[0][b][cc] + part[1][b][cc] + part[2][b][cc];
    if (ks2 == 0) v += ddirect[(long)b * H + i];
    o[(long)b * H + i] = v;
  }
}

__global__ __launch_bounds__(384) void cond_dh1_att_gemm(
    const float* __restrict__ dpstate_t,  // [B][A] fp32 (atomic-reduced)
    const bf16_t* __restrict__ WattB,     // [ngrpH*16][Apad]
    float* __restrict__ dh1_buf,          // [B][H] in/out (+=)
    int B, int H, int A, int Apad) {
  __shared__ float part[3][32][JB + 1];
  const int wg = blockIdx.x;
  const int wave = threadIdx.x / NATS_WAVE;
  const int m = wave / 3;
  const int ks = wave % 3;
  const int i0 = wg * JB;
  const int kchunk = ((Apad / 3 + 31) / 32) * 32;
  const int kbeg = ks * kchunk;
  const int kend = min(Apad, (ks + 1) * kchunk);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k = kbeg; k + 32 <= kend; k += 32) {
    bf16x8 a = frag_a_f32pad(dpstate_t, 16 * m, A, k, B, A);
    bf16x8 bfr = frag_bt_rowmajor(WattB, i0, Apad, k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
  }
  {
    const int lane = threadIdx.x & (NATS_WAVE - 1);
    const int col = lane & 15;
    const int rbase = 16 * m + (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 4; ++i) part[ks][rbase + i][col] = acc[i];
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < B * JB; idx += blockDim.x) {
    const int b = idx / JB;
    const int cc = idx % JB;
    const int i = i0 + cc;
    if (i >= H) continue;
    dh1_buf[(long)b * H + i] +=
        part[0][b][cc] + part[1][b][cc] + part[2][b][cc];
  }
}

// attention backward, stage 1 (one WAVE per (b,s)): dalpha[s,b] =
// ctx[s,b,:] . dctxpre[b,:] (+ acc-chain and upstream alpha grads) and the
// softmax-backward dot partial, accumulated by atomics into dot_buf.
__global__ __launch_bounds__(256) void cond_attn_bwd_dalpha(
    const bf16_t* __restrict__ ctx_bf,      // [Ts][B][C]
    const float* __restrict__ dctxpre_f32,  // [B][C]
    const float* __restrict__ alphas_t,     // [B][Ts]
    const float* __restrict__ daccA,        // [B][Ts]
    const float* __restrict__ mask_t,       // [B] or null
    const float* __restrict__ dalphas_t,    // [B][Ts] or null
    float* __restrict__ dal_buf,            // [Ts][B]
    float* __restrict__ dot_buf,            // [B] (zeroed per step)
    int B, int Ts, int C) {
  const int b = blockIdx.x;
  const int wave = threadIdx.x / NATS_WAVE;
  const int lane = threadIdx.x & (NATS_WAVE - 1);
  const int s = blockIdx.y * 4 + wave;
  if (s >= Ts) return;
  const bf16_t* crow = ctx_bf + ((long)s * B + b) * C;
  const float* drow = dctxpre_f32 + (long)b * C;
  float part = 0.f;
  const int C8 = C & ~7;
  for (int c = lane * 8; c < C8; c += NATS_WAVE * 8) {
    bf16x8 v = *(const bf16x8*)(crow + c);
    const float4 d0 = *(const float4*)(drow + c);
    const float4 d1 = *(const float4*)(drow + c + 4);
    part += (float)v[0] * d0.x + (float)v[1] * d0.y + (float)v[2] * d0.z +
            (float)v[3] * d0.w + (float)v[4] * d1.x + (float)v[5] * d1.y +
            (float)v[6] * d1.z + (float)v[7] * d1.w;
  }
  for (int c = C8 + lane; c < C; c += NATS_WAVE) {
    part += (float)crow[c] * drow[c];
  }
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1)
    part += __shfl_down(part, off);
  if (lane == 0) {
    float dal = part;
    const float mm = (mask_t != nullptr) ? mask_t[b] : 1.f;
    dal += mm * daccA[(long)b * Ts + s];
    if (dalphas_t != nullptr) dal += dalphas_t[(long)b * Ts + s];
    dal_buf[(long)s * B + b] = dal;
    // NOTE: the dot(alpha, dal) fold was MEASURED OUT here twice — a
    // per-s atomicAdd serializes ~Ts adds on one address (~120us/step,
    // round 1), and a per-BLOCK atomic tree still serialized Ts/4 adds
    // per address (dalpha 9us -> 47.9us/call, prof_r2). The separate
    // cond_attn_bwd_dot pass (~3us) stays.
  }
}

// softmax-backward dot: dot_buf[b] = sum_s alpha[b,s] * dal[s,b]
__global__ __launch_bounds__(256) void cond_attn_bwd_dot(
    const float* __restrict__ alphas_t, const float* __restrict__ dal_buf,
    float* __restrict__ dot_buf, int B, int Ts) {
  const int b = blockIdx.x;
  __shared__ float red[256 / NATS_WAVE];
  float part = 0.f;
  for (int s = threadIdx.x; s < Ts; s += blockDim.x)
    part += alphas_t[(long)b * Ts + s] * dal_buf[(long)s * B + b];
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1)
    part += __shfl_down(part, off);
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0)
    red[threadIdx.x / NATS_WAVE] = part;
  __syncthreads();
  if (threadIdx.x == 0) {
    float S = 0.f;
    for (int w = 0; w < (int)blockDim.x / NATS_WAVE; ++w) S += red[w];
    dot_buf[b] = S;
  }
}

// attention backward, stage 2 (grid (B, ceil(Ts/256))): softmax backward,
// per-(s,i) dpc into dpctx_acc + acc_alpha carry; tanh values stored to
// pc_buf [B][A][Ts] (coalesced in s) for the stage-3 reduction.
__global__ __launch_bounds__(256) void cond_attn_bwd_scatter(
    const float* __restrict__ alphas_t,     // [B][Ts]
    const float* __restrict__ dal_buf,      // [Ts][B]
    const float* __restrict__ dot_buf,      // [B]
    const float* __restrict__ pctx,         // [Ts][B][A]
    const float* __restrict__ pstate_t,     // [B][A]
    const float* __restrict__ accA_used_t,  // [B][Ts]
    const float* __restrict__ Dwei, const float* __restrict__ Uatt,
    float* __restrict__ daccA,              // [B][Ts] (+=, atomic)
    float* __restrict__ dpctx_acc,          // [Ts][B][A] (+=)
    bf16_t* __restrict__ pc_buf,            // [B][A][Tpad8]
    int B, int Ts, int A, int Tpad8) {
  // grid (B, s-chunks, A-chunks): the un-chunked variant ran only
  // B * Ts/256 blocks (2 waves/CU at the CNN/DM shape) and was
  // latency-bound on its serial per-A float4 RMW chain.
  const int b = blockIdx.x;
  const int s = blockIdx.y * blockDim.x + threadIdx.x;
  if (s >= Ts) return;
  const int ACH = gridDim.z;
  const int chunkA = ((A + ACH - 1) / ACH + 3) & ~3;  // 16B-aligned splits
  const int ibeg = blockIdx.z * chunkA;
  const int iend = min(A, ibeg + chunkA);
  if (ibeg >= iend) return;
  const float al = alphas_t[(long)b * Ts + s];
  const float de = al * (dal_buf[(long)s * B + b] - dot_buf[b]);
  const float accAu = accA_used_t[(long)b * Ts + s];
  const float* prow = pctx + ((long)s * B + b) * A;
  const float* srow = pstate_t + (long)b * A;
  float* dprow = dpctx_acc + ((long)s * B + b) * A;
  float daccA_add = 0.f;
  const int A4 = (A % 4 == 0) ? iend : ibeg;  // f32x4 needs A % 4 == 0
  int i = ibeg;
  for (; i < A4; i += 4) {
    const float4 p = *(const float4*)(prow + i);
    const float4 st = *(const float4*)(srow + i);
    const float4 dw = *(const float4*)(Dwei + i);
    const float4 ua = *(const float4*)(Uatt + i);
    float4 dpv = *(const float4*)(dprow + i);
    const float pc0 = tanhf(p.x + st.x + accAu * dw.x);
    const float pc1 = tanhf(p.y + st.y + accAu * dw.y);
    const float pc2 = tanhf(p.z + st.z + accAu * dw.z);
    const float pc3 = tanhf(p.w + st.w + accAu * dw.w);
    const float d0 = de * (1.f - pc0 * pc0) * ua.x;
    const float d1 = de * (1.f - pc1 * pc1) * ua.y;
    const float d2 = de * (1.f - pc2 * pc2) * ua.z;
    const float d3 = de * (1.f - pc3 * pc3) * ua.w;
    dpv.x += d0;
    dpv.y += d1;
    dpv.z += d2;
    dpv.w += d3;
    *(float4*)(dprow + i) = dpv;
    daccA_add += d0 * dw.x + d1 * dw.y + d2 * dw.z + d3 * dw.w;
    pc_buf[((long)b * A + i) * Tpad8 + s] = (bf16_t)pc0;
    pc_buf[((long)b * A + i + 1) * Tpad8 + s] = (bf16_t)pc1;
    pc_buf[((long)b * A + i + 2) * Tpad8 + s] = (bf16_t)pc2;
    pc_buf[((long)b * A + i + 3) * Tpad8 + s] = (bf16_t)pc3;
  }
  for (; i < iend; ++i) {
    const float pc = tanhf(prow[i] + srow[i] + accAu * Dwei[i]);
    const float dpc = de * (1.f - pc * pc) * Uatt[i];
    dprow[i] += dpc;
    daccA_add += dpc * Dwei[i];
    pc_buf[((long)b * A + i) * Tpad8 + s] = (bf16_t)pc;
  }
  if (ACH == 1) {
    daccA[(long)b * Ts + s] += daccA_add;
  } else {
    atomicAdd(daccA + (long)b * Ts + s, daccA_add);
  }
}

// attention backward, stage 3 (grid (b, s-chunk)): reduce over an
// s-range of the contiguous pc_buf rows -> dpstate / dD_wei / dU_att /
// dc_att by atomic combine. de and accA_used for the range staged in LDS.
__global__ __launch_bounds__(256) void cond_attn_bwd_reduce(
    const float* __restrict__ alphas_t,     // [B][Ts]
    const float* __restrict__ dal_buf,      // [Ts][B]
    const float* __restrict__ dot_buf,      // [B]
    const float* __restrict__ accA_used_t,  // [B][Ts]
    const bf16_t* __restrict__ pc_buf,      // [B][A][Tpad8]
    const float* __restrict__ Uatt,
    float* __restrict__ dpstate_t,          // [B][A]
    float* __restrict__ gdDwei,             // [A] (atomic)
    float* __restrict__ gdUatt,             // [A] (atomic)
    float* __restrict__ gdcatt,             // [1] (atomic)
    int B, int Ts, int A, int Apad, int Tpad8, int SCH) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const int chunk = (Ts + SCH - 1) / SCH;
  float* sm_de = (float*)smem_raw;      // [chunk]
  float* sm_au = sm_de + chunk;         // [chunk]
  __shared__ float red[256 / NATS_WAVE];
  const int b = blockIdx.x;
  const int sbeg = blockIdx.y * chunk;
  const int send = min(Ts, sbeg + chunk);
  const float dot = dot_buf[b];
  float dc = 0.f;
  for (int s = sbeg + threadIdx.x; s < send; s += blockDim.x) {
    const float de = alphas_t[(long)b * Ts + s] *
                     (dal_buf[(long)s * B + b] - dot);
    sm_de[s - sbeg] = de;
    sm_au[s - sbeg] = accA_used_t[(long)b * Ts + s];
    dc += de;
  }
#pragma unroll
  for (int off = NATS_WAVE / 2; off > 0; off >>= 1) dc += __shfl_down(dc, off);
  if ((threadIdx.x & (NATS_WAVE - 1)) == 0) red[threadIdx.x / NATS_WAVE] = dc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float S = 0.f;
    for (int w = 0; w < (int)blockDim.x / NATS_WAVE; ++w) S += red[w];
    atomicAdd(gdcatt, S);
  }
  for (int i = threadIdx.x; i < A; i += blockDim.x) {
    const bf16_t* prow = pc_buf + ((long)b * A + i) * Tpad8;
    const float ua = Uatt[i];
    float sps = 0.f, sdw = 0.f, sua = 0.f;
    int s = sbeg;
    const int a8beg = (sbeg + 7) & ~7;
    const int a8end = send & ~7;
    for (; s < min(a8beg, send); ++s) {
      const float pc = (float)prow[s];
      const float de = sm_de[s - sbeg];
      const float dpc = de * (1.f - pc * pc) * ua;
      sps += dpc;
      sdw += dpc * sm_au[s - sbeg];
      sua += de * pc;
    }
    for (; s + 8 <= a8end; s += 8) {
      bf16x8 pv = *(const bf16x8*)(prow + s);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float pc = (float)pv[k];
        const float de = sm_de[s + k - sbeg];
        const float dpc = de * (1.f - pc * pc) * ua;
        sps += dpc;
        sdw += dpc * sm_au[s + k - sbeg];
        sua += de * pc;
      }
    }
    for (; s < send; ++s) {
      const float pc = (float)prow[s];
      const float de = sm_de[s - sbeg];
      const float dpc = de * (1.f - pc * pc) * ua;
      sps += dpc;
      sdw += dpc * sm_au[s - sbeg];
      sua += de * pc;
    }
    atomicAdd(&dpstate_t[(long)b * A + i], sps);
    atomicAdd(&gdDwei[i], sdw);
    atomicAdd(&gdUatt[i], sua);
  }
}

inline int cdiv_i(int a, int b) { return (a + b - 1) / b; }

}  // namespace

// ================= host: forward =================
std::vector<torch::Tensor> cond_gru_fwd(
    torch::Tensor yg, torch::Tensor yc, c10::optional<torch::Tensor> mask,
    torch::Tensor init_state, torch::Tensor ctx_bf,
    c10::optional<torch::Tensor> ctx_mask, torch::Tensor pctx,
    torch::Tensor Upk2, torch::Tensor W1pk, torch::Tensor WattPk,
    torch::Tensor b1, torch::Tensor bx1, torch::Tensor Uatt,
    torch::Tensor catt, torch::Tensor Dwei, torch::Tensor Wcon,
    torch::Tensor Ucon,
    c10::optional<torch::Tensor> accC0, c10::optional<torch::Tensor> accA0) {
  const int T = yg.size(0), B = yg.size(1);
  const int H = yc.size(2);
  const int Ts = ctx_bf.size(0), C = ctx_bf.size(2);
  const int A = Uatt.size(0);
  TORCH_CHECK(B <= 32, "cond_gru: batch must be <= 32");
  TORCH_CHECK(yg.dtype() == torch::kBFloat16 && yg.is_contiguous());
  TORCH_CHECK(ctx_bf.dtype() == torch::kBFloat16 && ctx_bf.is_contiguous());
  TORCH_CHECK(pctx.dtype() == torch::kFloat32 && pctx.is_contiguous());
  const int Hpad = Upk2.size(1);
  const int K1 = W1pk.size(1);
  const int ngrpH = cdiv_i(H, JB);
  TORCH_CHECK(Upk2.size(0) == ngrpH * 3 * JB);
  TORCH_CHECK(W1pk.size(0) == ngrpH * 4 * JB);

  auto optsF = yg.options().dtype(torch::kFloat32);
  auto optsB = yg.options();
  auto h2_all = torch::empty({T, B, H}, optsF);
  auto h1_all = torch::empty({T, B, H}, optsF);
  auto ctxs_all = torch::empty({T, B, C}, optsF);
  auto alphas_all = torch::empty({T, B, Ts}, optsF);
  auto saved2 = torch::empty({T, B, 3 * H}, optsB);
  auto saved1 = torch::empty({T, B, 4 * H}, optsB);
  auto pstate_all = torch::empty({T, B, A}, optsF);
  auto ctxpre_all = torch::empty({T, B, C}, optsB);
  auto accA_used = torch::empty({T, B, Ts}, optsF);
  auto accC_used = torch::empty({T, B, C}, optsB);
  auto accA = accA0.has_value() ? accA0->contiguous().to(torch::kFloat32)
                                : torch::zeros({B, Ts}, optsF);
  auto accC = accC0.has_value() ? accC0->contiguous().to(torch::kFloat32)
                                : torch::zeros({B, C}, optsF);
  auto h2bf = torch::zeros({2, 32, Hpad}, optsB);
  const int GRU1_KS = 4;
  auto gru1_part = torch::empty({GRU1_KS, 4, 32, Hpad}, optsF);
  const int GRU2_KS = 4;
  auto gru2_part = torch::empty({GRU2_KS, 3, 32, Hpad}, optsF);
  auto hc_bf = torch::zeros({32, K1}, optsB);
  auto e_buf = torch::zeros({Ts, B}, optsF);  // escore accumulates into it
  const int PS_KS = 4;
  auto ps_part = torch::empty({PS_KS, 32, A}, optsF);
  auto ctxpre_f32 = torch::zeros({B, C}, optsF);  // gate_fwd re-zeroes
  const int SCH = std::max(1, std::min(8, Ts / 64));
  auto init_f = init_state.contiguous().to(torch::kFloat32);
  h2bf[0].slice(0, 0, B).slice(1, 0, H).copy_(init_f.to(torch::kBFloat16));

  const float* mask_p = nullptr;
  torch::Tensor mask_c;
  if (mask.has_value()) {
    mask_c = mask->contiguous().to(torch::kFloat32);
    mask_p = mask_c.data_ptr<float>();
  }
  const float* cmask_p = nullptr;
  torch::Tensor cmask_c;
  if (ctx_mask.has_value()) {
    cmask_c = ctx_mask->contiguous().to(torch::kFloat32);
    cmask_p = cmask_c.data_ptr<float>();
  }

  auto stream = at::cuda::getCurrentCUDAStream().stream();
  const int Apad16 = cdiv_i(A, 16) * 16;
  const long hbstride = (long)32 * Hpad;
  bf16_t* h2bf_p = (bf16_t*)h2bf.data_ptr();

  for (int t = 0; t < T; ++t) {
    const float* h2prev = (t == 0)
                              ? init_f.data_ptr<float>()
                              : h2_all.data_ptr<float>() + (long)(t - 1) * B * H;
    const float* mt = mask_p ? mask_p + (long)t * B : nullptr;
    // 1) GRU_2 -> h1 (bf16 into hc_bf cols [0,H)); split-K when K large
    if (Hpad >= 1024) {
      hipLaunchKernelGGL(nats_gru2_gemm_splitk, dim3(ngrpH, GRU2_KS),
                         dim3(384), 0, stream, h2bf_p + (t % 2) * hbstride,
                         (const bf16_t*)Upk2.data_ptr(),
                         gru2_part.data_ptr<float>(), Hpad);
      hipLaunchKernelGGL(nats_gru2_step_pointwise, dim3(cdiv_i(B * H, 256)),
                         dim3(256), 0, stream, gru2_part.data_ptr<float>(),
                         GRU2_KS, h2prev,
                         (const bf16_t*)yg.data_ptr() + (long)t * B * 2 * H,
                         (const bf16_t*)yc.data_ptr() + (long)t * B * H, mt,
                         h1_all.data_ptr<float>() + (long)t * B * H,
                         (bf16_t*)hc_bf.data_ptr(), K1,
                         (bf16_t*)saved2.data_ptr() + (long)t * B * 3 * H, B,
                         H, Hpad);
    } else {
      hipLaunchKernelGGL(nats_gru_step_fwd, dim3(ngrpH), dim3(384), 0, stream,
                         h2bf_p + (t % 2) * hbstride, h2prev,
                         (const bf16_t*)Upk2.data_ptr(),
                         (const bf16_t*)yg.data_ptr() + (long)t * B * 2 * H,
                         (const bf16_t*)yc.data_ptr() + (long)t * B * H, mt,
                         h1_all.data_ptr<float>() + (long)t * B * H,
                         (bf16_t*)hc_bf.data_ptr(), K1,
                         (bf16_t*)saved2.data_ptr() + (long)t * B * 3 * H, B,
                         H, Hpad);
    }
    // 2) pstate = h1 @ W_att (split-K partials; combined in softmax)
    hipLaunchKernelGGL(cond_small_gemm_bt, dim3(2, Apad16 / 16, PS_KS),
                       dim3(64), 0, stream, (const bf16_t*)hc_bf.data_ptr(),
                       (const bf16_t*)WattPk.data_ptr(),
                       ps_part.data_ptr<float>(), B, A, K1,
                       (int)WattPk.size(1));
    // 3) attention scores (s- and A-parallel) + softmax + acc_alpha
    hipLaunchKernelGGL(cond_attn_escore,
                       dim3(B, cdiv_i(Ts, 256), A >= 16 ? 4 : 1), dim3(256),
                       3 * A * sizeof(float), stream, pctx.data_ptr<float>(),
                       ps_part.data_ptr<float>(), PS_KS,
                       accA.data_ptr<float>(), Dwei.data_ptr<float>(),
                       Uatt.data_ptr<float>(), catt.data_ptr<float>(),
                       e_buf.data_ptr<float>(), B, Ts, A);
    // 1024 threads: the kernel runs only B blocks (latency-bound) —
    // wider blocks quarter the serial passes over Ts
    hipLaunchKernelGGL(cond_attn_softmax, dim3(B), dim3(1024), 0, stream,
                       accA.data_ptr<float>(),
                       accA_used.data_ptr<float>() + (long)t * B * Ts,
                       cmask_p, mt, e_buf.data_ptr<float>(),
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       ps_part.data_ptr<float>(), PS_KS,
                       pstate_all.data_ptr<float>() + (long)t * B * A, A, B,
                       Ts);
    // 4) weighted context (s-chunked partials) + gate + acc_ctx
    // (ctxpre_f32 arrives zeroed: gate_fwd re-zeroes it after reading)
    hipLaunchKernelGGL(cond_attn_ctx_partial,
                       dim3(B, cdiv_i(C, 256), SCH), dim3(256), 0, stream,
                       (const bf16_t*)ctx_bf.data_ptr(),
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       ctxpre_f32.data_ptr<float>(), B, Ts, C, SCH);
    hipLaunchKernelGGL(cond_attn_gate_fwd, dim3(B, cdiv_i(C, 256)), dim3(256),
                       0, stream, ctxpre_f32.data_ptr<float>(),
                       Ucon.data_ptr<float>(), Wcon.data_ptr<float>(),
                       accC.data_ptr<float>(),
                       (bf16_t*)accC_used.data_ptr() + (long)t * B * C,
                       (bf16_t*)ctxpre_all.data_ptr() + (long)t * B * C,
                       ctxs_all.data_ptr<float>() + (long)t * B * C,
                       (bf16_t*)hc_bf.data_ptr(), Hpad, K1, mt, B, C);
    // 5) GRU_1 -> h2: split-K for large K (raises block parallelism 4x),
    // single fused kernel when K is small (split overhead dominates)
    if (K1 >= 2048) {
      hipLaunchKernelGGL(cond_gru1_gemm_splitk, dim3(ngrpH, GRU1_KS),
                         dim3(1024), 0, stream,
                         (const bf16_t*)hc_bf.data_ptr(),
                         (const bf16_t*)W1pk.data_ptr(),
                         gru1_part.data_ptr<float>(), H, K1, Hpad);
      hipLaunchKernelGGL(cond_gru1_step_pointwise, dim3(cdiv_i(B * H, 256)),
                         dim3(256), 0, stream, gru1_part.data_ptr<float>(),
                         GRU1_KS, h1_all.data_ptr<float>() + (long)t * B * H,
                         b1.data_ptr<float>(), bx1.data_ptr<float>(), mt,
                         h2_all.data_ptr<float>() + (long)t * B * H,
                         h2bf_p + ((t + 1) % 2) * hbstride, Hpad,
                         (bf16_t*)saved1.data_ptr() + (long)t * B * 4 * H, B,
                         H, Hpad);
    } else {
      hipLaunchKernelGGL(cond_gru1_step_fwd, dim3(ngrpH), dim3(1024), 0,
                         stream, (const bf16_t*)hc_bf.data_ptr(),
                         h1_all.data_ptr<float>() + (long)t * B * H,
                         (const bf16_t*)W1pk.data_ptr(), b1.data_ptr<float>(),
                         bx1.data_ptr<float>(), mt,
                         h2_all.data_ptr<float>() + (long)t * B * H,
                         h2bf_p + ((t + 1) % 2) * hbstride, Hpad,
                         (bf16_t*)saved1.data_ptr() + (long)t * B * 4 * H, B,
                         H, K1);
    }
  }
  HIP_CHECK(hipGetLastError());
  return {h2_all, ctxs_all, alphas_all, accC, accA, h1_all, saved2, saved1,
          pstate_all, ctxpre_all, accA_used, accC_used};
}

// ================= host: backward =================
std::vector<torch::Tensor> cond_gru_bwd(
    torch::Tensor dh2_all, c10::optional<torch::Tensor> dctxs_all,
    c10::optional<torch::Tensor> dalphas_all,
    c10::optional<torch::Tensor> daccC_f, c10::optional<torch::Tensor> daccA_f,
    // saved forward state
    torch::Tensor yc, torch::Tensor h1_all, torch::Tensor h2_all,
    torch::Tensor ctxs_all, torch::Tensor alphas_all, torch::Tensor saved2,
    torch::Tensor saved1, torch::Tensor pstate_all, torch::Tensor ctxpre_all,
    torch::Tensor accA_used, torch::Tensor accC_used, torch::Tensor ctx_bf,
    torch::Tensor pctx, torch::Tensor init_state,
    c10::optional<torch::Tensor> mask,
    // packed weights for the in-loop GEMMs
    torch::Tensor U1cat,   // [ngrpH*16][K3Hpad] = [U_1|Ux_1]
    torch::Tensor W1cat,   // [ngrpC*16][K3Hpad] = [W_1|Wx_1]
    torch::Tensor U2cat,   // [ngrpH*16][K3Hpad] = [U|Ux]
    torch::Tensor WattB,   // [ngrpH*16][Apad32] = W_att
    torch::Tensor bx1, torch::Tensor Dwei, torch::Tensor Uatt,
    torch::Tensor Ucon, torch::Tensor Wcon) {
  const int T = dh2_all.size(0), B = dh2_all.size(1), H = dh2_all.size(2);
  const int Ts = ctx_bf.size(0), C = ctx_bf.size(2);
  const int A = Uatt.size(0);
  const int K3Hpad = U1cat.size(1);
  const int Apad32 = WattB.size(1);
  const int ngrpH = cdiv_i(H, JB);
  const int ngrpC = cdiv_i(C, JB);

  auto optsF = dh2_all.options().dtype(torch::kFloat32);
  auto optsB = dh2_all.options().dtype(torch::kBFloat16);
  auto dpre1_all = torch::empty({T, B, 4 * H}, optsB);
  auto dpre2_all = torch::empty({T, B, 4 * H}, optsB);
  auto dctxpre_all = torch::empty({T, B, C}, optsB);
  auto gdUcon = torch::zeros({C}, optsF);
  auto gdWcon = torch::zeros({C}, optsF);
  auto dpstate_all = torch::zeros({T, B, A}, optsF);  // reduce atomic-adds
  auto dpctx_acc = torch::zeros({Ts, B, A}, optsF);
  auto gdDwei = torch::zeros({A}, optsF);
  auto gdUatt = torch::zeros({A}, optsF);
  auto gdcatt = torch::zeros({1}, optsF);

  auto dh_carry = torch::zeros({2, B, H}, optsF);
  auto dh1_buf = torch::zeros({B, H}, optsF);
  auto ddirect_h1 = torch::empty({B, H}, optsF);
  auto ddirect2 = torch::empty({B, H}, optsF);
  auto dctx_dir = torch::empty({B, C}, optsF);
  auto dctxpre_f32 = torch::empty({B, C}, optsF);
  auto daccA = daccA_f.has_value() ? daccA_f->contiguous().to(torch::kFloat32)
                                   : torch::zeros({B, Ts}, optsF);
  auto daccC = daccC_f.has_value() ? daccC_f->contiguous().to(torch::kFloat32)
                                   : torch::zeros({B, C}, optsF);
  auto dal_buf = torch::empty({Ts, B}, optsF);
  auto dot_buf = torch::empty({B}, optsF);
  const int Tpad8 = (Ts + 7) / 8 * 8;
  auto pc_buf = torch::empty({B, A, Tpad8},
                             dh2_all.options().dtype(torch::kBFloat16));
  const int Apad = Apad32;
  auto dstep1 = torch::zeros({32, K3Hpad}, optsB);
  auto dstepC = torch::zeros({32, K3Hpad}, optsB);
  auto dstep2 = torch::zeros({32, K3Hpad}, optsB);
  auto init_f = init_state.contiguous().to(torch::kFloat32);

  const float* mask_all = nullptr;
  torch::Tensor mask_c;
  if (mask.has_value()) {
    mask_c = mask->contiguous().to(torch::kFloat32);
    mask_all = mask_c.data_ptr<float>();
  }
  torch::Tensor dctxs_c;
  const float* dctxs_p = nullptr;
  if (dctxs_all.has_value()) {
    dctxs_c = dctxs_all->contiguous().to(torch::kFloat32);
    dctxs_p = dctxs_c.data_ptr<float>();
  }
  torch::Tensor dalpha_c;
  const float* dalpha_p = nullptr;
  if (dalphas_all.has_value()) {
    dalpha_c = dalphas_all->contiguous().to(torch::kFloat32);
    dalpha_p = dalpha_c.data_ptr<float>();
  }

  auto stream = at::cuda::getCurrentCUDAStream().stream();
  auto dh2_c = dh2_all.contiguous().to(torch::kFloat32);
  const int pwH = (int)std::min<long>(512, (((long)B * H) + 255) / 256);
  const int pwHC =
      (int)std::min<long>(512, (((long)B * (H + C) + B) + 255) / 256);

  for (int t = T - 1; t >= 0; --t) {
    const float* mt = mask_all ? mask_all + (long)t * B : nullptr;
    const float* h2prev =
        (t == 0) ? init_f.data_ptr<float>()
                 : h2_all.data_ptr<float>() + (long)(t - 1) * B * H;
    // b1 (fused): GRU_1 pointwise + dctx passthrough + dot_buf re-zero
    hipLaunchKernelGGL(cond_gru1_bwd_pointwise, dim3(pwHC), dim3(256), 0,
                       stream, dh_carry.data_ptr<float>(),
                       dh_carry.data_ptr<float>() + (long)B * H,
                       dh2_c.data_ptr<float>() + (long)t * B * H,
                       (const bf16_t*)saved1.data_ptr() + (long)t * B * 4 * H,
                       h1_all.data_ptr<float>() + (long)t * B * H,
                       bx1.data_ptr<float>(), mt,
                       (bf16_t*)dstep1.data_ptr(), (bf16_t*)dstepC.data_ptr(),
                       K3Hpad, ddirect_h1.data_ptr<float>(),
                       (bf16_t*)dpre1_all.data_ptr() + (long)t * B * 4 * H,
                       dctxs_p ? dctxs_p + (long)t * B * C : nullptr,
                       daccC.data_ptr<float>(), dctx_dir.data_ptr<float>(), C,
                       dot_buf.data_ptr<float>(), B, H);
    // b2+b3+b4 fused launch: dh1 = ddirect_h1 + dstep1 @ [U_1|Ux_1]^T and
    // the dctx GEMM with the distraction-gate backward applied in the
    // epilogue (the former separate cond_gate_bwd pass)
    hipLaunchKernelGGL(cond_bwd_gemm_dual_gate,
                       dim3(std::max(ngrpH, ngrpC), 2), dim3(384), 0, stream,
                       (const bf16_t*)dstep1.data_ptr(),
                       (const bf16_t*)U1cat.data_ptr(),
                       ddirect_h1.data_ptr<float>(),
                       dh1_buf.data_ptr<float>(), H,
                       (const bf16_t*)dstepC.data_ptr(),
                       (const bf16_t*)W1cat.data_ptr(),
                       dctx_dir.data_ptr<float>(), C, K3Hpad,
                       ctxs_all.data_ptr<float>() + (long)t * B * C,
                       Ucon.data_ptr<float>(), Wcon.data_ptr<float>(),
                       daccC.data_ptr<float>(), dctxpre_f32.data_ptr<float>(),
                       (bf16_t*)dctxpre_all.data_ptr() + (long)t * B * C,
                       (const bf16_t*)ctxpre_all.data_ptr() + (long)t * B * C,
                       (const bf16_t*)accC_used.data_ptr() + (long)t * B * C,
                       gdUcon.data_ptr<float>(), gdWcon.data_ptr<float>(),
                       B);
    // b5: attention backward — wave-per-(b,s) dalpha with the softmax-bwd
    // dot folded in (block tree + one atomic per block), then the
    // (b,s)-parallel scatter (dpctx/daccA/pc_buf) and the per-(b,i)
    // s-contiguous reduce for dpstate/dD_wei/dU_att/dc_att
    hipLaunchKernelGGL(cond_attn_bwd_dalpha, dim3(B, cdiv_i(Ts, 4)),
                       dim3(256), 0, stream,
                       (const bf16_t*)ctx_bf.data_ptr(),
                       dctxpre_f32.data_ptr<float>(),
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       daccA.data_ptr<float>(), mt,
                       dalpha_p ? dalpha_p + (long)t * B * Ts : nullptr,
                       dal_buf.data_ptr<float>(), dot_buf.data_ptr<float>(),
                       B, Ts, C);
    hipLaunchKernelGGL(cond_attn_bwd_dot, dim3(B), dim3(256), 0, stream,
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       dal_buf.data_ptr<float>(), dot_buf.data_ptr<float>(),
                       B, Ts);
    hipLaunchKernelGGL(cond_attn_bwd_scatter,
                       dim3(B, cdiv_i(Ts, 256), A >= 32 ? 8 : (A >= 16 ? 4 : 1)),
                       dim3(256), 0, stream,
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       dal_buf.data_ptr<float>(), dot_buf.data_ptr<float>(),
                       pctx.data_ptr<float>(),
                       pstate_all.data_ptr<float>() + (long)t * B * A,
                       accA_used.data_ptr<float>() + (long)t * B * Ts,
                       Dwei.data_ptr<float>(), Uatt.data_ptr<float>(),
                       daccA.data_ptr<float>(), dpctx_acc.data_ptr<float>(),
                       (bf16_t*)pc_buf.data_ptr(), B, Ts, A, Tpad8);
    const int RSCH = std::max(1, std::min(12, Ts / 64));
    hipLaunchKernelGGL(cond_attn_bwd_reduce, dim3(B, RSCH), dim3(256),
                       2 * ((Ts + RSCH - 1) / RSCH) * sizeof(float), stream,
                       alphas_all.data_ptr<float>() + (long)t * B * Ts,
                       dal_buf.data_ptr<float>(), dot_buf.data_ptr<float>(),
                       accA_used.data_ptr<float>() + (long)t * B * Ts,
                       (const bf16_t*)pc_buf.data_ptr(),
                       Uatt.data_ptr<float>(),
                       dpstate_all.data_ptr<float>() + (long)t * B * A,
                       gdDwei.data_ptr<float>(), gdUatt.data_ptr<float>(),
                       gdcatt.data_ptr<float>(), B, Ts, A, Apad32, Tpad8,
                       RSCH);
    // b6a: dh1 += dpstate @ W_att^T — A fragments converted from the
    // fp32 dpstate in-register (no cond_dpstate_cast pass)
    hipLaunchKernelGGL(cond_dh1_att_gemm, dim3(ngrpH), dim3(384), 0, stream,
                       dpstate_all.data_ptr<float>() + (long)t * B * A,
                       (const bf16_t*)WattB.data_ptr(),
                       dh1_buf.data_ptr<float>(), B, H, A, Apad32);
    // b8: GRU_2 pointwise (dh1 -> gate preact grads)
    hipLaunchKernelGGL(nats_gru_step_bwd_pointwise, dim3(pwH), dim3(256), 0,
                       stream, dh1_buf.data_ptr<float>(), nullptr,
                       (const bf16_t*)saved2.data_ptr() + (long)t * B * 3 * H,
                       (const bf16_t*)yc.data_ptr() + (long)t * B * H, h2prev,
                       mt, (bf16_t*)dstep2.data_ptr(), K3Hpad,
                       ddirect2.data_ptr<float>(),
                       (bf16_t*)dpre2_all.data_ptr() + (long)t * B * 4 * H, B,
                       H);
    // b9: dh_{t-1} = [dpr1|dpu1|dpxl1] @ [U|Ux]^T + passthrough, split
    // over K halves written as two partials (summed by the next step's
    // GRU_1 pointwise — no atomics, no zeroing)
    hipLaunchKernelGGL(cond_dh_carry_gemm_split, dim3(ngrpH, 2), dim3(384),
                       0, stream, (const bf16_t*)dstep2.data_ptr(),
                       (const bf16_t*)U2cat.data_ptr(),
                       ddirect2.data_ptr<float>(), dh_carry.data_ptr<float>(),
                       B, H, K3Hpad);
  }
  HIP_CHECK(hipGetLastError());
  auto dh_carry_sum = dh_carry[0] + dh_carry[1];
  return {dpre1_all, dpre2_all, dctxpre_all, gdUcon, dpstate_all,
          dpctx_acc, gdDwei, gdUatt, gdcatt, dh_carry_sum, daccC, daccA,
          gdWcon};
}
