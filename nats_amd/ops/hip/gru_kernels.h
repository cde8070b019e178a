// Shared GRU kernel declarations (defined in gru_scan.hip, reused by the
// conditional-GRU decoder in cond_gru.hip — GRU_2's per-step math is
// identical to the encoder cell, nats.py:503-519 vs 336-356).
#pragma once

#include "common.h"

// forward step: fused [h@U_r | h@U_u | h@Ux] MFMA + gates/mask pointwise.
// ld_bfout = row stride of the bf16 h output (lets the decoder write h1
// into the [h1|ctx] packed GRU_1 operand buffer).
__global__ __launch_bounds__(384) void nats_gru_step_fwd(
    const bf16_t* __restrict__ h_bf, const float* __restrict__ h_prev,
    const bf16_t* __restrict__ Upk, const bf16_t* __restrict__ xg_t,
    const bf16_t* __restrict__ xc_t, const float* __restrict__ mask_t,
    float* __restrict__ h_out, bf16_t* __restrict__ h_bf_out, int ld_bfout,
    bf16_t* __restrict__ saved_t, int B, int H, int Hpad);

__global__ __launch_bounds__(384) void nats_gru2_gemm_splitk(const bf16_t* h_bf, const bf16_t* Upk,
                                      float* part, int Hpad);

__global__ void nats_gru2_step_pointwise(
    const float* part, int KS, const float* h_prev, const bf16_t* xg_t,
    const bf16_t* xc_t, const float* mask_t, float* h_out, bf16_t* h_bf_out,
    int ld_bfout, bf16_t* saved_t, int B, int H, int Hpad);

// backward pointwise: dh -> gate preactivation grads. dh_out_t may be null.
__global__ void nats_gru_step_bwd_pointwise(
    const float* __restrict__ dh_buf, const float* __restrict__ dh_out_t,
    const bf16_t* __restrict__ saved_t, const bf16_t* __restrict__ xc_t,
    const float* __restrict__ h_prev, const float* __restrict__ mask_t,
    bf16_t* __restrict__ dstep, int ld_dstep, float* __restrict__ ddirect,
    bf16_t* __restrict__ dpre_t, int B, int H);

// backward recurrent GEMM: out[b,i] = ddirect[b,i] + dstep[b,:] @ Wt[i,:]
// (generic over output width "H" = rows of Wt; K = Kpad).
__global__ __launch_bounds__(384) void nats_gru_step_bwd_gemm(
    const bf16_t* __restrict__ dstep, const bf16_t* __restrict__ Wt,
    const float* __restrict__ ddirect, float* __restrict__ out, int B, int H,
    int Kpad);

// Per-direction argument sets for the bidirectional encoder kernels
// (both directions are independent — one launch covers both via
// blockIdx.y, halving launch count and overlapping the work).
struct GruFwdArgs {
  const bf16_t* h_bf;
  const float* h_prev;
  const bf16_t* Upk;
  const bf16_t* xg_t;
  const bf16_t* xc_t;
  const float* mask_t;
  float* h_out;
  bf16_t* h_bf_out;
  bf16_t* saved_t;
};

struct GruBwdArgs {
  const bf16_t* dstep_in;   // dstep written at step t+1 (zeros at t=T-1)
  const float* ddirect_in;  // passthrough written at step t+1
  const float* dh_out_t;    // upstream grad at t
  const bf16_t* saved_t;
  const bf16_t* xc_t;
  const float* h_prev;
  const float* mask_t;
  bf16_t* dstep_out;
  float* ddirect_out;
  bf16_t* dpre_t;
};

__global__ __launch_bounds__(384) void nats_gru_step_fwd_bidir(
    GruFwdArgs a0, GruFwdArgs a1, int ld_bfout, int B, int H, int Hpad);

// fused backward step: phase 1 recomputes dh for this WG's columns from
// the PREVIOUS (t+1) step's dstep/ddirect via the recurrent GEMM, phase 2
// does the pointwise gate backward on the same columns (column-local).
__global__ __launch_bounds__(384) void nats_gru_step_bwd_fused_bidir(
    GruBwdArgs a0, GruBwdArgs a1, const bf16_t* Ubwd0, const bf16_t* Ubwd1,
    int B, int H, int Kpad);


