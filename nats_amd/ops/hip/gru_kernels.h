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
