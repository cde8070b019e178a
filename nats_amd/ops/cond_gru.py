"""Conditional-GRU decoder ops — HIP path.

The fused decoder-step kernels (GRU_2 + attention + distraction + GRU_1,
SURVEY §2.4 K10-K16) are staged work; until they land the GPU path runs
the eager tensor implementation (still on-GPU through rocBLAS/hipBLASLt,
correct but launch-bound). ops/__init__ routes here so the swap is a
one-line change.
"""

from . import eager


def cond_gru_scan_hip(y_gates, y_cand, mask, init_state, ctx, ctx_mask, pctx,
                      P):
    return eager.cond_gru_scan(y_gates, y_cand, mask, init_state, ctx,
                               ctx_mask, pctx, P)


def cond_gru_step_hip(h_prev, x_g, x_c, ctx, ctx_mask, pctx, acc_ctx,
                      acc_alpha, P):
    return eager.cond_gru_step(h_prev, x_g, x_c, ctx, ctx_mask, pctx,
                               acc_ctx, acc_alpha, P)
