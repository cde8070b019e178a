"""Autograd wrapper for the fused conditional-GRU decoder
(ops/hip/cond_gru.hip). Semantics = gru_cond_layer (nats.py:498-608).

The sequential per-step chain runs in HIP; every weight gradient that
factors over time is ONE hipBLASLt GEMM here (dU_1, dW_1, dW_att, ...,
plus dWc_att/db_att which flow through the python-side pctx computation
via the returned dpctx).
"""

import torch

from . import _hip_ext
from .gru import pack_fwd_weights, _pad_to

JB = 16


def _ceil(x, m):
    return (x + m - 1) // m * m


def _pack_rows(mat, out_rows, out_cols):
    """[rows,K] -> zero-padded [out_rows, out_cols] bf16 contiguous."""
    if mat.is_cuda and mat.dtype == torch.float32:
        ext = _hip_ext()
        if ext is not None:
            # fused pad(+transpose) kernel; .t() views pack without copy
            if mat.dim() == 2 and not mat.is_contiguous() and                     mat.t().is_contiguous():
                return ext.pack_pad(mat.t(), out_rows, out_cols, True)
            return ext.pack_pad(mat.contiguous(), out_rows, out_cols, False)
    return _pad_to(mat, out_rows, out_cols).to(torch.bfloat16).contiguous()


def pack_gru1_weights(U_1, W_1, Ux_1, Wx_1, Hpad, Cpad):
    """[ngrp*4*16, K1] with per-group rows [r2|u2|pxa|pxb]; K layout
    [h1-block (Hpad) | ctx-block (Cpad)], zero where an output group does
    not consume that operand block. One fused kernel on GPU."""
    if U_1.is_cuda and U_1.dtype == torch.float32:
        ext = _hip_ext()
        if ext is not None:
            return ext.pack_gru1_weights(U_1, W_1, Ux_1, Wx_1, Hpad, Cpad)
    H = Ux_1.shape[0]
    C = W_1.shape[0]
    ngrp = (H + JB - 1) // JB
    rows = ngrp * JB
    K1 = Hpad + Cpad
    P = U_1.new_zeros(4, rows, K1)
    P[0, :H, :H] = U_1[:, :H].t()
    P[0, :H, Hpad:Hpad + C] = W_1[:, :H].t()
    P[1, :H, :H] = U_1[:, H:].t()
    P[1, :H, Hpad:Hpad + C] = W_1[:, H:].t()
    P[2, :H, :H] = Ux_1.t()
    P[3, :H, Hpad:Hpad + C] = Wx_1.t()
    P = P.view(4, ngrp, JB, K1).permute(1, 0, 2, 3).contiguous()
    return P.view(ngrp * 4 * JB, K1).to(torch.bfloat16).contiguous()


class CondGRUScanFn(torch.autograd.Function):
    @staticmethod
    def forward(fctx, yg, yc, mask, init_state, ctx, ctx_mask, pctx,
                U, Ux, U_1, W_1, b_1, Wx_1, Ux_1, bx_1,
                W_att, U_att, c_att, W_con, U_con, D_wei,
                acc_ctx0=None, acc_alpha0=None):
        ext = _hip_ext()
        H = Ux.shape[1]
        C = ctx.shape[2]
        A = U_att.shape[0]
        Hpad, Cpad = _ceil(H, 32), _ceil(C, 32)
        yg = yg.to(torch.bfloat16).contiguous()
        yc = yc.to(torch.bfloat16).contiguous()
        ctx_bf = ctx.to(torch.bfloat16).contiguous()
        pctx_f = pctx.float().contiguous()
        Upk2 = pack_fwd_weights(U, Ux)
        W1pk = pack_gru1_weights(U_1.float(), W_1.float(), Ux_1.float(),
                                 Wx_1.float(), Hpad, Cpad)
        WattPk = _pack_rows(W_att.float().t(), _ceil(A, 16), Hpad)
        vecs = dict(
            b1=b_1.float().contiguous(), bx1=bx_1.float().contiguous(),
            Uatt=U_att.float().reshape(-1).contiguous(),
            Dwei=D_wei.float().reshape(-1).contiguous(),
            Wcon=W_con.float().reshape(-1).contiguous(),
            Ucon=U_con.float().reshape(-1).contiguous())
        outs = ext.cond_gru_fwd(
            yg, yc, mask, init_state, ctx_bf, ctx_mask, pctx_f, Upk2, W1pk,
            WattPk, vecs["b1"], vecs["bx1"], vecs["Uatt"],
            c_att.float().reshape(-1).contiguous(), vecs["Dwei"],
            vecs["Wcon"], vecs["Ucon"], acc_ctx0, acc_alpha0)
        (h2_all, ctxs_all, alphas_all, accC, accA, h1_all, saved2, saved1,
         pstate_all, ctxpre_all, accA_used, accC_used) = outs
        fctx.save_for_backward(
            yc, h1_all, h2_all, ctxs_all, alphas_all, saved2, saved1,
            pstate_all, ctxpre_all, accA_used, accC_used, ctx_bf, pctx_f,
            init_state,
            mask if mask is not None else torch.empty(0),
            U, Ux, U_1, W_1, Wx_1, Ux_1, bx_1, W_att, U_att, D_wei, W_con,
            U_con)
        fctx.in_dtypes = (yg.dtype, ctx.dtype, pctx.dtype, init_state.dtype)
        return h2_all, ctxs_all, alphas_all, accC, accA

    @staticmethod
    def backward(fctx, dh2_all, dctxs_all, dalphas_all, daccC_f, daccA_f):
        ext = _hip_ext()
        (yc, h1_all, h2_all, ctxs_all, alphas_all, saved2, saved1,
         pstate_all, ctxpre_all, accA_used, accC_used, ctx_bf, pctx_f,
         init_state, mask, U, Ux, U_1, W_1, Wx_1, Ux_1, bx_1, W_att, U_att,
         D_wei, W_con, U_con) = fctx.saved_tensors
        mask = mask if mask.numel() else None
        T, B, H = h2_all.shape
        C = ctx_bf.shape[2]
        A = U_att.shape[0]
        from .gru import pack_bwd_weights
        U1cat = pack_bwd_weights(U_1.float(), Ux_1.float())
        U2cat = pack_bwd_weights(U.float(), Ux.float())
        K3Hpad = U1cat.shape[1]
        ext2 = _hip_ext()
        if W_1.is_cuda and ext2 is not None:
            W1cat = ext2.pack_cat2(W_1.float(), Wx_1.float(),
                                   _ceil(C, 16), K3Hpad)
        else:
            W1cat = _pack_rows(
                torch.cat([W_1.float(), Wx_1.float()], dim=1),
                _ceil(C, 16), K3Hpad)
        Apad32 = _ceil(A, 32)
        WattB = _pack_rows(W_att.float(), _ceil(H, 16), Apad32)

        def _opt(t):
            return None if t is None else t

        outs = ext.cond_gru_bwd(
            dh2_all.contiguous().float(), _opt(dctxs_all), _opt(dalphas_all),
            _opt(daccC_f), _opt(daccA_f), yc, h1_all, h2_all, ctxs_all,
            alphas_all, saved2, saved1, pstate_all, ctxpre_all, accA_used,
            accC_used, ctx_bf, pctx_f, init_state, mask, U1cat, W1cat, U2cat,
            WattB, bx_1.float().contiguous(),
            D_wei.float().reshape(-1).contiguous(),
            U_att.float().reshape(-1).contiguous(),
            U_con.float().reshape(-1).contiguous(),
            W_con.float().reshape(-1).contiguous())
        (dpre1_all, dpre2_all, dctxpre_all, gdUcon, dpstate_all,
         dpctx_acc, gdDwei, gdUatt, gdcatt, dh_carry, daccC, daccA,
         gdWcon) = outs

        TB = T * B
        h1f = h1_all.reshape(TB, H).to(torch.bfloat16)
        dpr2u2 = dpre1_all[..., :2 * H].reshape(TB, 2 * H)
        dpx2 = dpre1_all[..., 2 * H:3 * H].reshape(TB, H)
        dpxa = dpre1_all[..., 3 * H:].reshape(TB, H)
        dU_1 = (h1f.t() @ dpr2u2).float()
        dUx_1 = (h1f.t() @ dpxa).float()
        ctxsf = ctxs_all.reshape(TB, C).to(torch.bfloat16)
        dW_1 = (ctxsf.t() @ dpr2u2).float()
        dWx_1 = (ctxsf.t() @ dpx2).float()
        db_1 = dpr2u2.float().sum(0)
        dbx_1 = dpxa.float().sum(0)
        dW_att = (h1f.t() @
                  dpstate_all.reshape(TB, A).to(torch.bfloat16)).float()

        # dU_con/dW_con accumulated in the gate epilogue (one atomic per
        # column per step) — no (T,B,C) re-streams here
        dU_con = gdUcon.reshape(C, 1)
        dW_con = gdWcon.reshape(C, 1)

        # GRU_2 (same algebra as the encoder scan backward)
        dyg = dpre2_all[..., :2 * H]
        dyc = dpre2_all[..., 2 * H:3 * H]
        h2prev = torch.cat([init_state.float().unsqueeze(0), h2_all[:-1]],
                           dim=0).reshape(TB, H).to(torch.bfloat16)
        dU = (h2prev.t() @ dpre2_all[..., :2 * H].reshape(TB, 2 * H)).float()
        dUx = (h2prev.t() @ dpre2_all[..., 3 * H:].reshape(TB, H)).float()

        # encoder-context grad from the attention weighted sum, factored
        # over time: dctx[s,b,c] = sum_t alpha[t,b,s] * dctx_pre[t,b,c]
        dctx_enc = torch.einsum(
            "tbs,tbc->sbc", alphas_all.to(torch.bfloat16),
            dctxpre_all).float()

        ygd, ctxd, pctxd, initd = fctx.in_dtypes
        return (dyg.to(ygd), dyc.to(ygd), None, dh_carry.to(initd),
                dctx_enc.to(ctxd), None, dpctx_acc.to(pctxd),
                dU, dUx, dU_1, dW_1, db_1, dWx_1, dUx_1, dbx_1, dW_att,
                gdUatt.reshape(A, 1), gdcatt.reshape(1),
                dW_con, dU_con, gdDwei.reshape(1, A), None, None)


def _decoder_params(P):
    return (P["decoder_U"], P["decoder_Ux"], P["decoder_U_1"],
            P["decoder_W_1"], P["decoder_b_1"], P["decoder_Wx_1"],
            P["decoder_Ux_1"], P["decoder_bx_1"], P["decoder_W_att"],
            P["decoder_U_att"], P["decoder_c_att"], P["decoder_W_con"],
            P["decoder_U_con"], P["decoder_D_wei"])


def cond_gru_scan_hip(y_gates, y_cand, mask, init_state, ctx, ctx_mask, pctx,
                      P):
    if mask is not None:
        mask = mask.float().contiguous()
    if ctx_mask is not None:
        ctx_mask = ctx_mask.float().contiguous()
    return CondGRUScanFn.apply(y_gates, y_cand, mask, init_state, ctx,
                               ctx_mask, pctx, *_decoder_params(P), None,
                               None)


_STEP_PACK_KEYS = ("decoder_U", "decoder_Ux", "decoder_U_1", "decoder_W_1",
                   "decoder_Ux_1", "decoder_Wx_1", "decoder_W_att",
                   "decoder_b_1", "decoder_bx_1", "decoder_U_att",
                   "decoder_c_att", "decoder_D_wei", "decoder_W_con",
                   "decoder_U_con")


def _step_packed(P):
    """Packed decoder weights for the one-step path, cached on the
    ParameterDict and invalidated by in-place updates (tensor _version):
    decode runs hundreds of steps against frozen weights, and re-packing
    (zero-fill + scatter + transpose + bf16 cast of every decoder matrix)
    per step dominated the non-graph beam loop."""
    vers = tuple((P[k]._version, getattr(P[k], "_nats_update_epoch", 0))
                 for k in _STEP_PACK_KEYS)
    cached = getattr(P, "_nats_step_pack", None)
    if cached is not None and cached[0] == vers:
        return cached[1]
    H = P["decoder_Ux"].shape[1]
    A = P["decoder_U_att"].shape[0]
    C = P["decoder_W_1"].shape[0]
    Hpad, Cpad = _ceil(H, 32), _ceil(C, 32)
    packed = (
        pack_fwd_weights(P["decoder_U"], P["decoder_Ux"]),
        pack_gru1_weights(P["decoder_U_1"].float(), P["decoder_W_1"].float(),
                          P["decoder_Ux_1"].float(),
                          P["decoder_Wx_1"].float(), Hpad, Cpad),
        _pack_rows(P["decoder_W_att"].float().t(), _ceil(A, 16), Hpad),
        P["decoder_b_1"].float().contiguous(),
        P["decoder_bx_1"].float().contiguous(),
        P["decoder_U_att"].float().reshape(-1).contiguous(),
        P["decoder_c_att"].float().reshape(-1).contiguous(),
        P["decoder_D_wei"].float().reshape(-1).contiguous(),
        P["decoder_W_con"].float().reshape(-1).contiguous(),
        P["decoder_U_con"].float().reshape(-1).contiguous(),
    )
    try:
        P._nats_step_pack = (vers, packed)
    except Exception:
        pass
    return packed


@torch.no_grad()
def cond_gru_step_hip(h_prev, x_g, x_c, ctx, ctx_mask, pctx, acc_ctx,
                      acc_alpha, P):
    """One decode step = T=1 fused scan with carried accumulators."""
    ext = _hip_ext()
    (Upk2, W1pk, WattPk, b1, bx1, Uatt, catt, Dwei, Wcon,
     Ucon) = _step_packed(P)
    outs = ext.cond_gru_fwd(
        x_g.unsqueeze(0).to(torch.bfloat16).contiguous(),
        x_c.unsqueeze(0).to(torch.bfloat16).contiguous(),
        None, h_prev, ctx.to(torch.bfloat16).contiguous(),
        ctx_mask.float().contiguous() if ctx_mask is not None else None,
        pctx.float().contiguous(), Upk2, W1pk, WattPk, b1, bx1, Uatt, catt,
        Dwei, Wcon, Ucon,
        acc_ctx.float().contiguous(), acc_alpha.float().contiguous())
    h2_all, ctxs_all, alphas_all, accC, accA = outs[:5]
    return h2_all[0], ctxs_all[0], alphas_all[0], accC, accA
