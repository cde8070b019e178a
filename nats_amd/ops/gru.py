"""Autograd wrapper for the fused HIP GRU scan (ops/hip/gru_scan.hip).

Custom BPTT: the sequential recurrences (forward h-chain, reverse dh-chain)
run in the HIP kernels; the time-batched weight/input gradients are single
hipBLASLt GEMMs over the stored per-step preactivation grads — the same
split the reference's Theano graph implied (scan grads + batched dots,
nats.py:1340).
"""


import torch
import torch.nn.functional as F

from . import _hip_ext

JB = 16


def _pad_to(x, rows, cols):
    return F.pad(x, (0, cols - x.shape[1], 0, rows - x.shape[0]))


def pack_fwd_weights(U, Ux):
    """[ngrp*3*16, Hpad] bf16: per group g rows = [r-cols | u-cols | x-cols]
    of output columns [g*16,(g+1)*16), transposed (row = output col).
    On GPU one fused kernel (ops/hip/pack.hip) replaces the 6-op torch
    chain (pad x3, stack, permute-contiguous, cast) — these packs run
    every training step (weights change every update)."""
    if U.is_cuda and U.dtype == torch.float32:
        ext = _hip_ext()
        if ext is not None:
            return ext.pack_fwd_weights(U, Ux)
    H = Ux.shape[1]
    ngrp = (H + JB - 1) // JB
    rows = ngrp * JB
    Hpad = ((H + 31) // 32) * 32
    Ut = _pad_to(U[:, :H].t(), rows, Hpad)
    Uu = _pad_to(U[:, H:].t(), rows, Hpad)
    Uxt = _pad_to(Ux.t(), rows, Hpad)
    P = torch.stack([Ut, Uu, Uxt], dim=0)            # (3, rows, Hpad)
    P = P.view(3, ngrp, JB, Hpad).permute(1, 0, 2, 3).contiguous()
    return P.view(ngrp * 3 * JB, Hpad).to(torch.bfloat16).contiguous()


def pack_bwd_weights(U, Ux):
    """[ngrp*16, K3pad] bf16: row i = [U[i, :2H] | Ux[i, :]] zero-padded."""
    H = Ux.shape[1]
    ngrp = (H + JB - 1) // JB
    rows = ngrp * JB
    K3 = 3 * H
    K3pad = ((K3 + 31) // 32) * 32
    if U.is_cuda and U.dtype == torch.float32:
        ext = _hip_ext()
        if ext is not None:
            return ext.pack_cat2(U, Ux, rows, K3pad)
    cat = torch.cat([U, Ux], dim=1)                   # (H, 3H)
    return _pad_to(cat, rows, K3pad).to(torch.bfloat16).contiguous()


class GRUScanFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xg, xc, mask, U, Ux, h0):
        ext = _hip_ext()
        xg_dtype = xg.dtype
        xg = xg.to(torch.bfloat16).contiguous()
        xc = xc.to(torch.bfloat16).contiguous()
        Upk = pack_fwd_weights(U, Ux)
        h_all, saved = ext.gru_scan_fwd(xg, xc, mask, Upk, h0)
        ctx.save_for_backward(xc, h_all, saved, U, Ux,
                              mask if mask is not None else torch.empty(0),
                              h0 if h0 is not None else torch.empty(0))
        ctx.xg_dtype = xg_dtype
        return h_all

    @staticmethod
    def backward(ctx, dh_out):
        ext = _hip_ext()
        xc, h_all, saved, U, Ux, mask, h0 = ctx.saved_tensors
        mask = mask if mask.numel() else None
        h0v = h0 if h0.numel() else None
        T, B, H = h_all.shape
        Ubwd = pack_bwd_weights(U, Ux)
        dpre_all, dh0 = ext.gru_scan_bwd(
            dh_out.contiguous().float(), h_all, saved, xc, mask, Ubwd, h0v)
        dxg = dpre_all[..., :2 * H]
        dxc = dpre_all[..., 2 * H:3 * H]
        # time-batched weight grads: dU = hprev^T @ [dpr|dpu],
        # dUx = hprev^T @ dpxl (one GEMM each over all T*B positions)
        if h0v is not None:
            h_prev = torch.cat([h0v.unsqueeze(0).float(), h_all[:-1]], dim=0)
        else:
            h_prev = torch.cat([torch.zeros_like(h_all[:1]), h_all[:-1]],
                               dim=0)
        flat_h = h_prev.reshape(T * B, H).to(torch.bfloat16)
        dU = (flat_h.t() @ dpre_all[..., :2 * H].reshape(T * B, 2 * H)).float()
        dUx = (flat_h.t() @ dpre_all[..., 3 * H:].reshape(T * B, H)).float()
        need_h0 = ctx.needs_input_grad[5]
        return (dxg.to(ctx.xg_dtype), dxc.to(ctx.xg_dtype), None,
                dU.to(U.dtype), dUx.to(Ux.dtype),
                dh0 if need_h0 else None)


def gru_scan_hip(x_gates, x_cand, mask, U, Ux, h0=None):
    if mask is not None:
        mask = mask.float().contiguous()
    return GRUScanFn.apply(x_gates, x_cand, mask, U, Ux, h0)


class BidirGRUScanFn(torch.autograd.Function):
    """Both encoder directions in one fused launch sequence (h0 = 0)."""

    @staticmethod
    def forward(ctx, xg0, xc0, mask0, U0, Ux0, xg1, xc1, mask1, U1, Ux1):
        ext = _hip_ext()
        ctx.xg_dtype = xg0.dtype
        xg0 = xg0.to(torch.bfloat16).contiguous()
        xc0 = xc0.to(torch.bfloat16).contiguous()
        xg1 = xg1.to(torch.bfloat16).contiguous()
        xc1 = xc1.to(torch.bfloat16).contiguous()
        h_all0, saved0, h_all1, saved1 = ext.gru_scan_fwd_bidir(
            xg0, xc0, mask0, pack_fwd_weights(U0, Ux0),
            xg1, xc1, mask1, pack_fwd_weights(U1, Ux1))
        e = torch.empty(0)
        ctx.save_for_backward(xc0, h_all0, saved0, U0, Ux0,
                              mask0 if mask0 is not None else e,
                              xc1, h_all1, saved1, U1, Ux1,
                              mask1 if mask1 is not None else e)
        return h_all0, h_all1

    @staticmethod
    def backward(ctx, dh0_out, dh1_out):
        ext = _hip_ext()
        (xc0, h_all0, saved0, U0, Ux0, mask0,
         xc1, h_all1, saved1, U1, Ux1, mask1) = ctx.saved_tensors
        mask0 = mask0 if mask0.numel() else None
        mask1 = mask1 if mask1.numel() else None
        T, B, H = h_all0.shape
        dpre0, dh00, dpre1, dh01 = ext.gru_scan_bwd_bidir(
            dh0_out.contiguous().float(), h_all0, saved0, xc0, mask0,
            pack_bwd_weights(U0, Ux0),
            dh1_out.contiguous().float(), h_all1, saved1, xc1, mask1,
            pack_bwd_weights(U1, Ux1))

        def wgrads(h_all, dpre):
            h_prev = torch.cat([torch.zeros_like(h_all[:1]), h_all[:-1]], 0)
            fh = h_prev.reshape(T * B, H).to(torch.bfloat16)
            dU = (fh.t() @ dpre[..., :2 * H].reshape(T * B, 2 * H)).float()
            dUx = (fh.t() @ dpre[..., 3 * H:].reshape(T * B, H)).float()
            return dU, dUx

        dU0, dUx0 = wgrads(h_all0, dpre0)
        dU1, dUx1 = wgrads(h_all1, dpre1)
        xt = ctx.xg_dtype
        pt = U0.dtype
        return (dpre0[..., :2 * H].to(xt), dpre0[..., 2 * H:3 * H].to(xt),
                None, dU0.to(pt), dUx0.to(pt),
                dpre1[..., :2 * H].to(xt), dpre1[..., 2 * H:3 * H].to(xt),
                None, dU1.to(pt), dUx1.to(pt))


def gru_scan_bidir_hip(xg0, xc0, mask0, U0, Ux0, xg1, xc1, mask1, U1, Ux1):
    if mask0 is not None:
        mask0 = mask0.float().contiguous()
    if mask1 is not None:
        mask1 = mask1.float().contiguous()
    return BidirGRUScanFn.apply(xg0, xc0, mask0, U0, Ux0, xg1, xc1, mask1,
                                U1, Ux1)
