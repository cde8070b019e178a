"""Autograd wrapper for the fused vocab softmax+CE HIP kernels."""

import torch

from . import _hip_ext


class SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        ext = _hip_ext()
        logits = logits.to(torch.bfloat16).contiguous()
        targets = targets.contiguous()
        nll, stats = ext.softmax_ce_fwd(logits, targets)
        ctx.save_for_backward(logits, targets, stats)
        return nll

    @staticmethod
    def backward(ctx, dnll):
        ext = _hip_ext()
        logits, targets, stats = ctx.saved_tensors
        dlogits = ext.softmax_ce_bwd(logits, targets, stats,
                                     dnll.contiguous().float())
        return dlogits, None


def softmax_xent_hip(logits, targets):
    return SoftmaxXentFn.apply(logits, targets)
