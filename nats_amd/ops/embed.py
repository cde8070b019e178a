"""Embedding gather with fused decoder shift + scatter-add backward
(ops/hip/embed.hip; SURVEY §2.4 K1/K8, reference op sites
nats.py:700-701 and 730-734)."""

import torch

from . import _hip_ext


class EmbedGatherFn(torch.autograd.Function):
    @staticmethod
    def forward(fctx, Wemb, ids, shift_rows):
        ext = _hip_ext()
        out = ext.embed_gather(Wemb.contiguous(), ids.contiguous(),
                               int(shift_rows))
        fctx.save_for_backward(ids)
        fctx.V = Wemb.shape[0]
        fctx.shift_rows = int(shift_rows)
        return out

    @staticmethod
    def backward(fctx, dout):
        ext = _hip_ext()
        (ids,) = fctx.saved_tensors
        dW = ext.embed_scatter_add(dout.contiguous(), ids.contiguous(),
                                   fctx.V, fctx.shift_rows)
        return dW, None, None


def embed_gather_hip(Wemb, ids, shift=False):
    """out[t] = Wemb[ids[t-1]] if shift else Wemb[ids[t]]; the shifted
    variant writes zeros into the first timestep (BOS row,
    nats.py:730-734). ids (T, B) or (B,)."""
    shift_rows = ids.shape[-1] if (shift and ids.dim() == 2) else 0
    return EmbedGatherFn.apply(Wemb, ids, shift_rows)
