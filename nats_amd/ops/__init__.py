"""Op dispatch layer.

Every hot op has two implementations:

  * ``nats_amd.ops.eager`` — plain PyTorch, runs on CPU (and GPU), serves as
    the numerics oracle for the HIP kernels (the Theano graph equations at
    nats.py:336-372, 498-572, 753-770 are the spec),
  * ``nats_amd.ops.hip`` (built from ``nats_amd/ops/hip/*.hip``) — the
    hand-written CDNA4 kernels used on MI355X.

Dispatch rule: CUDA tensors use the HIP kernels; if the extension is not
importable on a GPU machine the op RAISES (no silent eager fallback — set
``NATS_AMD_ALLOW_EAGER_GPU=1`` to override for debugging). CPU tensors use
eager. ``NATS_AMD_FORCE_EAGER=1`` forces eager everywhere (used in tests to
A/B the kernels).
"""

import os

import torch

from . import eager

_HIP = None
_HIP_TRIED = False


def _hip_ext():
    """Import the compiled HIP extension lazily (built in-tree by setup or
    __graft_entry__.build())."""
    global _HIP, _HIP_TRIED
    if not _HIP_TRIED:
        _HIP_TRIED = True
        try:
            from . import hip_ext as m  # thin loader module
            _HIP = m.load()
        except Exception:
            _HIP = None
    return _HIP


def _use_hip(*tensors):
    if os.environ.get("NATS_AMD_FORCE_EAGER"):
        return False
    if not any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor)):
        return False
    ext = _hip_ext()
    if ext is None:
        if os.environ.get("NATS_AMD_ALLOW_EAGER_GPU"):
            return False
        raise RuntimeError(
            "nats_amd HIP extension is not built but tensors are on GPU. "
            "Run `python -c 'import __graft_entry__; __graft_entry__.build()'` "
            "or set NATS_AMD_ALLOW_EAGER_GPU=1 to run the (slow) eager path.")
    return True


def gru_scan(x_gates, x_cand, mask, U, Ux, h0=None):
    """GRU scan over time. Returns hidden states (T, B, H).

    Semantics: nats.py:336-372 (gru_layer step). ``x_gates``/``x_cand`` are
    the hoisted input projections x@W+b (T,B,2H) and x@Wx+bx (T,B,H).
    """
    if _use_hip(x_gates, U):
        from .gru import gru_scan_hip
        B = x_gates.shape[1]
        if B <= MAX_KERNEL_BATCH:
            return gru_scan_hip(x_gates, x_cand, mask, U, Ux, h0)
        return torch.cat([gru_scan_hip(
            x_gates[:, a:b], x_cand[:, a:b],
            mask[:, a:b] if mask is not None else None, U, Ux,
            h0[a:b] if h0 is not None else None)
            for a, b in _batch_chunks(B)], dim=1)
    return eager.gru_scan(x_gates, x_cand, mask, U, Ux, h0)


MAX_KERNEL_BATCH = 32  # MFMA M-tiling of the step kernels (2x 16-row tiles)


def _batch_chunks(B):
    return [(b0, min(b0 + MAX_KERNEL_BATCH, B))
            for b0 in range(0, B, MAX_KERNEL_BATCH)]


_CHUNK_STREAMS = {}


def _chunk_streams(device, n):
    """Per-device side streams for concurrent batch-chunk chains."""
    key = device.index or 0
    pool = _CHUNK_STREAMS.setdefault(key, [])
    while len(pool) < n:
        pool.append(torch.cuda.Stream(device=device))
    return pool[:n]


def gru_scan_bidir(xg0, xc0, mask0, U0, Ux0, xg1, xc1, mask1, U1, Ux1):
    """Both encoder directions (independent scans) in one fused launch
    sequence on GPU; two eager scans on CPU. Returns (h_fwd, h_bwd).

    Batches beyond the kernels' 32-row MFMA tiling are processed in exact
    per-chunk passes (batch rows are independent)."""
    if _use_hip(xg0, U0):
        from .gru import gru_scan_bidir_hip
        B = xg0.shape[1]
        # the persistent scans take up to 64 rows natively (two 32-row
        # chunk jobs concurrent in one launch — same barrier count as 32)
        direct_max = MAX_KERNEL_BATCH
        if _hip_ext() is not None and _hip_ext().gru_persistent_ok(
                U0.shape[0]):
            direct_max = 2 * MAX_KERNEL_BATCH
        if B <= direct_max:
            return gru_scan_bidir_hip(xg0, xc0, mask0, U0, Ux0, xg1, xc1,
                                      mask1, U1, Ux1)
        outs = [gru_scan_bidir_hip(
            xg0[:, a:b], xc0[:, a:b],
            mask0[:, a:b] if mask0 is not None else None, U0, Ux0,
            xg1[:, a:b], xc1[:, a:b],
            mask1[:, a:b] if mask1 is not None else None, U1, Ux1)
            for a, b in _batch_chunks(B)]
        return (torch.cat([o[0] for o in outs], dim=1),
                torch.cat([o[1] for o in outs], dim=1))
    return (eager.gru_scan(xg0, xc0, mask0, U0, Ux0),
            eager.gru_scan(xg1, xc1, mask1, U1, Ux1))


def cond_gru_scan(y_gates, y_cand, mask, init_state, ctx, ctx_mask, pctx, P):
    """Conditional-GRU decoder scan (training). Returns
    (h2s, ctxs, alphas, acc_ctx, acc_alpha).

    Semantics: nats.py:498-608 (gru_cond_layer scan path).
    """
    if _use_hip(y_gates, ctx):
        from .cond_gru import cond_gru_scan_hip
        B = y_gates.shape[1]
        if B <= MAX_KERNEL_BATCH:
            return cond_gru_scan_hip(y_gates, y_cand, mask, init_state, ctx,
                                     ctx_mask, pctx, P)
        # batches beyond the 32-row MFMA tiling run as CONCURRENT chunk
        # chains on separate HIP streams: the per-step decoder kernels
        # occupy a fraction of the chip each, and the chains are
        # independent (autograd replays each backward on its recorded
        # stream, so the reverse chains overlap too)
        chunks = _batch_chunks(B)
        cur = torch.cuda.current_stream()
        streams = _chunk_streams(y_gates.device, len(chunks))
        outs = []
        for (a, b), s in zip(chunks, streams):
            s.wait_stream(cur)
            with torch.cuda.stream(s):
                outs.append(cond_gru_scan_hip(
                    y_gates[:, a:b], y_cand[:, a:b],
                    mask[:, a:b] if mask is not None else None,
                    init_state[a:b], ctx[:, a:b],
                    ctx_mask[:, a:b] if ctx_mask is not None else None,
                    pctx[:, a:b], P))
        for s in streams:
            cur.wait_stream(s)
        for o in outs:
            for t in o:
                t.record_stream(cur)
        return tuple(torch.cat([o[i] for o in outs],
                               dim=1 if i < 3 else 0) for i in range(5))
    return eager.cond_gru_scan(y_gates, y_cand, mask, init_state, ctx,
                               ctx_mask, pctx, P)


def cond_gru_step(h_prev, x_g, x_c, ctx, ctx_mask, pctx, acc_ctx, acc_alpha, P):
    """One decoder step (beam search / sampler one_step path).

    Semantics: nats.py:592-594 with mask == all-ones.
    """
    if _use_hip(h_prev, ctx):
        from .cond_gru import cond_gru_step_hip
        B = h_prev.shape[0]
        if B <= MAX_KERNEL_BATCH:
            return cond_gru_step_hip(h_prev, x_g, x_c, ctx, ctx_mask, pctx,
                                     acc_ctx, acc_alpha, P)
        # beam batches beyond the 32-row tiling: concurrent chunk chains
        # on side streams (doubles the serving micro-batch row budget)
        chunks = _batch_chunks(B)
        cur = torch.cuda.current_stream()
        streams = _chunk_streams(h_prev.device, len(chunks))
        outs = []
        for (a, b), s in zip(chunks, streams):
            s.wait_stream(cur)
            with torch.cuda.stream(s):
                outs.append(cond_gru_step_hip(
                    h_prev[a:b], x_g[a:b], x_c[a:b], ctx[:, a:b],
                    ctx_mask[:, a:b] if ctx_mask is not None else None,
                    pctx[:, a:b], acc_ctx[a:b], acc_alpha[a:b], P))
        for s in streams:
            cur.wait_stream(s)
        for o in outs:
            for t in o:
                t.record_stream(cur)
        return tuple(torch.cat([o[i] for o in outs], dim=0)
                     for i in range(5))
    return eager.cond_gru_step(h_prev, x_g, x_c, ctx, ctx_mask, pctx,
                               acc_ctx, acc_alpha, P)


def embed_gather(Wemb, ids, shift=False):
    """Embedding gather, optionally fused with the decoder's
    shift-right-by-one (BOS row = zeros). Semantics: nats.py:700-701
    (plain) and 730-734 (shifted target embedding).
    """
    if (Wemb.dtype == torch.float32
            and not os.environ.get("NATS_EMBED_EAGER")
            and _use_hip(Wemb)):
        from .embed import embed_gather_hip
        return embed_gather_hip(Wemb, ids, shift=shift)
    emb = Wemb[ids.clamp_min(0)]
    if shift and ids.dim() == 2:
        shifted = torch.zeros_like(emb)
        shifted[1:] = emb[:-1]
        return shifted
    return emb


def softmax_xent(logits, targets):
    """Per-position NLL of a softmax over the vocabulary.

    logits (N, V) float, targets (N,) int64 -> (N,) float32.
    Semantics: nats.py:763-768 (softmax + categorical_crossentropy).
    """
    if _use_hip(logits):
        from .softmax_ce import softmax_xent_hip
        return softmax_xent_hip(logits, targets)
    return eager.softmax_xent(logits, targets)
