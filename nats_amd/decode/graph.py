"""hipGraph-captured beam-decode step.

The reference's beam loop calls a compiled f_next per step (nats.py:960);
our f_next is itself a chain of ~15 kernel launches (embedding, fused
cond-GRU step, readout GEMMs, softmax). Capturing that chain into a
hipGraph (torch.cuda.CUDAGraph on ROCm) collapses per-step launch
overhead into one graph replay.

Shapes are static: the stepper is captured for a fixed (beam k, source
length Ts); the beam keeps k rows alive (dead/padded rows carry copies of
row 0, harmless — callers slice [:live_k]). Host-side feedback between
replays (parent gather after rank selection) writes the static input
buffers with index copies.
"""

import torch


import weakref

_STEPPER_CACHE = weakref.WeakKeyDictionary()


def _weights_version(model):
    """Monotone tag over the decoder weights: in-place optimizer updates
    bump tensor _version, which must invalidate captured graphs (packed
    weights are baked into the graph at capture — ops/cond_gru.py
    _step_packed). The fused adadelta kernel updates through raw pointers
    instead, so it bumps _nats_update_epoch (ops/optim.py) — include both."""
    return tuple((p._version, getattr(p, "_nats_update_epoch", 0))
                 for p in model.P.values())


def get_stepper(model, ctx0, pctx0, k):
    """Cached stepper per (model, Ts, k): the captured graph is reused
    across sentences of the same source length — only the static context
    buffers are refreshed. Any weight update forces a fresh capture."""
    ver = _weights_version(model)
    per_model = _STEPPER_CACHE.setdefault(model, {})
    if per_model.get("__ver__") != ver:
        per_model.clear()
        per_model["__ver__"] = ver
    key = (int(ctx0.shape[0]), int(k))
    st = per_model.get(key)
    if st is None:
        plain_keys = [kk for kk in per_model
                      if isinstance(kk, tuple) and kk[0] != "batched"
                      and kk != "__ver__"]
        if len(plain_keys) >= 8:
            per_model.pop(plain_keys[0], None)
        st = GraphDecodeStepper(model, ctx0, pctx0, k)
        per_model[key] = st
    else:
        st.set_context(ctx0, pctx0)
    return st


def get_batched_stepper(model, ctx_pad, ctx_mask, pctx_pad, R):
    """Cached BatchedGraphStepper per (model, Ts, S, R) — reused across
    same-shaped micro-batches (serving); invalidated on weight updates."""
    ver = _weights_version(model)
    per_model = _STEPPER_CACHE.setdefault(model, {})
    if per_model.get("__ver__") != ver:
        per_model.clear()
        per_model["__ver__"] = ver
    key = ("batched", int(ctx_pad.shape[0]), int(ctx_pad.shape[1]), int(R))
    st = per_model.get(key)
    if st is None:
        # bound capture memory: keep the most recent shapes only
        batched_keys = [k for k in per_model
                        if isinstance(k, tuple) and k[0] == "batched"]
        if len(batched_keys) >= 8:
            per_model.pop(batched_keys[0], None)
        st = BatchedGraphStepper(model, ctx_pad, ctx_mask, pctx_pad, R)
        per_model[key] = st
    else:
        st.set_context(ctx_pad, ctx_mask, pctx_pad)
    return st


class BatchedGraphStepper:
    """hipGraph-captured f_next for the batched multi-sentence beam
    (decode/batched.py — the production serving path).

    Static row budget R (= S*k): live rows occupy a prefix, dead rows are
    padded with row 0's state (harmless — callers slice [:rows]). The
    per-step context gather ctx_pad[:, sent_idx] happens INSIDE the
    graph from a static sent_idx buffer, so hypothesis->sentence routing
    changes never force recapture."""

    def __init__(self, model, ctx_pad, ctx_mask, pctx_pad, R):
        self.model = model
        self.R = R
        device = ctx_pad.device
        Ts, S, C = ctx_pad.shape
        H = model.options["dim"]
        # own the static buffers (a caller's tensor may alias freed memory
        # by the next micro-batch)
        self.ctx_pad = ctx_pad.contiguous().clone()
        self.ctx_mask = ctx_mask.float().contiguous().clone()
        self.pctx_pad = pctx_pad.contiguous().clone()
        self.sent_idx = torch.zeros(R, dtype=torch.int64, device=device)
        self.y_in = torch.zeros(R, dtype=torch.int64, device=device)
        self.state_in = torch.zeros(R, H, device=device)
        self.accC_in = torch.zeros(R, C, device=device)
        self.accA_in = torch.zeros(R, Ts, device=device)
        self.graph = None
        self.outs = None

    def set_context(self, ctx_pad, ctx_mask, pctx_pad):
        self.ctx_pad.copy_(ctx_pad)
        self.ctx_mask.copy_(ctx_mask)
        self.pctx_pad.copy_(pctx_pad)

    def _run(self):
        ctx_b = self.ctx_pad.index_select(1, self.sent_idx)
        cmask_b = self.ctx_mask.index_select(1, self.sent_idx)
        pctx_b = self.pctx_pad.index_select(1, self.sent_idx)
        return self.model.f_next(self.y_in, ctx_b, cmask_b, pctx_b,
                                 self.state_in, self.accC_in, self.accA_in,
                                 sample_draw=False)

    def capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._run()
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            outs = self._run()
            self.outs = tuple(o.clone() if isinstance(o, torch.Tensor) else o
                              for o in outs)
        return self

    def step(self, sent_idx, y, state, acc_ctx, acc_alpha):
        """rows = len(y) (<= R) live hypothesis rows; returns (probs, h2,
        alpha, ctx_t, accC, accA), each R rows — callers slice [:rows]."""
        rows = y.shape[0]
        self.sent_idx[:rows] = sent_idx
        self.y_in[:rows] = y
        self.state_in[:rows] = state
        self.accC_in[:rows] = acc_ctx
        self.accA_in[:rows] = acc_alpha
        if rows < self.R:
            self.sent_idx[rows:] = sent_idx[0]
            self.y_in[rows:] = y[0]
            self.state_in[rows:] = state[0]
            self.accC_in[rows:] = acc_ctx[0]
            self.accA_in[rows:] = acc_alpha[0]
        if self.graph is None:
            self.capture()
        self.graph.replay()
        probs, _, h2, alpha, ctx_t, accC, accA = self.outs
        return probs, h2, alpha, ctx_t, accC, accA


class GraphDecodeStepper:
    """Replayable f_next for beam search (argmax/beam mode, no sampling)."""

    def __init__(self, model, ctx0, pctx0, k):
        """ctx0 (Ts,1,C), pctx0 (Ts,1,A) from f_init/project_ctx."""
        self.model = model
        self.k = k
        device = ctx0.device
        Ts, _, C = ctx0.shape
        H = model.options["dim"]
        self.ctx = ctx0.expand(Ts, k, C).contiguous()
        self.pctx = pctx0.expand(Ts, k, pctx0.shape[2]).contiguous()
        self.y_in = torch.zeros(k, dtype=torch.int64, device=device)
        self.state_in = torch.zeros(k, H, device=device)
        self.accC_in = torch.zeros(k, C, device=device)
        self.accA_in = torch.zeros(k, Ts, device=device)
        self.graph = None
        self.outs = None

    def set_context(self, ctx0, pctx0):
        """Refresh the static context buffers for a new source sequence
        (same Ts) without recapturing the graph."""
        Ts, _, C = ctx0.shape
        self.ctx.copy_(ctx0.expand(Ts, self.k, C))
        self.pctx.copy_(pctx0.expand(Ts, self.k, pctx0.shape[2]))

    def _run(self):
        return self.model.f_next(self.y_in, self.ctx, None, self.pctx,
                                 self.state_in, self.accC_in, self.accA_in,
                                 sample_draw=False)

    def capture(self):
        # warmup (allocator + kernels) on a side stream, then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._run()
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            outs = self._run()
            # keep stable output storage across replays
            self.outs = tuple(o.clone() if isinstance(o, torch.Tensor) else o
                              for o in outs)
        return self

    def step(self, y, state, acc_ctx, acc_alpha):
        """Run one decode step for `live` rows (<= k); returns
        (probs, state, alpha, ctx_t, acc_ctx, acc_alpha) each k rows —
        caller slices [:live]."""
        live = y.shape[0]
        self.y_in[:live] = y
        if live < self.k:
            self.y_in[live:] = y[0]
        self.state_in[:live] = state
        self.accC_in[:live] = acc_ctx
        self.accA_in[:live] = acc_alpha
        if live < self.k:
            self.state_in[live:] = state[0]
            self.accC_in[live:] = acc_ctx[0]
            self.accA_in[live:] = acc_alpha[0]
        if self.graph is None:
            self.capture()
        self.graph.replay()
        probs, _, h2, alpha, ctx_t, accC, accA = self.outs
        return probs, h2, alpha, ctx_t, accC, accA
