"""The distraction-based seq2seq summarization model.

Re-implements the reference's training graph (build_model, nats.py:658-772)
and sampler graph (build_sampler, nats.py:776-874) as one nn.Module whose
parameters carry the canonical ``.npz`` key names (see models/init.py).

Architecture:
  * bidirectional GRU encoder over source tokens (nats.py:699-713),
  * masked mean pooling -> tanh FF -> decoder init state (nats.py:717-724),
  * conditional GRU decoder: GRU_2 -> attention (+ distraction over
    attention-weight history) -> distraction gate over content vectors ->
    GRU_1 (nats.py:498-572),
  * readout: tanh(h@W + emb@W' + ctx@W'') -> vocab projection -> softmax CE
    (nats.py:753-770).
"""

from collections import OrderedDict

import numpy
import torch
from torch import nn

from .. import ops
from .init import init_params, param_shapes


def default_options(**overrides):
    """Hyperparameter schema mirroring train()'s keyword defaults
    (nats.py:1230-1257). The full dict is persisted with checkpoints."""
    opts = dict(
        dim_word=100,
        dim=1000,
        dim_att=100,
        encoder="gru",
        decoder="gru_cond",
        patience=10,
        max_epochs=5000,
        finish_after=10000000,
        dispFreq=100,
        decay_c=0.0,
        clip_c=-1.0,
        lrate=0.01,
        n_words=100000,
        maxlen=100,
        optimizer="adadelta",
        batch_size=16,
        valid_batch_size=16,
        saveto="model.npz",
        validFreq=1000,
        saveFreq=1000,
        sampleFreq=100,
        datasets=[],
        valid_datasets=[],
        dictionary="",
        use_dropout=False,
        reload_=False,
        verbose=False,
        enc_depth=1,  # stacked bi-GRU encoder layers (framework extension)
    )
    opts.update(overrides)
    return opts


class NatsModel(nn.Module):
    """Distraction-based attentional seq2seq (parameters = .npz schema)."""

    def __init__(self, options, params=None, seed=None):
        super().__init__()
        self.options = dict(options)
        if params is None:
            params = init_params(self.options, seed=seed)
        self.P = nn.ParameterDict()
        for k, shape in param_shapes(self.options):
            v = params[k]
            if isinstance(v, numpy.ndarray):
                t = torch.from_numpy(numpy.ascontiguousarray(v))
            else:
                t = torch.as_tensor(v)
            assert tuple(t.shape) == tuple(shape), (k, tuple(t.shape), shape)
            self.P[k] = nn.Parameter(t)

    # -- checkpoint interop ------------------------------------------------
    def get_params(self):
        """Pull parameters to a numpy OrderedDict (unzip, nats.py:37-41)."""
        out = OrderedDict()
        for k, _ in param_shapes(self.options):
            out[k] = self.P[k].detach().cpu().float().numpy()
        return out

    def set_params(self, params, strict=False):
        """Push a numpy dict into the module (zipp/load_params,
        nats.py:31-33, 81-89): missing keys warn and keep current values."""
        import warnings
        with torch.no_grad():
            for k in self.P:
                if k not in params:
                    if strict:
                        raise KeyError(k)
                    warnings.warn("%s is not in the archive" % k)
                    continue
                self.P[k].copy_(torch.as_tensor(
                    numpy.ascontiguousarray(params[k]), dtype=self.P[k].dtype))

    # -- graph pieces ------------------------------------------------------
    def embed(self, ids, shift=False):
        """Wemb gather; ids (T,B) or (B,) int64. shift=True fuses the
        decoder's shift-right-with-zero-BOS row (nats.py:730-734) into
        the gather kernel (ops/hip/embed.hip)."""
        return ops.embed_gather(self.P["Wemb"], ids, shift=shift)

    def encode(self, x, x_mask=None):
        """Bidirectional encode + decoder init state.

        x (T,B) int64; x_mask (T,B) float or None. When x_mask is None the
        context mean is the plain time-mean — the sampler's behaviour
        (nats.py:810); with a mask it is the masked mean (nats.py:717).
        Returns (ctx (T,B,2H), init_state (B,H)).
        """
        P = self.P
        emb = self.embed(x)
        maskr = x_mask.flip(0) if x_mask is not None else None

        def bi_layer(inp, pf, pr):
            inpr = inp.flip(0)
            xg = inp @ P[pf + "_W"] + P[pf + "_b"]
            xc = inp @ P[pf + "_Wx"] + P[pf + "_bx"]
            xgr = inpr @ P[pr + "_W"] + P[pr + "_b"]
            xcr = inpr @ P[pr + "_Wx"] + P[pr + "_bx"]
            proj, projr = ops.gru_scan_bidir(
                xg, xc, x_mask, P[pf + "_U"], P[pf + "_Ux"],
                xgr, xcr, maskr, P[pr + "_U"], P[pr + "_Ux"])
            return torch.cat([proj, projr.flip(0)], dim=-1)

        ctx = bi_layer(emb, "encoder", "encoder_r")
        for l in range(1, self.options.get("enc_depth", 1)):
            ctx = bi_layer(ctx, "encoder_l%d" % l, "encoder_r_l%d" % l)
        if x_mask is not None:
            ctx_mean = ((ctx * x_mask.unsqueeze(-1)).sum(0) /
                        x_mask.sum(0).unsqueeze(-1))
        else:
            ctx_mean = ctx.mean(0)
        init_state = torch.tanh(ctx_mean @ self.P["ff_state_W"] +
                                self.P["ff_state_b"])
        return ctx, init_state

    def project_ctx(self, ctx):
        """Attention keys, hoisted out of the decode loop (nats.py:493-494)."""
        return ctx @ self.P["decoder_Wc_att"] + self.P["decoder_b_att"]

    def _dec_inputs(self, emb):
        P = self.P
        yg = emb @ P["decoder_W"] + P["decoder_b"]
        yc = emb @ P["decoder_Wx"] + P["decoder_bx"]
        return yg, yc

    def readout_logits(self, h, emb, ctxs):
        """tanh-fused readout + vocab projection (nats.py:753-761).

        All args (...,*) leading dims broadcastable; returns logits (...,V).
        """
        P = self.P
        logit = torch.tanh(h @ P["ff_logit_lstm_W"] + P["ff_logit_lstm_b"] +
                           emb @ P["ff_logit_prev_W"] + P["ff_logit_prev_b"] +
                           ctxs @ P["ff_logit_ctx_W"] + P["ff_logit_ctx_b"])
        return logit @ P["ff_logit_W"] + P["ff_logit_b"]

    def forward(self, x, x_mask, y, y_mask):
        """Training graph -> per-sequence NLL (B,) (nats.py:658-772).

        NOTE (measured negative, round 2): running the decoder input prep
        (embed + hoisted GEMMs) on a side stream concurrent with the
        persistent encoder scans REGRESSED the step 39.2 -> 41.2 ms —
        the scans are latency-critical (one grid barrier per timestep)
        and any co-resident work inflates every barrier interval; the
        idle CUs are not free capacity. Reverted to sequential."""
        ctx, init_state = self.encode(x, x_mask)
        pctx = self.project_ctx(ctx)

        emb_shifted = self.embed(y, shift=True)
        yg, yc = self._dec_inputs(emb_shifted)

        h2s, ctxs, alphas, _, _ = ops.cond_gru_scan(
            yg, yc, y_mask, init_state, ctx, x_mask, pctx, self.P)

        logits = self.readout_logits(h2s, emb_shifted, ctxs)
        T, B, V = logits.shape
        cost = ops.softmax_xent(logits.reshape(T * B, V), y.reshape(-1))
        cost = cost.reshape(T, B)
        return (cost * y_mask).sum(0)

    # -- sampler (f_init / f_next equivalents, nats.py:776-874) ------------
    @torch.no_grad()
    def f_init(self, x, x_mask=None):
        """Encode one (or B) source(s); returns (init_state (B,H),
        ctx (T,B,2H)) like f_init (nats.py:815-817).

        x_mask=None is the reference sampler's unmasked path. A mask makes
        batched multi-sentence decode exact: padded steps are mask-blended
        through the scans and excluded from the init-state mean, so each
        column equals its own unpadded single-sentence encode."""
        ctx, init_state = self.encode(x, x_mask)
        return init_state, ctx

    @torch.no_grad()
    def f_next(self, y_prev, ctx, ctx_mask, pctx, state, acc_ctx, acc_alpha,
               generator=None, sample_draw=True):
        """One decode step (f_next, nats.py:821-871).

        y_prev (B,) int64, -1 marks BOS (embedding = zeros, nats.py:827-829).
        Returns (probs (B,V), sample (B,), state (B,H), alpha (B,Ts),
        ctx_t (B,C), acc_ctx, acc_alpha).
        """
        emb = self.embed(y_prev.clamp_min(0))
        emb = torch.where((y_prev < 0).unsqueeze(1), torch.zeros_like(emb), emb)
        yg, yc = self._dec_inputs(emb)
        h2, ctx_t, alpha_t, acc_ctx, acc_alpha = ops.cond_gru_step(
            state, yg, yc, ctx, ctx_mask, pctx, acc_ctx, acc_alpha, self.P)
        logits = self.readout_logits(h2, emb, ctx_t)
        probs = torch.softmax(logits.float(), dim=-1)
        if not sample_draw:
            sample = probs.argmax(dim=-1)
        elif generator is not None:
            sample = torch.multinomial(probs, 1, generator=generator).squeeze(1)
        else:
            sample = torch.multinomial(probs, 1).squeeze(1)
        return probs, sample, h2, alpha_t, ctx_t, acc_ctx, acc_alpha
