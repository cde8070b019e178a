"""Eager (plain PyTorch) reference implementations of the hot ops.

These are the numerics oracles for the HIP kernels. Each function's
semantics are the Theano graph equations of the reference, cited per-op.
They are deliberately written as straight-line tensor code (no custom
autograd) so torch autograd provides the exact BPTT the reference got from
``tensor.grad`` (nats.py:1340).
"""

import torch


def gru_scan(x_gates, x_cand, mask, U, Ux, h0=None):
    """GRU over time (nats.py:336-372).

    x_gates: (T,B,2H) = x@W + b     (hoisted, nats.py:328-329)
    x_cand:  (T,B,H)  = x@Wx + bx   (nats.py:331-332)
    mask:    (T,B) float or None (None = all ones, nats.py:317-318)
    U: (H,2H), Ux: (H,H); h0: (B,H) or None (zeros, nats.py:360)

    Per step (nats.py:336-356):
      preact = h_{t-1} @ U + x_t ;  r = sigmoid(preact[:, :H]),
      u = sigmoid(preact[:, H:]) ;  hbar = tanh((h_{t-1} @ Ux) * r + xx_t)
      h_t = u * h_{t-1} + (1-u) * hbar ;  h_t = m*h_t + (1-m)*h_{t-1}
    """
    T, B = x_gates.shape[0], x_gates.shape[1]
    H = Ux.shape[1]
    h = h0 if h0 is not None else x_gates.new_zeros(B, H)
    out = []
    for t in range(T):
        preact = torch.sigmoid(h @ U + x_gates[t])
        r, u = preact[:, :H], preact[:, H:]
        hbar = torch.tanh((h @ Ux) * r + x_cand[t])
        h_new = u * h + (1.0 - u) * hbar
        if mask is not None:
            m = mask[t].unsqueeze(1)
            h_new = m * h_new + (1.0 - m) * h
        h = h_new
        out.append(h)
    return torch.stack(out, dim=0)


def _attention(h1, ctx, ctx_mask, pctx, acc_ctx, acc_alpha, P):
    """Bahdanau attention + the two distraction terms (nats.py:527-546).

    Returns (alpha (Ts,B), ctx_t (B,C)). The softmax is computed with
    max-subtraction for stability; because the reference normalises
    exp(e)*mask / sum(exp(e)*mask) (nats.py:537-540), subtracting the
    per-column max before exp yields the identical result.
    """
    pstate = h1 @ P["decoder_W_att"]                       # (B,A)
    pc = pctx + pstate.unsqueeze(0)                        # (Ts,B,A)
    # distraction over the attention-weight history (nats.py:532):
    # (Ts,B,1) @ (1,A) -> (Ts,B,A)
    pc = pc + acc_alpha.t().unsqueeze(-1) * P["decoder_D_wei"].reshape(-1)
    pc = torch.tanh(pc)
    e = (pc @ P["decoder_U_att"]).squeeze(-1) + P["decoder_c_att"]  # (Ts,B)
    e = e - e.max(dim=0, keepdim=True).values
    alpha = torch.exp(e)
    if ctx_mask is not None:
        alpha = alpha * ctx_mask
    alpha = alpha / alpha.sum(dim=0, keepdim=True)
    ctx_t = (ctx * alpha.unsqueeze(-1)).sum(dim=0)         # (B,C)
    # distraction over input content vectors (nats.py:545-546)
    ctx_t = torch.tanh(P["decoder_U_con"].reshape(-1) * ctx_t +
                       acc_ctx * P["decoder_W_con"].reshape(-1))
    return alpha, ctx_t


def cond_gru_step(h_prev, x_g, x_c, ctx, ctx_mask, pctx, acc_ctx, acc_alpha,
                  P, m=None):
    """One conditional-GRU step (nats.py:498-572).

    h_prev (B,H); x_g (B,2H) = emb@W+b; x_c (B,H) = emb@Wx+bx;
    ctx (Ts,B,C); ctx_mask (Ts,B) or None; pctx (Ts,B,A) = ctx@Wc_att+b_att;
    acc_ctx (B,C); acc_alpha (B,Ts); m (B,) mask or None.

    Returns (h2, ctx_t, alpha_T, new_acc_ctx, new_acc_alpha) where
    alpha_T is (B,Ts) (the reference returns alpha.T, nats.py:572).

    NOTE GRU_2 applies sigmoid to the FULL 2H preactivation then slices
    (nats.py:505-510) — mathematically identical to slicing first.
    """
    H = h_prev.shape[1]
    # GRU_2 (nats.py:503-519)
    preact1 = torch.sigmoid(h_prev @ P["decoder_U"] + x_g)
    r1, u1 = preact1[:, :H], preact1[:, H:]
    h1 = torch.tanh((h_prev @ P["decoder_Ux"]) * r1 + x_c)
    h1 = u1 * h_prev + (1.0 - u1) * h1
    if m is not None:
        mm = m.unsqueeze(1)
        h1 = mm * h1 + (1.0 - mm) * h_prev

    alpha, ctx_t = _attention(h1, ctx, ctx_mask, pctx, acc_ctx, acc_alpha, P)

    # GRU_1 (nats.py:551-565)
    preact2 = torch.sigmoid(h1 @ P["decoder_U_1"] + P["decoder_b_1"] +
                            ctx_t @ P["decoder_W_1"])
    r2, u2 = preact2[:, :H], preact2[:, H:]
    h2 = torch.tanh((h1 @ P["decoder_Ux_1"] + P["decoder_bx_1"]) * r2 +
                    ctx_t @ P["decoder_Wx_1"])
    h2 = u2 * h1 + (1.0 - u2) * h2
    if m is not None:
        h2 = mm * h2 + (1.0 - mm) * h1

    # accumulate histories (nats.py:569-570)
    if m is not None:
        new_acc_ctx = mm * ctx_t + acc_ctx
        new_acc_alpha = mm * alpha.t() + acc_alpha
    else:
        new_acc_ctx = ctx_t + acc_ctx
        new_acc_alpha = alpha.t() + acc_alpha
    return h2, ctx_t, alpha.t(), new_acc_ctx, new_acc_alpha


def cond_gru_scan(y_gates, y_cand, mask, init_state, ctx, ctx_mask, pctx, P):
    """Decoder scan over target time (nats.py:596-608).

    Returns (h2s (T,B,H), ctxs (T,B,C), alphas (T,B,Ts),
    acc_ctx (B,C), acc_alpha (B,Ts)) — the final accumulators.
    """
    T, B = y_gates.shape[0], y_gates.shape[1]
    Ts = ctx.shape[0]
    C = ctx.shape[2]
    h = init_state
    acc_ctx = ctx.new_zeros(B, C)
    acc_alpha = ctx.new_zeros(B, Ts)
    h2s, ctxs, alphas = [], [], []
    for t in range(T):
        m = mask[t] if mask is not None else None
        h, ctx_t, alpha_t, acc_ctx, acc_alpha = cond_gru_step(
            h, y_gates[t], y_cand[t], ctx, ctx_mask, pctx, acc_ctx, acc_alpha,
            P, m=m)
        h2s.append(h)
        ctxs.append(ctx_t)
        alphas.append(alpha_t)
    return (torch.stack(h2s, 0), torch.stack(ctxs, 0), torch.stack(alphas, 0),
            acc_ctx, acc_alpha)


def softmax_xent(logits, targets):
    """Per-position negative log-likelihood (nats.py:763-768).

    logits (N,V), targets (N,) -> (N,) in logits.dtype's accumulation
    precision (fp32).
    """
    logp = torch.log_softmax(logits.float(), dim=-1)
    return -logp.gather(1, targets.unsqueeze(1)).squeeze(1)
