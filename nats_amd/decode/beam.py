"""Beam search / stochastic sampling with decode-time distraction rerank.

Semantics mirror gen_sample (nats.py:879-1076) exactly, including:
  * UNK suppression via probs[:,1]=1e-20 when use_unk=False (nats.py:973-974),
  * candidate scores = hyp_scores - log p, top (k - dead_k) by argsort,
  * the distraction rerank (ii>0, any lambda>0): per live hypothesis,
    penalties from its own history —
      -kl_factor * min_t KL(alpha_t || alpha_cur)     (scipy entropy conv.)
      +ctx_factor * max_t cosine_dist(ctx_t, ctx_cur)
      +state_factor * max_t cosine_dist(s_t, s_cur)
    penalties STEER SELECTION ONLY: the accumulated hypothesis costs are
    taken from the UN-reranked cand_flat (nats.py:1004),
  * EOS (id 0) retirement, live/dead bookkeeping, flush of still-live
    hypotheses at maxlen (nats.py:1068-1074),
  * stochastic mode adds the sampled word's PROBABILITY (not log-prob) to
    the running score (nats.py:969) — reproduced verbatim.

Length normalisation is caller-side (gen driver / train-loop sampling).
"""

import numpy
import torch


def _kl_div(p, q, eps=0.0):
    """scipy.stats.entropy(p, q): normalises both, sum p*log(p/q)."""
    p = numpy.asarray(p, dtype="float64")
    q = numpy.asarray(q, dtype="float64")
    p = p / p.sum()
    q = q / q.sum()
    with numpy.errstate(divide="ignore", invalid="ignore"):
        v = numpy.where(p > 0, p * numpy.log(p / q), 0.0)
    return float(numpy.sum(v))


def _cosine_dist(u, v):
    """scipy.spatial.distance.cosine: 1 - u.v/(|u||v|)."""
    u = numpy.asarray(u, dtype="float64")
    v = numpy.asarray(v, dtype="float64")
    denom = numpy.linalg.norm(u) * numpy.linalg.norm(v)
    if denom == 0:
        return 0.0
    return float(1.0 - numpy.dot(u, v) / denom)


def distraction_penalties(hyp_dec_alphas, hyp_ctxs, hyp_states_dis,
                          dec_alphas, ctxs, next_state,
                          kl_factor, ctx_factor, state_factor):
    """Per-hypothesis rerank penalties (nats.py:981-999).

    History lists are per live hypothesis; dec_alphas/ctxs/next_state are
    the CURRENT step outputs (numpy, rows = live hyps). Returns three
    (live_k,) float32 arrays.
    """
    live_k = len(hyp_dec_alphas)
    alphac_score = numpy.zeros((live_k,), dtype="float32")
    ctxs_score = numpy.zeros((live_k,), dtype="float32")
    state_score = numpy.zeros((live_k,), dtype="float32")
    for idx in range(live_k):
        alpha_reg, ctx_reg, state_reg = [], [], []
        for alphas, _ctxs, states in zip(hyp_dec_alphas[idx], hyp_ctxs[idx],
                                         hyp_states_dis[idx]):
            alpha_reg.append(_kl_div(alphas, dec_alphas[idx, :]))
            ctx_reg.append(_cosine_dist(_ctxs, ctxs[idx, :]))
            state_reg.append(_cosine_dist(states, next_state[idx, :]))
        if alpha_reg:
            alphac_score[idx] = -kl_factor * min(alpha_reg)
            ctxs_score[idx] = ctx_factor * max(ctx_reg)
            state_score[idx] = state_factor * max(state_reg)
    return alphac_score, ctxs_score, state_score


def distraction_penalties_gpu(hist_a, hist_c, hist_s, cur_a, cur_c, cur_s,
                              kl_factor, ctx_factor, state_factor):
    """Batched device-side rerank penalties (same math as
    distraction_penalties, vectorised over (history step, hypothesis) —
    the host scipy loops cost ~3x decode throughput at beam 10).

    hist_* are (n_hist, live_k, dim) fp32 device tensors; cur_* (live_k,
    dim). Returns a (live_k,) fp32 tensor of summed penalties."""
    from nats_amd.ops import _hip_ext
    ext = _hip_ext()
    if ext is not None and hist_a.is_cuda and hasattr(ext, "rerank_penalties"):
        return ext.rerank_penalties(
            hist_a.contiguous(), hist_c.contiguous(), hist_s.contiguous(),
            cur_a.contiguous(), cur_c.contiguous(), cur_s.contiguous(),
            kl_factor, ctx_factor, state_factor)
    p = hist_a / hist_a.sum(-1, keepdim=True)
    q = (cur_a / cur_a.sum(-1, keepdim=True)).unsqueeze(0)
    logterm = torch.where(p > 0, p * (torch.log(p) - torch.log(q)),
                          torch.zeros_like(p))
    a_pen = -kl_factor * logterm.sum(-1).min(dim=0).values

    def cosd(h, c):
        num = (h * c.unsqueeze(0)).sum(-1)
        den = h.norm(dim=-1) * c.norm(dim=-1).unsqueeze(0)
        d = 1.0 - num / den
        return torch.where(den == 0, torch.zeros_like(d), d)

    c_pen = ctx_factor * cosd(hist_c, cur_c).max(dim=0).values
    s_pen = state_factor * cosd(hist_s, cur_s).max(dim=0).values
    return a_pen + c_pen + s_pen


@torch.no_grad()
def gen_sample(model, x, k=1, maxlen=30, stochastic=True, argmax=False,
               use_unk=False, kl_factor=0.0, ctx_factor=0.0, state_factor=0.0,
               generator=None, use_graph=False):
    """Generate one summary by beam search or stochastic sampling.

    model: NatsModel; x: (T,1) int64 tensor on the model's device.
    Returns (sample, sample_score, sample_dec_alphas) exactly like the
    reference: beam mode gives lists (one per finished hypothesis);
    stochastic mode gives a flat token list and a scalar score.
    """
    if k > 1:
        assert not stochastic, "Beam search does not support stochastic sampling"

    device = x.device
    sample = []
    sample_score = [] if not stochastic else 0.0
    sample_dec_alphas = []

    live_k = 1
    dead_k = 0

    hyp_samples = [[]] * live_k
    hyp_scores = numpy.zeros(live_k).astype("float32")
    hyp_dec_alphas = [[]] * live_k
    hyp_ctxs = [[]] * live_k
    hyp_states_dis = [[]] * live_k

    init_state, ctx0 = model.f_init(x)
    pctx0 = model.project_ctx(ctx0)
    next_state = init_state                               # (1,H)
    next_w = torch.full((1,), -1, dtype=torch.int64, device=device)
    C = ctx0.shape[2]
    Ts = ctx0.shape[0]
    acc_ctx = torch.zeros((live_k, C), device=device, dtype=ctx0.dtype)
    acc_alpha = torch.zeros((live_k, Ts), device=device, dtype=ctx0.dtype)

    # hipGraph-captured decode step (beam mode on GPU): the whole f_next
    # kernel chain replays as one graph per step
    stepper = None
    if use_graph and not stochastic and device.type == "cuda":
        from .graph import get_stepper
        stepper = get_stepper(model, ctx0.float(), pctx0.float(), k)

    any_lambda = (kl_factor > 0.0 or ctx_factor > 0.0 or state_factor > 0.0)
    # device-resident rerank histories (GPU path); the numpy histories are
    # still kept — they are the returned alignment output
    gpu_rerank = any_lambda and device.type == "cuda"
    hist_a_dev = hist_c_dev = hist_s_dev = None

    for ii in range(maxlen):
        if stepper is not None:
            probs_k, h2_k, alpha_k, ctxs_k, accC_k, accA_k = stepper.step(
                next_w, next_state.float(), acc_ctx.float(),
                acc_alpha.float())
            probs = probs_k[:live_k]
            w_sample = probs.argmax(dim=-1)
            next_state = h2_k[:live_k]
            dec_alphas = alpha_k[:live_k]
            ctxs = ctxs_k[:live_k]
            acc_ctx = accC_k[:live_k]
            acc_alpha = accA_k[:live_k]
        else:
            ctx = ctx0.expand(Ts, live_k, C)
            pctx = pctx0.expand(Ts, live_k, pctx0.shape[2])
            probs, w_sample, next_state, dec_alphas, ctxs, acc_ctx, \
                acc_alpha = model.f_next(
                    next_w, ctx, None, pctx, next_state, acc_ctx, acc_alpha,
                    generator=generator, sample_draw=stochastic)

        if stochastic:
            if argmax:
                nw = int(probs[0].argmax())
            else:
                nw = int(w_sample[0])
            sample.append(nw)
            sample_score += float(probs[0, nw])
            if nw == 0:
                break
            # The reference feeds the SAMPLED word back into f_next
            # unconditionally (next_w = ret[1], nats.py:961); argmax only
            # selects which word is RECORDED (nats.py:965-966).
            next_w = w_sample[:1].to(torch.int64).reshape(1)
            continue

        next_p = probs.float().cpu().numpy()
        if not use_unk:
            next_p[:, 1] = 1e-20

        cand_scores = hyp_scores[:, None] - numpy.log(next_p)
        cand_flat = cand_scores.flatten()
        ranks_flat = cand_flat.argsort()[: (k - dead_k)]

        if ii > 0 and any_lambda:
            if gpu_rerank:
                pen = distraction_penalties_gpu(
                    hist_a_dev, hist_c_dev, hist_s_dev, dec_alphas.float(),
                    ctxs.float(), next_state.float(), kl_factor, ctx_factor,
                    state_factor).cpu().numpy()
                new_cand = cand_scores + pen[:, None]
            else:
                da = dec_alphas.float().cpu().numpy()
                cs = ctxs.float().cpu().numpy()
                ns = next_state.float().cpu().numpy()
                a_s, c_s, s_s = distraction_penalties(
                    hyp_dec_alphas, hyp_ctxs, hyp_states_dis, da, cs, ns,
                    kl_factor, ctx_factor, state_factor)
                new_cand = (cand_scores + a_s[:, None] + c_s[:, None] +
                            s_s[:, None])
            ranks_flat = new_cand.flatten().argsort()[: (k - dead_k)]

        voc_size = next_p.shape[1]
        trans_indices = ranks_flat // voc_size
        word_indices = ranks_flat % voc_size
        costs = cand_flat[ranks_flat]   # UN-reranked costs (nats.py:1004)

        da = dec_alphas.float().cpu().numpy()
        cs = ctxs.float().cpu().numpy()
        ns_np = next_state.float().cpu().numpy()

        new_hyp_samples = []
        new_hyp_scores = numpy.zeros(k - dead_k).astype("float32")
        new_hyp_states = []
        new_hyp_dec_alphas = []
        new_hyp_ctxs = []
        new_hyp_acc_ctx = []
        new_hyp_acc_alpha = []
        new_hyp_states_dis = []
        acc_ctx_np = acc_ctx.float().cpu().numpy()
        acc_alpha_np = acc_alpha.float().cpu().numpy()

        for idx, (ti, wi) in enumerate(zip(trans_indices, word_indices)):
            ti = int(ti)
            new_hyp_samples.append(hyp_samples[ti] + [int(wi)])
            new_hyp_scores[idx] = costs[idx]
            new_hyp_states.append(ns_np[ti].copy())
            new_hyp_dec_alphas.append(hyp_dec_alphas[ti] + [da[ti, :].copy()])
            new_hyp_ctxs.append(hyp_ctxs[ti] + [cs[ti, :].copy()])
            new_hyp_acc_ctx.append(acc_ctx_np[ti].copy())
            new_hyp_acc_alpha.append(acc_alpha_np[ti].copy())
            new_hyp_states_dis.append(hyp_states_dis[ti] + [ns_np[ti, :].copy()])

        if gpu_rerank:
            sel_t = torch.as_tensor(trans_indices.astype(numpy.int64),
                                    device=device)
            cur_a = dec_alphas.float()[sel_t].unsqueeze(0)
            cur_c = ctxs.float()[sel_t].unsqueeze(0)
            cur_s = next_state.float()[sel_t].unsqueeze(0)
            if hist_a_dev is None:
                hist_a_new, hist_c_new, hist_s_new = cur_a, cur_c, cur_s
            else:
                hist_a_new = torch.cat([hist_a_dev[:, sel_t], cur_a], dim=0)
                hist_c_new = torch.cat([hist_c_dev[:, sel_t], cur_c], dim=0)
                hist_s_new = torch.cat([hist_s_dev[:, sel_t], cur_s], dim=0)

        new_live_k = 0
        live_pos = []
        hyp_samples, hyp_scores_l, hyp_states = [], [], []
        hyp_dec_alphas, hyp_ctxs, hyp_states_dis = [], [], []
        hyp_acc_ctx, hyp_acc_alpha = [], []

        for idx in range(len(new_hyp_samples)):
            if new_hyp_samples[idx][-1] == 0:
                sample.append(new_hyp_samples[idx])
                sample_score.append(float(new_hyp_scores[idx]))
                sample_dec_alphas.append(new_hyp_dec_alphas[idx])
                dead_k += 1
            else:
                new_live_k += 1
                live_pos.append(idx)
                hyp_samples.append(new_hyp_samples[idx])
                hyp_scores_l.append(new_hyp_scores[idx])
                hyp_states.append(new_hyp_states[idx])
                hyp_dec_alphas.append(new_hyp_dec_alphas[idx])
                hyp_ctxs.append(new_hyp_ctxs[idx])
                hyp_acc_ctx.append(new_hyp_acc_ctx[idx])
                hyp_acc_alpha.append(new_hyp_acc_alpha[idx])
                hyp_states_dis.append(new_hyp_states_dis[idx])

        hyp_scores = numpy.array(hyp_scores_l, dtype="float32")
        live_k = new_live_k

        if new_live_k < 1:
            break
        if dead_k >= k:
            break

        if gpu_rerank:
            keep = torch.as_tensor(live_pos, dtype=torch.int64, device=device)
            hist_a_dev = hist_a_new[:, keep]
            hist_c_dev = hist_c_new[:, keep]
            hist_s_dev = hist_s_new[:, keep]

        next_w = torch.tensor([w[-1] for w in hyp_samples],
                              dtype=torch.int64, device=device)
        next_state = torch.from_numpy(
            numpy.array(hyp_states, dtype="float32")).to(device)
        acc_ctx = torch.from_numpy(
            numpy.array(hyp_acc_ctx, dtype="float32")).to(device)
        acc_alpha = torch.from_numpy(
            numpy.array(hyp_acc_alpha, dtype="float32")).to(device)

    if not stochastic and live_k > 0:
        for idx in range(live_k):
            sample.append(hyp_samples[idx])
            sample_score.append(float(hyp_scores[idx]))
            sample_dec_alphas.append(hyp_dec_alphas[idx])

    return sample, sample_score, sample_dec_alphas
