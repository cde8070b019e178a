"""Batched multi-sentence beam search.

Serving-throughput extension beyond the reference (which decodes one
sentence per process, gen.py:30-45): S sentences' beams share every
decode step as one (S*k)-row batch through the fused decoder kernels
(row budget: the cond-GRU kernels take B <= 32). Per-sentence semantics
are identical to decode.beam.gen_sample — sources are padded to the
longest and masked via ctx_mask, so attention, the distraction terms and
the histories are unaffected by padding.
"""

import numpy
import torch

from .beam import distraction_penalties_gpu


@torch.no_grad()
def gen_sample_batched(model, xs, k=1, maxlen=30, use_unk=False,
                       kl_factor=0.0, ctx_factor=0.0, state_factor=0.0):
    """Beam-decode a list of sources jointly.

    xs: list of (T_i, 1) int64 tensors (same device as model).
    Returns a list of (sample, sample_score, sample_dec_alphas) tuples,
    one per sentence, each shaped exactly like gen_sample's beam output
    (alignment alphas are trimmed to the sentence's own length).
    """
    S = len(xs)
    assert S >= 1
    device = xs[0].device
    any_lambda = (kl_factor > 0.0 or ctx_factor > 0.0 or state_factor > 0.0)

    # ONE batched masked encode for all sentences (identical per-column to
    # each sentence's own unpadded encode — see f_init): pads to the
    # longest source so the encoder scans run S columns per launch instead
    # of S separate T-step launches.
    lens = [int(x.shape[0]) for x in xs]
    Ts = max(lens)
    x_pad = torch.zeros(Ts, S, dtype=torch.int64, device=device)
    x_mask = torch.zeros(Ts, S, device=device)
    for i, x in enumerate(xs):
        x_pad[:lens[i], i] = x[:, 0]
        x_mask[:lens[i], i] = 1.0
    init_all, ctx_all = model.f_init(x_pad, x_mask)
    C = ctx_all.shape[2]
    H = init_all.shape[1]
    inits = [init_all[i:i + 1] for i in range(S)]
    ctxs0 = [ctx_all[:lens[i], i:i + 1] for i in range(S)]

    # per-sentence beam state (numpy bookkeeping identical to gen_sample)
    st = [dict(live=1, dead=0, samples=[[]],
               scores=numpy.zeros(1, dtype="float32"),
               alphas=[[]], ctx_hist=[[]], st_hist=[[]],
               out_samples=[], out_scores=[], out_alphas=[])
          for _ in range(S)]
    # device state per sentence
    dstate = [dict(state=inits[i], w=torch.full((1,), -1, dtype=torch.int64,
                                                device=device),
                   acc_c=torch.zeros(1, C, device=device),
                   acc_a=torch.zeros(1, Ts, device=device),
                   ha=None, hc=None, hs=None)
              for i in range(S)]

    ctx_pad = torch.zeros(Ts, S, C, device=device, dtype=ctxs0[0].dtype)
    ctx_mask_s = torch.zeros(Ts, S, device=device)
    for i, c in enumerate(ctxs0):
        ctx_pad[:lens[i], i] = c[:, 0]
        ctx_mask_s[:lens[i], i] = 1.0
    pctx_pad = model.project_ctx(ctx_pad)

    for ii in range(maxlen):
        rows = []   # (sentence, local row)
        for i in range(S):
            for r in range(st[i]["live"]):
                rows.append((i, r))
        if not rows:
            break
        B = len(rows)
        sent_idx = torch.tensor([i for i, _ in rows], device=device)
        y = torch.cat([dstate[i]["w"] for i in range(S)
                       if st[i]["live"] > 0])
        state = torch.cat([dstate[i]["state"] for i in range(S)
                           if st[i]["live"] > 0])
        acc_c = torch.cat([dstate[i]["acc_c"] for i in range(S)
                           if st[i]["live"] > 0])
        acc_a = torch.cat([dstate[i]["acc_a"] for i in range(S)
                           if st[i]["live"] > 0])
        ctx_b = ctx_pad[:, sent_idx]
        cmask_b = ctx_mask_s[:, sent_idx]
        pctx_b = pctx_pad[:, sent_idx]

        probs, _, h2, alpha, ctx_t, acc_c, acc_a = model.f_next(
            y, ctx_b, cmask_b, pctx_b, state, acc_c, acc_a,
            sample_draw=False)

        probs_np = probs.float().cpu().numpy()
        if not use_unk:
            probs_np[:, 1] = 1e-20
        alpha_np = alpha.float().cpu().numpy()
        ctxt_np = ctx_t.float().cpu().numpy()
        h2_np = h2.float().cpu().numpy()
        accc_np = acc_c.float().cpu().numpy()
        acca_np = acc_a.float().cpu().numpy()

        # per-sentence selection (same math as gen_sample)
        row0 = 0
        for i in range(S):
            live = st[i]["live"]
            if live == 0:
                continue
            sl = slice(row0, row0 + live)
            row0 += live
            p_i = probs_np[sl]
            cand = st[i]["scores"][:, None] - numpy.log(p_i)
            cand_flat = cand.flatten()
            want = k - st[i]["dead"]
            ranks = cand_flat.argsort()[:want]
            if ii > 0 and any_lambda and dstate[i]["ha"] is not None:
                pen = distraction_penalties_gpu(
                    dstate[i]["ha"], dstate[i]["hc"], dstate[i]["hs"],
                    alpha[sl].float(), ctx_t[sl].float(), h2[sl].float(),
                    kl_factor, ctx_factor, state_factor).cpu().numpy()
                ranks = (cand + pen[:, None]).flatten().argsort()[:want]
            V = p_i.shape[1]
            tis = (ranks // V).astype(int)
            wis = (ranks % V).astype(int)
            costs = cand_flat[ranks]

            new_samples, new_scores, new_states = [], [], []
            new_alphas, new_ctxh, new_sth = [], [], []
            new_accc, new_acca, new_words = [], [], []
            for rank_i, (ti, wi) in enumerate(zip(tis, wis)):
                new_samples.append(st[i]["samples"][ti] + [int(wi)])
                new_scores.append(float(costs[rank_i]))
                new_states.append(h2_np[sl][ti])
                new_alphas.append(st[i]["alphas"][ti] +
                                  [alpha_np[sl][ti, :lens[i]].copy()])
                new_ctxh.append(st[i]["ctx_hist"][ti] +
                                [ctxt_np[sl][ti].copy()])
                new_sth.append(st[i]["st_hist"][ti] + [h2_np[sl][ti].copy()])
                new_accc.append(accc_np[sl][ti])
                new_acca.append(acca_np[sl][ti])
                new_words.append(int(wi))

            # device histories for the rerank (pre-filter selection order)
            if any_lambda:
                sel = torch.tensor(tis, dtype=torch.int64, device=device)
                base = sl.start
                cur_a = alpha[base:base + live][sel].unsqueeze(0).float()
                cur_c = ctx_t[base:base + live][sel].unsqueeze(0).float()
                cur_s = h2[base:base + live][sel].unsqueeze(0).float()
                if dstate[i]["ha"] is None:
                    ha, hc, hs = cur_a, cur_c, cur_s
                else:
                    ha = torch.cat([dstate[i]["ha"][:, sel], cur_a], 0)
                    hc = torch.cat([dstate[i]["hc"][:, sel], cur_c], 0)
                    hs = torch.cat([dstate[i]["hs"][:, sel], cur_s], 0)

            keep = []
            samples, scores, states_l = [], [], []
            alphas_l, ctxh_l, sth_l, accc_l, acca_l, words_l = \
                [], [], [], [], [], []
            for idx2 in range(len(new_samples)):
                if new_samples[idx2][-1] == 0:
                    st[i]["out_samples"].append(new_samples[idx2])
                    st[i]["out_scores"].append(new_scores[idx2])
                    st[i]["out_alphas"].append(new_alphas[idx2])
                    st[i]["dead"] += 1
                else:
                    keep.append(idx2)
                    samples.append(new_samples[idx2])
                    scores.append(new_scores[idx2])
                    states_l.append(new_states[idx2])
                    alphas_l.append(new_alphas[idx2])
                    ctxh_l.append(new_ctxh[idx2])
                    sth_l.append(new_sth[idx2])
                    accc_l.append(new_accc[idx2])
                    acca_l.append(new_acca[idx2])
                    words_l.append(new_words[idx2])

            st[i].update(samples=samples, alphas=alphas_l, ctx_hist=ctxh_l,
                         st_hist=sth_l,
                         scores=numpy.array(scores, dtype="float32"))
            st[i]["live"] = len(samples)
            if st[i]["dead"] >= k:
                st[i]["live"] = 0
            if st[i]["live"] > 0:
                dstate[i]["w"] = torch.tensor(words_l, dtype=torch.int64,
                                              device=device)
                dstate[i]["state"] = torch.from_numpy(
                    numpy.array(states_l, dtype="float32")).to(device)
                dstate[i]["acc_c"] = torch.from_numpy(
                    numpy.array(accc_l, dtype="float32")).to(device)
                dstate[i]["acc_a"] = torch.from_numpy(
                    numpy.array(acca_l, dtype="float32")).to(device)
                if any_lambda:
                    kt = torch.tensor(keep, dtype=torch.int64, device=device)
                    dstate[i]["ha"] = ha[:, kt]
                    dstate[i]["hc"] = hc[:, kt]
                    dstate[i]["hs"] = hs[:, kt]

    results = []
    for i in range(S):
        out_s = list(st[i]["out_samples"])
        out_c = list(st[i]["out_scores"])
        out_a = list(st[i]["out_alphas"])
        for idx2 in range(st[i]["live"]):
            out_s.append(st[i]["samples"][idx2])
            out_c.append(float(st[i]["scores"][idx2]))
            out_a.append(st[i]["alphas"][idx2])
        results.append((out_s, out_c, out_a))
    return results
