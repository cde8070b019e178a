"""Batched multi-sentence beam search.

Serving-throughput extension beyond the reference (which decodes one
sentence per process, gen.py:30-45): S sentences' beams share every
decode step as one (S*k)-row batch through the fused decoder kernels
(row budget: the cond-GRU kernels take B <= 32). Per-sentence semantics
are identical to decode.beam.gen_sample — sources are padded to the
longest and masked via ctx_mask, so attention, the distraction terms and
the histories are unaffected by padding.

Execution layout (MI355X): all recurrent state (h2, acc_ctx, acc_alpha,
rerank histories) stays device-resident across steps and hypothesis
reshuffles are device index_selects; the only per-step host transfers
are the top-k candidate ids/costs and the selected attention rows for
the alignment output (two small synchronisations per step — the naive
numpy bookkeeping round-trip was ~25% of decode time in the kernel
trace, profiles/decode_kernel_stats.csv).

Numerics note: results are mathematically identical to per-sentence
gen_sample, but not bitwise — the batched masked-mean init state and
torch reductions over different padded lengths round differently at the
last ulp. On a trained model beams are insensitive to this; a
random-init model with near-tie hypotheses can flip a beam choice
(observed only with inflated-readout test models).
"""

import math

import numpy
import torch

from .beam import distraction_penalties_gpu


@torch.no_grad()
def gen_sample_batched(model, xs, k=1, maxlen=30, use_unk=False,
                       kl_factor=0.0, ctx_factor=0.0, state_factor=0.0,
                       use_graph=True):
    """Beam-decode a list of sources jointly.

    xs: list of (T_i, 1) int64 tensors (same device as model).
    maxlen: an int, or a per-sentence list (serving: each request brings
    its own cap — a sentence whose cap is reached flushes its live
    hypotheses and drops out of the shared batch, gen_sample
    semantics nats.py:1068-1074 per sentence).
    Returns a list of (sample, sample_score, sample_dec_alphas) tuples,
    one per sentence, each shaped exactly like gen_sample's beam output
    (alignment alphas are trimmed to the sentence's own length).
    """
    S = len(xs)
    assert S >= 1
    maxlens = ([int(maxlen)] * S if isinstance(maxlen, int)
               else [int(m) for m in maxlen])
    assert len(maxlens) == S
    device = xs[0].device
    any_lambda = (kl_factor > 0.0 or ctx_factor > 0.0 or state_factor > 0.0)

    # ONE batched masked encode for all sentences (identical per-column to
    # each sentence's own unpadded encode — see f_init): pads to the
    # longest source so the encoder scans run S columns per launch instead
    # of S separate T-step launches.
    lens = [int(x.shape[0]) for x in xs]
    Ts = max(lens)
    if use_graph and device.type == "cuda":
        # bucket the padded source length so serving traffic with varied
        # lengths reuses a handful of captured graphs instead of one per
        # distinct Ts (padding is exact: ctx_mask zeroes the pad columns
        # through attention and the encoder mask-blend)
        Ts = (Ts + 63) // 64 * 64
    x_pad = torch.zeros(Ts, S, dtype=torch.int64, device=device)
    src_mask = torch.zeros(Ts, S, device=device)
    for i, x in enumerate(xs):
        x_pad[:lens[i], i] = x[:, 0]
        src_mask[:lens[i], i] = 1.0
    init_all, ctx_all = model.f_init(x_pad, src_mask)
    C = ctx_all.shape[2]

    # per-sentence host bookkeeping (sample token lists + scores; identical
    # math to gen_sample) — everything tensor-valued lives on device
    st = [dict(live=1, dead=0, samples=[[]],
               scores=numpy.zeros(1, dtype="float32"),
               alphas=[[]], out_samples=[], out_scores=[], out_alphas=[])
          for _ in range(S)]
    dstate = [dict(state=init_all[i:i + 1],
                   w=torch.full((1,), -1, dtype=torch.int64, device=device),
                   acc_c=torch.zeros(1, C, device=device),
                   acc_a=torch.zeros(1, Ts, device=device),
                   ha=None, hc=None, hs=None)
              for i in range(S)]

    ctx_pad = torch.zeros(Ts, S, C, device=device, dtype=ctx_all.dtype)
    for i in range(S):
        ctx_pad[:lens[i], i] = ctx_all[:lens[i], i]
    ctx_mask_s = src_mask
    pctx_pad = model.project_ctx(ctx_pad)

    # hipGraph-captured f_next for the shared step (fixed row budget S*k;
    # the per-step sentence gather runs inside the graph)
    stepper = None
    if use_graph and device.type == "cuda":
        from .graph import get_batched_stepper
        stepper = get_batched_stepper(model, ctx_pad.float(), ctx_mask_s,
                                      pctx_pad.float(), S * k)

    NEG_UNK = math.log(1e-20)

    for ii in range(max(maxlens)):
        # sentences at their own maxlen flush live hypotheses and leave
        for i in range(S):
            if st[i]["live"] > 0 and ii >= maxlens[i]:
                for idx2 in range(st[i]["live"]):
                    st[i]["out_samples"].append(st[i]["samples"][idx2])
                    st[i]["out_scores"].append(float(st[i]["scores"][idx2]))
                    st[i]["out_alphas"].append(st[i]["alphas"][idx2])
                st[i].update(samples=[], alphas=[],
                             scores=numpy.zeros(0, dtype="float32"))
                st[i]["live"] = 0
        alive = [i for i in range(S) if st[i]["live"] > 0]
        if not alive:
            break
        sent_idx = torch.tensor(
            sum(([i] * st[i]["live"] for i in alive), []), device=device)
        y = torch.cat([dstate[i]["w"] for i in alive])
        state = torch.cat([dstate[i]["state"] for i in alive])
        acc_c = torch.cat([dstate[i]["acc_c"] for i in alive])
        acc_a = torch.cat([dstate[i]["acc_a"] for i in alive])
        if stepper is None:
            ctx_b = ctx_pad[:, sent_idx]
            cmask_b = ctx_mask_s[:, sent_idx]
            pctx_b = pctx_pad[:, sent_idx]

        if stepper is not None:
            rows = y.shape[0]
            outs = stepper.step(sent_idx, y, state.float(), acc_c.float(),
                                acc_a.float())
            probs, h2, alpha, ctx_t, acc_c, acc_a = [
                o[:rows] for o in outs]
        else:
            probs, _, h2, alpha, ctx_t, acc_c, acc_a = model.f_next(
                y, ctx_b, cmask_b, pctx_b, state, acc_c, acc_a,
                sample_draw=False)
        V = probs.shape[1]

        # ---- phase 1: per-sentence top-k ON DEVICE, one host transfer ----
        logp = probs.float().log()
        if not use_unk:
            logp[:, 1] = NEG_UNK
        sel_parts = []   # per alive sentence: (ranks (want,), costs (want,))
        row0 = 0
        for i in alive:
            live = st[i]["live"]
            sl = slice(row0, row0 + live)
            row0 += live
            cand = torch.as_tensor(st[i]["scores"],
                                   device=device)[:, None] - logp[sl]
            cand_flat = cand.flatten()
            want = k - st[i]["dead"]
            sel_flat = cand_flat
            if ii > 0 and any_lambda and dstate[i]["ha"] is not None:
                pen = distraction_penalties_gpu(
                    dstate[i]["ha"], dstate[i]["hc"], dstate[i]["hs"],
                    alpha[sl].float(), ctx_t[sl].float(), h2[sl].float(),
                    kl_factor, ctx_factor, state_factor)
                sel_flat = (cand + pen[:, None]).flatten()
            ranks = sel_flat.topk(want, largest=False).indices
            sel_parts.append((ranks, cand_flat[ranks]))  # UN-reranked costs
        flat_host = torch.cat([torch.cat([r.double(), c.double()])
                               for r, c in sel_parts]).cpu().numpy()

        # ---- phase 2: bookkeeping + device-side state reshuffle ----
        alpha_gather = []   # device rows to fetch for alignment histories
        pend = []           # (i, sl, tis, wis, costs, sel_dev)
        row0 = 0
        off = 0
        for i in alive:
            live = st[i]["live"]
            sl = slice(row0, row0 + live)
            row0 += live
            want = k - st[i]["dead"]
            ranks = flat_host[off:off + want].astype(numpy.int64)
            costs = flat_host[off + want:off + 2 * want].astype("float32")
            off += 2 * want
            tis = ranks // V
            wis = ranks % V
            sel_dev = torch.as_tensor(tis, device=device) + sl.start
            alpha_gather.append(alpha.float()[sel_dev])
            pend.append((i, sl, tis, wis, costs, sel_dev))
        alpha_sel = torch.cat(alpha_gather).cpu().numpy()

        arow = 0
        for i, sl, tis, wis, costs, sel_dev in pend:
            want = len(tis)
            a_np = alpha_sel[arow:arow + want]
            arow += want

            new_samples, new_scores, new_alphas, new_words = [], [], [], []
            for r, (ti, wi) in enumerate(zip(tis, wis)):
                new_samples.append(st[i]["samples"][ti] + [int(wi)])
                new_scores.append(float(costs[r]))
                new_alphas.append(st[i]["alphas"][ti] +
                                  [a_np[r, :lens[i]].copy()])
                new_words.append(int(wi))

            # device histories for the rerank (pre-filter selection order)
            if any_lambda:
                cur_a = alpha.float()[sel_dev].unsqueeze(0)
                cur_c = ctx_t.float()[sel_dev].unsqueeze(0)
                cur_s = h2.float()[sel_dev].unsqueeze(0)
                if dstate[i]["ha"] is None:
                    ha, hc, hs = cur_a, cur_c, cur_s
                else:
                    ti_dev = sel_dev - sl.start
                    ha = torch.cat([dstate[i]["ha"][:, ti_dev], cur_a], 0)
                    hc = torch.cat([dstate[i]["hc"][:, ti_dev], cur_c], 0)
                    hs = torch.cat([dstate[i]["hs"][:, ti_dev], cur_s], 0)

            keep, samples, scores, alphas_l, words_l = [], [], [], [], []
            for idx2 in range(want):
                if new_samples[idx2][-1] == 0:
                    st[i]["out_samples"].append(new_samples[idx2])
                    st[i]["out_scores"].append(new_scores[idx2])
                    st[i]["out_alphas"].append(new_alphas[idx2])
                    st[i]["dead"] += 1
                else:
                    keep.append(idx2)
                    samples.append(new_samples[idx2])
                    scores.append(new_scores[idx2])
                    alphas_l.append(new_alphas[idx2])
                    words_l.append(new_words[idx2])

            st[i].update(samples=samples, alphas=alphas_l,
                         scores=numpy.array(scores, dtype="float32"))
            st[i]["live"] = len(samples)
            if st[i]["dead"] >= k:
                st[i]["live"] = 0
            if st[i]["live"] > 0:
                kt = torch.tensor(keep, dtype=torch.int64, device=device)
                sel_keep = sel_dev[kt]
                dstate[i]["w"] = torch.tensor(words_l, dtype=torch.int64,
                                              device=device)
                dstate[i]["state"] = h2[sel_keep].float()
                dstate[i]["acc_c"] = acc_c[sel_keep].float()
                dstate[i]["acc_a"] = acc_a[sel_keep].float()
                if any_lambda:
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
