#!/usr/bin/env python3
"""Decode benchmark — beam-k summaries/sec (BASELINE.json configs[3]:
distraction beam search, hipGraph-captured decode, LCSTS shape).

Synthetic source sequences, random-init weights (no network). Prints one
JSON line per variant (plain beam / distraction-rerank beam / graph)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch

from nats_amd.decode.beam import gen_sample
from nats_amd.models.distraction import NatsModel, default_options


def run(model, xs, k, maxlen, use_graph, lam, batched=False):
    n_tokens = 0
    t0 = time.perf_counter()
    if batched:
        from nats_amd.decode.batched import gen_sample_batched
        sb = max(1, 64 // k)
        for base in range(0, len(xs), sb):
            outs = gen_sample_batched(model, xs[base:base + sb], k=k,
                                      maxlen=maxlen, use_unk=True,
                                      kl_factor=lam, ctx_factor=lam,
                                      state_factor=lam, use_graph=use_graph)
            for sample, _, _ in outs:
                n_tokens += sum(len(s) for s in sample)
    else:
        for x in xs:
            sample, score, alphas = gen_sample(
                model, x, k=k, maxlen=maxlen, stochastic=False, use_unk=True,
                kl_factor=lam, ctx_factor=lam, state_factor=lam,
                use_graph=use_graph)
            n_tokens += sum(len(s) for s in sample)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return len(xs) / dt, n_tokens


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=20, help="summaries to decode")
    ap.add_argument("--k", type=int, default=10, help="beam width")
    ap.add_argument("--src", type=int, default=120)
    ap.add_argument("--maxlen", type=int, default=30)
    ap.add_argument("--dim", type=int, default=500)
    ap.add_argument("--vocab", type=int, default=4000)
    args = ap.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    opts = default_options(dim=args.dim, dim_word=100, dim_att=100,
                           n_words=args.vocab)
    model = NatsModel(opts, seed=0).to(device).eval()
    rng = numpy.random.RandomState(0)
    xs = [torch.tensor(
        list(rng.randint(2, args.vocab, size=args.src)) + [0],
        dtype=torch.int64, device=device).reshape(-1, 1)
        for _ in range(args.n)]

    variants = [("beam", False, 0.0, False),
                ("beam+distraction", False, 0.5, False),
                ("beam+batched", False, 0.0, True),
                ("beam+distraction+batched", False, 0.5, True)]
    if device == "cuda":
        variants.append(("beam+hipgraph", True, 0.0, False))
        variants.append(("beam+batched+hipgraph", True, 0.0, True))
        variants.append(("beam+distraction+batched+hipgraph", True, 0.5,
                         True))
    for name, graph, lam, batched in variants:
        run(model, xs[:2], args.k, args.maxlen, graph, lam, batched)
        sps, ntok = run(model, xs, args.k, args.maxlen, graph, lam, batched)
        print(json.dumps({
            "metric": "summaries_per_sec", "variant": name, "value": sps,
            "beam": args.k, "n": args.n, "src_len": args.src,
            "dim": args.dim, "vocab": args.vocab, "device": device,
            "data": "synthetic"}))


if __name__ == "__main__":
    main()
