#!/usr/bin/env python3
"""Bisect the whole-step graph-capture failure: at which model scale do
replays stop moving parameters? Sweeps (dim, vocab, src) from the known-
good test shape to the bench shape; prints param-delta per replay."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch

from nats_amd.data.synthetic import synthetic_batch
from nats_amd.engine.optim import build_optimizer
from nats_amd.models.distraction import NatsModel, default_options


def probe(dim, V, src, tgt, B):
    from nats_amd.utils.step_graph import GraphedTrainStep
    opts = default_options(dim=dim, dim_word=100, dim_att=100, n_words=V,
                           batch_size=B, optimizer="adadelta", clip_c=100.0,
                           maxlen=src + 1)
    model = NatsModel(opts, seed=1).cuda()
    opt = build_optimizer("adadelta", list(model.P.items()), clip_c=100.0)
    rng = numpy.random.RandomState(1)
    batch = [torch.from_numpy(a).cuda()
             for a in synthetic_batch(rng, B, src, tgt, V)]

    def csum():
        with torch.no_grad():
            return float(torch.stack(
                [p.float().abs().sum() for p in model.parameters()]).sum())

    try:
        st = GraphedTrainStep(model, opt, None, batch)
    except Exception as e:
        print("dim=%d V=%d src=%d: CAPTURE FAILED: %s"
              % (dim, V, src, str(e)[:160]))
        return
    costs, deltas = [], []
    prev = csum()
    for _ in range(3):
        c = st.step(*batch)
        torch.cuda.synchronize()
        costs.append(round(float(c.detach()), 3))
        cur = csum()
        deltas.append(round(cur - prev, 6))
        prev = cur
    print("dim=%d V=%d src=%d B=%d: costs=%s param_dsum=%s"
          % (dim, V, src, B, costs, deltas), flush=True)


def main():
    probe(96, 600, 48, 12, 4)       # known-good test shape
    probe(96, 30000, 48, 12, 4)     # big vocab only
    probe(500, 4000, 120, 30, 20)   # lcsts shape
    probe(1000, 4000, 120, 30, 20)  # big dim (split-K paths)
    probe(1000, 30000, 120, 30, 20)  # big dim + vocab
    probe(1000, 30000, 800, 100, 20)  # full bench shape
    print("done")


if __name__ == "__main__":
    main()
