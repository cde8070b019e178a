#!/usr/bin/env python3
"""Isolate the embed-kernel training divergence at the bench shape:
one full train step with HIP embed vs eager embed (NATS_EMBED_EAGER),
comparing cost and Wemb.grad / total grad norm."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch

from nats_amd.data.synthetic import synthetic_batch
from nats_amd.models.distraction import NatsModel, default_options


def one_backward(use_hip):
    os.environ.pop("NATS_EMBED_EAGER", None)
    if not use_hip:
        os.environ["NATS_EMBED_EAGER"] = "1"
    opts = default_options(dim=1000, dim_word=100, dim_att=100,
                           n_words=30000, batch_size=20, clip_c=100.0,
                           maxlen=801)
    model = NatsModel(opts, seed=1234).cuda()
    rng = numpy.random.RandomState(1234)
    x, xm, y, ym = [torch.from_numpy(a).cuda()
                    for a in synthetic_batch(rng, 20, 800, 100, 30000)]
    with torch.autocast("cuda", dtype=torch.bfloat16):
        cost = model(x, xm, y, ym).mean()
    cost.backward()
    g = {k: p.grad.detach().clone() for k, p in model.P.items()
         if p.grad is not None}
    return float(cost.detach()), g


def main():
    c_hip, g_hip = one_backward(True)
    c_eager, g_eager = one_backward(False)
    print("cost hip=%.4f eager=%.4f" % (c_hip, c_eager))
    tot_h = sum(float((v ** 2).sum()) for v in g_hip.values()) ** 0.5
    tot_e = sum(float((v ** 2).sum()) for v in g_eager.values()) ** 0.5
    print("global grad norm hip=%.4f eager=%.4f" % (tot_h, tot_e))
    for k in sorted(g_hip):
        a, b = g_hip[k], g_eager[k]
        err = float((a - b).abs().max())
        mx = float(b.abs().max())
        if err > 1e-2 * max(mx, 1e-6) or k == "Wemb":
            print("%-22s max|d|=%.4e ref_max=%.4e norm_hip=%.4e "
                  "norm_eager=%.4e" % (k, err, mx,
                                       float(a.norm()), float(b.norm())))


if __name__ == "__main__":
    main()
