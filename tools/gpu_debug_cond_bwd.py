"""GPU bisect for cond_gru backward: report rel-diff for every gradient."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from nats_amd.models.distraction import NatsModel, default_options
from nats_amd.ops import eager
from nats_amd.ops.cond_gru import cond_gru_scan_hip


def main():
    torch.manual_seed(5)
    g = torch.Generator().manual_seed(5)
    T, B, H, Ts, A, E = 6, 4, 24, 8, 10, 10
    C = 2 * H
    opts = default_options(dim_word=E, dim=H, dim_att=A, n_words=50)
    model = NatsModel(opts, seed=5)
    with torch.no_grad():
        for k in ("decoder_W_att", "decoder_Wc_att", "decoder_U_att",
                  "decoder_D_wei", "decoder_W_con", "decoder_U_con"):
            model.P[k].mul_(40.0)
    yg = torch.randn(T, B, 2 * H, generator=g)
    yc = torch.randn(T, B, H, generator=g)
    init = torch.randn(B, H, generator=g) * 0.1
    ctx = torch.randn(Ts, B, C, generator=g)
    lens = torch.randint(2, T + 1, (B,), generator=g)
    mask = (torch.arange(T).unsqueeze(1) < lens.unsqueeze(0)).float()
    slens = torch.randint(2, Ts + 1, (B,), generator=g)
    ctx_mask = (torch.arange(Ts).unsqueeze(1) < slens.unsqueeze(0)).float()

    keys = ["decoder_U", "decoder_Ux", "decoder_U_1", "decoder_W_1",
            "decoder_b_1", "decoder_Wx_1", "decoder_Ux_1", "decoder_bx_1",
            "decoder_W_att", "decoder_U_att", "decoder_c_att",
            "decoder_W_con", "decoder_U_con", "decoder_D_wei",
            "decoder_Wc_att", "decoder_b_att"]

    def run(dev, fn):
        P = {k: v.detach().clone().to(dev).requires_grad_(k in keys)
             for k, v in model.P.items()}
        ins = [t.clone().to(dev).requires_grad_(True)
               for t in (yg, yc, init, ctx)]
        pctx = ins[3] @ P["decoder_Wc_att"] + P["decoder_b_att"]
        h2s, ctxs, alphas, accC, accA = fn(
            ins[0], ins[1], mask.to(dev), ins[2], ins[3], ctx_mask.to(dev),
            pctx, P)
        torch.manual_seed(0)
        w1 = torch.randn(h2s.shape)
        w2 = torch.randn(ctxs.shape)
        ((h2s.float() * w1.to(dev)).sum() +
         (ctxs.float() * w2.to(dev)).sum()).backward()
        grads = {k: P[k].grad.cpu().float() for k in keys}
        gins = {n: t.grad.cpu().float()
                for n, t in zip(["yg", "yc", "init", "ctx"], ins)}
        return grads, gins

    rg, ri = run("cpu", eager.cond_gru_scan)
    hg, hi = run("cuda", cond_gru_scan_hip)
    for name, (a, b) in list({k: (rg[k], hg[k]) for k in rg}.items()) + \
            list({k: (ri[k], hi[k]) for k in ri}.items()):
        denom = a.abs().max().clamp_min(1e-4)
        rel = float((a - b).abs().max() / denom)
        print("%-18s ref|max|=%9.4f rel=%.4f %s" %
              (name, float(a.abs().max()), rel, "BAD" if rel > 0.08 else ""))


if __name__ == "__main__":
    main()
