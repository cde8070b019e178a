"""Manual GPU bisect for cond_gru_fwd: diff every per-step intermediate
against the eager oracle. Run on a GPU box:
    python tools/gpu_debug_cond.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from nats_amd.models.distraction import NatsModel, default_options
from nats_amd.ops import eager
from nats_amd.ops.cond_gru import (_ceil, _pack_rows, pack_gru1_weights,
                                   CondGRUScanFn)
from nats_amd.ops.gru import pack_fwd_weights
from nats_amd.ops import _hip_ext


def main():
    ext = _hip_ext()
    torch.manual_seed(5)
    g = torch.Generator().manual_seed(5)
    T, B, H, Ts, A, E = 3, 5, 32, 9, 12, 10
    C = 2 * H
    opts = default_options(dim_word=E, dim=H, dim_att=A, n_words=50)
    model = NatsModel(opts, seed=5)
    P = {k: v.detach() for k, v in model.P.items()}
    yg = torch.randn(T, B, 2 * H, generator=g)
    yc = torch.randn(T, B, H, generator=g)
    init = torch.randn(B, H, generator=g) * 0.1
    ctx = torch.randn(Ts, B, C, generator=g)
    pctx = ctx @ P["decoder_Wc_att"] + P["decoder_b_att"]

    # eager per-step intermediates
    h = init.clone()
    acc_c = torch.zeros(B, C)
    acc_a = torch.zeros(B, Ts)
    eag = []
    for t in range(T):
        # replicate cond_gru_step but keep h1/pstate/alpha too
        Hd = H
        preact1 = torch.sigmoid(h @ P["decoder_U"] + yg[t])
        r1, u1 = preact1[:, :Hd], preact1[:, Hd:]
        h1 = torch.tanh((h @ P["decoder_Ux"]) * r1 + yc[t])
        h1 = u1 * h + (1 - u1) * h1
        pstate = h1 @ P["decoder_W_att"]
        alpha, ctx_t = eager._attention(h1, ctx, None, pctx, acc_c, acc_a, P)
        preact2 = torch.sigmoid(h1 @ P["decoder_U_1"] + P["decoder_b_1"] +
                                ctx_t @ P["decoder_W_1"])
        r2, u2 = preact2[:, :Hd], preact2[:, Hd:]
        h2 = torch.tanh((h1 @ P["decoder_Ux_1"] + P["decoder_bx_1"]) * r2 +
                        ctx_t @ P["decoder_Wx_1"])
        h2 = u2 * h1 + (1 - u2) * h2
        acc_c = ctx_t + acc_c
        acc_a = alpha.t() + acc_a
        eag.append(dict(h1=h1, pstate=pstate, alpha=alpha.t(), ctx_t=ctx_t,
                        h2=h2, acc_c=acc_c.clone(), acc_a=acc_a.clone()))
        h = h2

    # HIP
    dev = "cuda"
    Pg = {k: v.to(dev) for k, v in P.items()}
    Hpad, Cpad = _ceil(H, 32), _ceil(C, 32)
    Upk2 = pack_fwd_weights(Pg["decoder_U"], Pg["decoder_Ux"])
    W1pk = pack_gru1_weights(Pg["decoder_U_1"], Pg["decoder_W_1"],
                             Pg["decoder_Ux_1"], Pg["decoder_Wx_1"], Hpad,
                             Cpad)
    WattPk = _pack_rows(Pg["decoder_W_att"].t(), _ceil(A, 16), Hpad)
    outs = ext.cond_gru_fwd(
        yg.to(dev).to(torch.bfloat16), yc.to(dev).to(torch.bfloat16), None,
        init.to(dev), ctx.to(dev).to(torch.bfloat16), None,
        pctx.to(dev).float().contiguous(), Upk2, W1pk, WattPk,
        Pg["decoder_b_1"].contiguous(), Pg["decoder_bx_1"].contiguous(),
        Pg["decoder_U_att"].reshape(-1).contiguous(), 0.0,
        Pg["decoder_D_wei"].reshape(-1).contiguous(),
        Pg["decoder_W_con"].reshape(-1).contiguous(),
        Pg["decoder_U_con"].reshape(-1).contiguous(), None, None)
    (h2_all, ctxs_all, alphas_all, accC, accA, h1_all, saved2, saved1,
     pstate_all, ctxpre_all, accA_used, accC_used) = outs

    def diff(name, t, hip, ref):
        hipc = hip.float().cpu()
        d = (hipc - ref).abs().max().item()
        nan = torch.isnan(hipc).any().item()
        print("t=%d %-8s maxdiff=%.5f nan=%s" % (t, name, d, nan))

    for t in range(T):
        e = eag[t]
        diff("h1", t, h1_all[t], e["h1"])
        diff("pstate", t, pstate_all[t], e["pstate"])
        diff("alpha", t, alphas_all[t], e["alpha"])
        diff("ctx_t", t, ctxs_all[t], e["ctx_t"])
        diff("h2", t, h2_all[t], e["h2"])
    diff("accC", T - 1, accC, eag[-1]["acc_c"])
    diff("accA", T - 1, accA, eag[-1]["acc_a"])


if __name__ == "__main__":
    main()
