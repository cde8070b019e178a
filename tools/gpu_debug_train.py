"""Stage-by-stage NaN hunt at the train-step test config."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch

from nats_amd.data.synthetic import synthetic_batch
from nats_amd.models.distraction import NatsModel, default_options
from nats_amd import ops


def chk(name, t):
    t = t.float()
    print("%-12s finite=%s min=%.3g max=%.3g" %
          (name, bool(torch.isfinite(t).all()), float(t.min()),
           float(t.max())))


def main():
    opts = default_options(dim_word=24, dim=48, dim_att=12, n_words=500)
    model = NatsModel(opts, seed=2).cuda()
    rng = numpy.random.RandomState(3)
    x, xm, y, ym = [torch.from_numpy(a).cuda()
                    for a in synthetic_batch(rng, 8, 30, 10, 500)]
    with torch.autocast("cuda", dtype=torch.bfloat16):
        ctx, init_state = model.encode(x, xm)
        chk("ctx", ctx)
        chk("init", init_state)
        pctx = model.project_ctx(ctx)
        chk("pctx", pctx)
        emb = model.embed(y)
        emb_shifted = torch.zeros_like(emb)
        emb_shifted[1:] = emb[:-1]
        yg, yc = model._dec_inputs(emb_shifted)
        chk("yg", yg)
        h2s, ctxs, alphas, accC, accA = ops.cond_gru_scan(
            yg, yc, ym, init_state, ctx, xm, pctx, model.P)
        chk("h2s", h2s)
        chk("ctxs", ctxs)
        chk("alphas", alphas)
        logits = model.readout_logits(h2s, emb_shifted, ctxs)
        chk("logits", logits)
        T, B, V = logits.shape
        cost = ops.softmax_xent(logits.reshape(T * B, V), y.reshape(-1))
        chk("nll", cost)
        cost = (cost.reshape(T, B) * ym).sum(0)
        chk("cost", cost)
        # run fp32 CPU for comparison
    cpu_model = NatsModel(opts, params=model.get_params())
    ref = cpu_model(x.cpu(), xm.cpu(), y.cpu(), ym.cpu())
    chk("cost_cpu", ref)
    print("rel:", float(((cost.float().cpu() - ref) / ref).abs().max()))

    # backward NaN hunt, GPU vs CPU grads
    cost.mean().backward()
    ref.mean().backward()
    bad = 0
    for k, p in model.P.items():
        g = p.grad
        gc = cpu_model.P[k].grad
        if g is None:
            print("%-20s NO GRAD" % k)
            continue
        fin = bool(torch.isfinite(g).all())
        rel = float((g.float().cpu() - gc).abs().max() /
                    gc.abs().max().clamp_min(1e-5))
        flag = ""
        if not fin:
            flag = "NONFINITE"
            bad += 1
        elif rel > 0.1:
            flag = "MISMATCH"
            bad += 1
        if flag:
            print("%-20s finite=%s rel=%.4f %s" % (k, fin, rel, flag))
    print("bad grads:", bad)




def train_loop():
    """Replicate test_model_train_step_gpu exactly, printing costs."""
    from nats_amd.engine.optim import build_optimizer
    opts = default_options(dim_word=24, dim=48, dim_att=12, n_words=500)
    model = NatsModel(opts, seed=2).cuda()
    opt = build_optimizer("adadelta", list(model.P.items()), clip_c=1.0)
    print("optimizer:", type(opt).__name__)
    rng = numpy.random.RandomState(3)
    x, xm, y, ym = [torch.from_numpy(a).cuda()
                    for a in synthetic_batch(rng, 8, 30, 10, 500)]
    for i in range(3):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            cost = model(x, xm, y, ym).mean()
        cost.backward()
        gnorm = opt.step()
        print("step %d cost=%.4f gnorm=%s finite_params=%s" %
              (i, float(cost), float(gnorm),
               all(bool(torch.isfinite(p).all()) for p in model.P.values())))


if __name__ == "__main__":
    import sys as _s
    if len(_s.argv) > 1 and _s.argv[1] == "loop":
        train_loop()
        raise SystemExit
    main()
