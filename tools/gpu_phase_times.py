"""Wall-time per model phase on the cnn_dm bench config."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy, torch
from nats_amd.data.synthetic import synthetic_batch
from nats_amd.models.distraction import NatsModel, default_options
from nats_amd import ops

def t(label, fn, n=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): out = fn()
    torch.cuda.synchronize()
    print("%-16s %8.2f ms" % (label, 1e3*(time.perf_counter()-t0)/n))
    return out

def main():
    opts = default_options(dim_word=100, dim=1000, dim_att=100, n_words=30000)
    model = NatsModel(opts, seed=1).cuda()
    rng = numpy.random.RandomState(0)
    x, xm, y, ym = [torch.from_numpy(a).cuda()
                    for a in synthetic_batch(rng, 20, 800, 100, 30000)]
    amp = torch.autocast("cuda", dtype=torch.bfloat16)
    with amp, torch.no_grad():
        ctx, init = t("encode", lambda: model.encode(x, xm))
        pctx = model.project_ctx(ctx)
        emb = model.embed(y); es = torch.zeros_like(emb); es[1:] = emb[:-1]
        yg, yc = model._dec_inputs(es)
        dec = t("decoder", lambda: ops.cond_gru_scan(
            yg, yc, ym, init, ctx, xm, pctx, model.P))
        h2s, ctxs = dec[0], dec[1]
        logits = t("readout", lambda: model.readout_logits(h2s, es, ctxs))
        T, B, V = logits.shape
        t("softmax_ce", lambda: ops.softmax_xent(
            logits.reshape(T*B, V), y.reshape(-1)))
    # bwd probes
    with amp:
        def enc_bwd():
            ctx2, init2 = model.encode(x, xm)
            (ctx2.float().sum() + init2.float().sum()).backward()
        t("enc fwd+bwd", enc_bwd, n=2)
        def dec_bwd():
            ctx2, init2 = model.encode(x, xm)
            pctx2 = model.project_ctx(ctx2)
            d = ops.cond_gru_scan(yg, yc, ym, init2, ctx2, xm, pctx2, model.P)
            (d[0].float().sum() + d[1].float().sum()).backward()
        t("enc+dec f+b", dec_bwd, n=2)

main()
