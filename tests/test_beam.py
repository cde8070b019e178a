import numpy
import torch

from nats_amd.decode.beam import (_cosine_dist, _kl_div,
                                  distraction_penalties, gen_sample)
from nats_amd.models.distraction import NatsModel


def _model(tiny_options):
    return NatsModel(tiny_options, seed=7)


def test_kl_and_cosine_match_scipy():
    from scipy.stats import entropy
    from scipy.spatial.distance import cosine
    rng = numpy.random.RandomState(0)
    for _ in range(5):
        p = rng.rand(9) + 1e-3
        q = rng.rand(9) + 1e-3
        assert abs(_kl_div(p, q) - entropy(p, q)) < 1e-10
        u, v = rng.randn(6), rng.randn(6)
        assert abs(_cosine_dist(u, v) - cosine(u, v)) < 1e-10


def test_beam_returns_k_hypotheses(tiny_options):
    model = _model(tiny_options)
    x = torch.randint(2, tiny_options["n_words"], (8, 1))
    sample, score, alphas = gen_sample(model, x, k=4, maxlen=12,
                                       stochastic=False)
    assert 1 <= len(sample) <= 4
    assert len(sample) == len(score) == len(alphas)
    for s, al in zip(sample, alphas):
        assert len(s) >= 1
        # one alpha history entry per generated token
        assert len(al) == len(s)
        for a in al:
            assert a.shape == (8,)


def test_beam_unk_suppression(tiny_options):
    """use_unk=False forces p(UNK)=1e-20 -> UNK never chosen."""
    model = _model(tiny_options)
    x = torch.randint(2, tiny_options["n_words"], (6, 1))
    sample, _, _ = gen_sample(model, x, k=3, maxlen=15, stochastic=False,
                              use_unk=False)
    for s in sample:
        assert 1 not in s[:-1]  # (eos terminator may be absent)


def test_stochastic_sampling(tiny_options):
    model = _model(tiny_options)
    x = torch.randint(2, tiny_options["n_words"], (6, 1))
    g = torch.Generator().manual_seed(3)
    sample, score, _ = gen_sample(model, x, k=1, maxlen=10, stochastic=True,
                                  generator=g)
    assert isinstance(sample, list)
    assert isinstance(score, float)
    assert len(sample) <= 10


def test_distraction_penalties_shapes():
    rng = numpy.random.RandomState(1)
    live_k, Ts, C, H = 3, 5, 8, 6
    hist_a = [[rng.rand(Ts) for _ in range(2)] for _ in range(live_k)]
    hist_c = [[rng.randn(C) for _ in range(2)] for _ in range(live_k)]
    hist_s = [[rng.randn(H) for _ in range(2)] for _ in range(live_k)]
    cur_a = rng.rand(live_k, Ts)
    cur_c = rng.randn(live_k, C)
    cur_s = rng.randn(live_k, H)
    a, c, s = distraction_penalties(hist_a, hist_c, hist_s, cur_a, cur_c,
                                    cur_s, 1.0, 2.0, 3.0)
    assert a.shape == c.shape == s.shape == (live_k,)
    assert (a <= 0).all()  # -kl_factor * min KL, KL >= 0


def test_distraction_rerank_changes_selection_not_costs(tiny_options):
    """Penalties steer selection; accumulated costs stay un-reranked
    (nats.py:1004)."""
    model = _model(tiny_options)
    x = torch.randint(2, tiny_options["n_words"], (6, 1))
    s0, c0, _ = gen_sample(model, x, k=3, maxlen=8, stochastic=False,
                           use_unk=True)
    s1, c1, _ = gen_sample(model, x, k=3, maxlen=8, stochastic=False,
                           use_unk=True, kl_factor=5.0, ctx_factor=5.0,
                           state_factor=5.0)
    # all returned costs are sums of -log p along the path: positive
    for c in c0 + c1:
        assert c > 0


def _oracle_beam(model, x, k, maxlen, use_unk):
    """Independent re-derivation of the reference beam loop (nats.py:
    940-1074) written against f_init/f_next directly — no shared code
    with decode.beam. Used to cross-check gen_sample's bookkeeping."""
    import torch
    init, ctx0 = model.f_init(x)
    pctx0 = model.project_ctx(ctx0)
    Ts, _, C = ctx0.shape
    hyps = [dict(toks=[], cost=0.0, state=init[0],
                 accC=torch.zeros(C), accA=torch.zeros(Ts))]
    done = []
    for step in range(maxlen):
        if not hyps or len(done) >= k:
            break
        B = len(hyps)
        y = torch.tensor([h["toks"][-1] if h["toks"] else -1 for h in hyps])
        state = torch.stack([h["state"] for h in hyps])
        accC = torch.stack([h["accC"] for h in hyps])
        accA = torch.stack([h["accA"] for h in hyps])
        probs, _, h2, alpha, ctx_t, accC2, accA2 = model.f_next(
            y, ctx0.expand(Ts, B, C), None,
            pctx0.expand(Ts, B, pctx0.shape[2]), state, accC, accA,
            sample_draw=False)
        p = probs.double().numpy()
        if not use_unk:
            p[:, 1] = 1e-20
        import numpy as np
        cand = np.array([h["cost"] for h in hyps])[:, None] - np.log(p)
        flat = cand.flatten()
        order = flat.argsort()[:k - len(done)]
        new = []
        for r in order:
            ti, wi = int(r) // p.shape[1], int(r) % p.shape[1]
            h = dict(toks=hyps[ti]["toks"] + [wi], cost=float(flat[r]),
                     state=h2[ti], accC=accC2[ti], accA=accA2[ti])
            if wi == 0:
                done.append(h)
            else:
                new.append(h)
        hyps = new
    done += hyps  # flush still-live at maxlen
    return (sorted(tuple(h["toks"]) for h in done),
            sorted(round(h["cost"], 4) for h in done))


def test_gen_sample_matches_independent_oracle(tiny_options):
    """Cross-check the production beam (decode/beam.py) against a from-
    scratch oracle on a model with well-separated probabilities."""
    import torch
    from nats_amd.models.distraction import NatsModel
    from nats_amd.decode.beam import gen_sample
    model = NatsModel(tiny_options, seed=21).eval()
    with torch.no_grad():
        for key in ("ff_logit_lstm_W", "ff_logit_ctx_W", "ff_logit_prev_W",
                    "ff_logit_W"):
            model.P[key].mul_(50.0)
    g = torch.Generator().manual_seed(2)
    for trial in range(3):
        x = torch.randint(2, tiny_options["n_words"], (6 + trial, 1),
                          generator=g)
        x[-1] = 0
        with torch.no_grad():
            s, c, a = gen_sample(model, x, k=4, maxlen=7, stochastic=False,
                                 use_unk=True)
            otoks, ocosts = _oracle_beam(model, x, k=4, maxlen=7,
                                         use_unk=True)
        assert sorted(map(tuple, s)) == otoks
        assert sorted(round(v, 4) for v in c) == ocosts
