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
