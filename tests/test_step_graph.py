"""Whole-step hipGraph capture: parity with the eager step (GPU)."""

import numpy
import pytest
import torch

pytestmark = pytest.mark.gpu


def _setup(seed=7, B=4, src=48, tgt=12, V=600):
    from nats_amd.data.synthetic import synthetic_batch
    from nats_amd.engine.optim import build_optimizer
    from nats_amd.models.distraction import NatsModel, default_options

    device = torch.device("cuda", 0)
    opts = default_options(dim=96, dim_word=32, dim_att=24, n_words=V,
                           batch_size=B, optimizer="adadelta", clip_c=100.0)
    model = NatsModel(opts, seed=seed).to(device)
    opt = build_optimizer("adadelta", list(model.P.items()), clip_c=100.0)
    rng = numpy.random.RandomState(seed)
    batch = [torch.from_numpy(a).to(device)
             for a in synthetic_batch(rng, B, src, tgt, V)]
    return model, opt, batch


def _eager_steps(model, opt, batch, n):
    costs = []
    for _ in range(n):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            cost = model(*batch).mean()
        cost.backward()
        opt.step()
        costs.append(float(cost.detach()))
    return costs


def test_graphed_step_matches_eager():
    """N graphed replays == N eager steps from the same init (the capture
    snapshot/restore must inject no extra updates)."""
    from nats_amd.utils.step_graph import GraphedStepCache

    model_e, opt_e, batch = _setup()
    eager_costs = _eager_steps(model_e, opt_e, batch, 4)

    model_g, opt_g, batch_g = _setup()  # same seeds -> same init/batch
    cache = GraphedStepCache(model_g, opt_g)
    gstep = cache.get(*batch_g)
    assert gstep is not None, "capture failed on GPU"
    graph_costs = []
    for _ in range(4):
        c = gstep.step(*batch_g)
        graph_costs.append(float(c.detach()))

    # identical kernels, identical order -> tight agreement (bf16 noise
    # from allocator-address-dependent atomics only)
    for ce, cg in zip(eager_costs, graph_costs):
        assert abs(ce - cg) / max(abs(ce), 1.0) < 5e-3, (eager_costs,
                                                         graph_costs)
    # parameters after N steps agree
    for (k, pe), (k2, pg) in zip(model_e.P.items(), model_g.P.items()):
        err = float((pe - pg).abs().max())
        ref = float(pe.abs().max())
        assert err / max(ref, 1e-6) < 5e-3, (k, err, ref)


def test_graphed_step_updates_decode_caches():
    """Graph replays bump the per-param update epoch so the packed-weight
    decode caches can't serve stale weights."""
    from nats_amd.utils.step_graph import GraphedStepCache

    model, opt, batch = _setup()
    cache = GraphedStepCache(model, opt)
    gstep = cache.get(*batch)
    assert gstep is not None
    p = next(iter(model.P.values()))
    e0 = getattr(p, "_nats_update_epoch", 0)
    gstep.step(*batch)
    assert getattr(p, "_nats_update_epoch", 0) > e0
