import math

import numpy
import pytest
import torch

from nats_amd.engine.optim import (Adadelta, Adam, RMSProp, SGD,
                                   build_optimizer, clip_grads_global_norm)


def _mkparam(v):
    p = torch.nn.Parameter(torch.tensor(v, dtype=torch.float32))
    return p


def test_clip_global_norm():
    p1, p2 = _mkparam([3.0, 0.0]), _mkparam([0.0, 4.0])
    p1.grad = torch.tensor([3.0, 0.0])
    p2.grad = torch.tensor([0.0, 4.0])
    # ||g|| = 5 > 1 -> scaled to norm 1
    norm = clip_grads_global_norm([p1, p2], clip_c=1.0)
    assert abs(norm - 5.0) < 1e-6
    total = math.sqrt(float(p1.grad.pow(2).sum() + p2.grad.pow(2).sum()))
    assert abs(total - 1.0) < 1e-6
    # below threshold: untouched
    p1.grad = torch.tensor([0.1, 0.0])
    p2.grad = torch.tensor([0.0, 0.1])
    clip_grads_global_norm([p1, p2], clip_c=1.0)
    assert abs(float(p1.grad[0]) - 0.1) < 1e-7


def test_adadelta_formula():
    """One step vs hand numpy (nats.py:1145-1173; rho=.95 eps=1e-6)."""
    p = _mkparam([1.0, -2.0])
    g = numpy.array([0.5, -0.25], dtype="float64")
    p.grad = torch.tensor(g, dtype=torch.float32)
    opt = Adadelta([("p", p)])
    opt.step()
    rg2 = 0.05 * g * g
    ud = -numpy.sqrt(0.0 + 1e-6) / numpy.sqrt(rg2 + 1e-6) * g
    expect = numpy.array([1.0, -2.0]) + ud
    numpy.testing.assert_allclose(p.detach().numpy(), expect, rtol=1e-5)
    # second step exercises both EMAs
    p.grad = torch.tensor(g, dtype=torch.float32)
    opt.step()
    rg2_2 = 0.95 * rg2 + 0.05 * g * g
    ru2 = 0.05 * ud * ud
    ud2 = -numpy.sqrt(ru2 + 1e-6) / numpy.sqrt(rg2_2 + 1e-6) * g
    numpy.testing.assert_allclose(p.detach().numpy(), expect + ud2, rtol=1e-5)


def test_adam_nonstandard_constants():
    """The reference's adam: lr0=2e-4 hardcoded, b1=.1, b2=.001
    (nats.py:1114-1117) — lr argument ignored."""
    p = _mkparam([0.0])
    g = 2.0
    p.grad = torch.tensor([g])
    opt = Adam([("p", p)], lrate=123.0)  # lrate must be ignored
    opt.step()
    b1, b2, lr0, e = 0.1, 0.001, 2e-4, 1e-8
    fix1, fix2 = 1 - b1 ** 1, 1 - b2 ** 1
    lr_t = lr0 * math.sqrt(fix2) / fix1
    m_t = b1 * g
    v_t = b2 * g * g
    expect = -lr_t * m_t / (math.sqrt(v_t) + e)
    assert abs(float(p) - expect) < 1e-9


def test_rmsprop_formula():
    p = _mkparam([1.0])
    g = 0.3
    p.grad = torch.tensor([g])
    opt = RMSProp([("p", p)])
    opt.step()
    rg, rg2 = 0.05 * g, 0.05 * g * g
    ud = -1e-4 * g / math.sqrt(rg2 - rg ** 2 + 1e-4)
    assert abs(float(p) - (1.0 + ud)) < 1e-7


def test_sgd():
    p = _mkparam([1.0])
    p.grad = torch.tensor([0.5])
    opt = SGD([("p", p)], lrate=0.1)
    opt.step()
    assert abs(float(p) - 0.95) < 1e-7


def test_build_optimizer_names():
    p = _mkparam([1.0])
    for name, cls in [("adadelta", Adadelta), ("adam", Adam),
                      ("rmsprop", RMSProp), ("sgd", SGD)]:
        opt = build_optimizer(name, [("p", p)])
        assert isinstance(opt, cls)
    with pytest.raises(ValueError):
        build_optimizer("lamb", [("p", p)])


def test_optimizer_state_roundtrip(tmp_path):
    from nats_amd.engine.checkpoint import (load_optimizer_state,
                                            save_optimizer_state)
    p = _mkparam([1.0, 2.0])
    opt = Adadelta([("p", p)])
    p.grad = torch.tensor([0.1, 0.2])
    opt.step()
    saveto = str(tmp_path / "m.npz")
    save_optimizer_state(saveto, opt)
    p2 = _mkparam([1.0, 2.0])
    opt2 = Adadelta([("p", p2)])
    assert load_optimizer_state(saveto, opt2)
    assert opt2.t == opt.t
    torch.testing.assert_close(opt2.state["p"]["rg2"], opt.state["p"]["rg2"])
