"""Layer registry — parity with the reference's string-keyed layer table
(nats.py:106-114: ``layers = {'ff': ..., 'gru': ..., 'gru_cond': ...}``,
``get_layer(name) -> (param_init, feedforward)``).

The reference used this indirection to let ``options['encoder']`` /
``options['decoder']`` select cell types; our model honours the same
option keys and this registry exposes the same lookup for extensions."""

import torch

from ..ops import eager


def param_init_ff(options, params, prefix="ff", nin=None, nout=None,
                  ortho=True):
    from .init import norm_weight
    import numpy
    nin = nin if nin is not None else options["dim"]
    nout = nout if nout is not None else options["dim"]
    params[prefix + "_W"] = norm_weight(nin, nout, scale=0.01, ortho=ortho)
    params[prefix + "_b"] = numpy.zeros((nout,), dtype="float32")
    return params


def fflayer(params, state_below, options, prefix="ff", activ=torch.tanh):
    """activ(x @ W + b) (nats.py:263-267; activ is a callable here, not an
    eval'd lambda string)."""
    return activ(state_below @ params[prefix + "_W"] + params[prefix + "_b"])


def gru_layer(params, state_below, options, prefix="gru", mask=None):
    """Registry-level GRU application (gru_layer, nats.py:305-374)."""
    xg = state_below @ params[prefix + "_W"] + params[prefix + "_b"]
    xc = state_below @ params[prefix + "_Wx"] + params[prefix + "_bx"]
    from .. import ops
    return [ops.gru_scan(xg, xc, mask, params[prefix + "_U"],
                         params[prefix + "_Ux"])]


def gru_cond_layer(params, state_below, options, prefix="decoder", mask=None,
                   context=None, context_mask=None, one_step=False,
                   init_state=None, acc_ctx=None, acc_alpha=None):
    """Registry-level conditional-GRU application (nats.py:454-609)."""
    assert context is not None, "Context must be provided"
    assert context.dim() == 3, "Context must be 3-d"
    if one_step:
        assert init_state is not None, "previous state must be provided"
    xg = state_below @ params[prefix + "_W"] + params[prefix + "_b"]
    xc = state_below @ params[prefix + "_Wx"] + params[prefix + "_bx"]
    pctx = context @ params[prefix + "_Wc_att"] + params[prefix + "_b_att"]
    from .. import ops
    if one_step:
        return list(eager.cond_gru_step(init_state, xg, xc, context,
                                        context_mask, pctx, acc_ctx,
                                        acc_alpha, params))
    B = state_below.shape[1]
    if init_state is None:
        init_state = state_below.new_zeros(B, options["dim"])
    return list(ops.cond_gru_scan(xg, xc, mask, init_state, context,
                                  context_mask, pctx, params))


layers = {
    "ff": (param_init_ff, fflayer),
    "gru": (None, gru_layer),
    "gru_cond": (None, gru_cond_layer),
}


def get_layer(name):
    """(param_init, feedforward) pair for a registered layer name."""
    return layers[name]


def dropout_layer(state_before, use_noise, p=0.5, generator=None):
    """Binomial dropout with the reference's switch semantics
    (nats.py:50-63: scale by p at test time instead of inverted dropout).
    Present for capability parity — the reference never invokes it in any
    graph, and neither does the default model here."""
    if use_noise:
        mask = torch.bernoulli(torch.full_like(state_before, 1.0 - p),
                               generator=generator)
        return state_before * mask
    return state_before * (1.0 - p)
