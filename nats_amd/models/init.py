"""Parameter initialisation and the canonical ``.npz`` checkpoint schema.

The key/shape layout reproduces the reference exactly so checkpoints are
interchangeable (schema declared across nats.py:613-654 and the layer inits
at nats.py:271-302, 378-451, 251-260). All float32.

With V=n_words, E=dim_word, H=dim, A=dim_att, C=2H:

  Wemb (V,E); encoder{,_r}_{W (E,2H), b (2H,), U (H,2H), Wx (E,H), bx (H,),
  Ux (H,H)}; ff_state_{W (C,H), b (H,)}; decoder_{W (E,2H), b (2H,),
  U (H,2H), Wx (E,H), Ux (H,H), bx (H,), U_1 (H,2H), W_1 (C,2H), b_1 (2H,),
  Wx_1 (C,H), Ux_1 (H,H), bx_1 (H,), W_att (H,A), Wc_att (C,A), b_att (A,),
  U_att (A,1), c_att (1,), W_con (C,1), U_con (C,1), D_wei (1,A)};
  ff_logit_lstm_{W (H,E), b (E,)}; ff_logit_prev_{W (E,E), b (E,)};
  ff_logit_ctx_{W (C,E), b (E,)}; ff_logit_{W (E,V), b (V,)}.
"""

from collections import OrderedDict

import numpy


def ortho_weight(ndim, rng=None):
    """SVD-orthogonal square init (nats.py:118-129)."""
    rng = rng or numpy.random
    W = rng.randn(ndim, ndim)
    u, s, v = numpy.linalg.svd(W)
    return u.astype("float32")


def norm_weight(nin, nout=None, scale=0.01, ortho=True, rng=None):
    """Gaussian init, orthogonal when square (nats.py:132-142)."""
    rng = rng or numpy.random
    if nout is None:
        nout = nin
    if nout == nin and ortho:
        W = ortho_weight(nin, rng)
    else:
        W = scale * rng.randn(nin, nout)
    return W.astype("float32")


def _gru_shapes(prefix, nin, dim):
    return [
        (prefix + "_W", (nin, 2 * dim)),
        (prefix + "_b", (2 * dim,)),
        (prefix + "_U", (dim, 2 * dim)),
        (prefix + "_Wx", (nin, dim)),
        (prefix + "_bx", (dim,)),
        (prefix + "_Ux", (dim, dim)),
    ]


def param_shapes(options):
    """Ordered (key, shape) list for a given options dict."""
    V = options["n_words"]
    E = options["dim_word"]
    H = options["dim"]
    A = options["dim_att"]
    C = 2 * H
    shapes = [("Wemb", (V, E))]
    shapes += _gru_shapes("encoder", E, H)
    shapes += _gru_shapes("encoder_r", E, H)
    # stacked bi-GRU encoder layers (framework extension beyond the
    # single-layer reference; enc_depth=1 keeps the exact reference
    # schema — BASELINE configs[4] uses enc_depth=4 at dim 2048)
    for l in range(1, options.get("enc_depth", 1)):
        shapes += _gru_shapes("encoder_l%d" % l, C, H)
        shapes += _gru_shapes("encoder_r_l%d" % l, C, H)
    shapes += [("ff_state_W", (C, H)), ("ff_state_b", (H,))]
    # decoder: GRU_2 (input = y embedding)
    shapes += [
        ("decoder_W", (E, 2 * H)),
        ("decoder_U", (H, 2 * H)),
        ("decoder_b", (2 * H,)),
        ("decoder_Wx", (E, H)),
        ("decoder_Ux", (H, H)),
        ("decoder_bx", (H,)),
        # GRU_1 (input = content vector)
        ("decoder_U_1", (H, 2 * H)),
        ("decoder_W_1", (C, 2 * H)),
        ("decoder_b_1", (2 * H,)),
        ("decoder_Wx_1", (C, H)),
        ("decoder_Ux_1", (H, H)),
        ("decoder_bx_1", (H,)),
        # attention MLP
        ("decoder_W_att", (H, A)),
        ("decoder_Wc_att", (C, A)),
        ("decoder_b_att", (A,)),
        ("decoder_U_att", (A, 1)),
        ("decoder_c_att", (1,)),
        # distraction
        ("decoder_W_con", (C, 1)),
        ("decoder_U_con", (C, 1)),
        ("decoder_D_wei", (1, A)),
    ]
    shapes += [
        ("ff_logit_lstm_W", (H, E)),
        ("ff_logit_lstm_b", (E,)),
        ("ff_logit_prev_W", (E, E)),
        ("ff_logit_prev_b", (E,)),
        ("ff_logit_ctx_W", (C, E)),
        ("ff_logit_ctx_b", (E,)),
        ("ff_logit_W", (E, V)),
        ("ff_logit_b", (V,)),
    ]
    return shapes


# Static copy of the schema (shapes symbolic) for documentation/tests.
PARAM_SCHEMA = [k for k, _ in param_shapes(
    dict(n_words=1, dim_word=1, dim=1, dim_att=1))]


# Gate-stacked matrices: two independently drawn (nin, H) blocks side by
# side (param_init_gru nats.py:283-291; param_init_gru_cond nats.py:392-411).
_GATE_STACK_NORM = {"encoder_W", "encoder_r_W", "decoder_W"}
_GATE_STACK_ORTHO = {"encoder_U", "encoder_r_U", "decoder_U", "decoder_U_1"}
# Readout layers created with ortho=False (nats.py:641-649) — matters for the
# square ff_logit_prev_W, which is Gaussian rather than orthogonal.
_NO_ORTHO = {"ff_logit_lstm_W", "ff_logit_prev_W", "ff_logit_ctx_W"}


def _init_one(key, shape, rng):
    """Initialise one tensor per the reference's distribution choices."""
    if len(shape) == 1:
        return numpy.zeros(shape, dtype="float32")
    nin, nout = shape
    if key in _GATE_STACK_NORM:
        h = nout // 2
        return numpy.concatenate(
            [norm_weight(nin, h, rng=rng), norm_weight(nin, h, rng=rng)], axis=1)
    if key in _GATE_STACK_ORTHO:
        return numpy.concatenate([ortho_weight(nin, rng), ortho_weight(nin, rng)],
                                 axis=1)
    if key in _NO_ORTHO:
        return norm_weight(nin, nout, ortho=False, rng=rng)
    # decoder_W_1 is a single norm_weight(C, 2H) draw (nats.py:412), and every
    # remaining matrix is norm_weight with the default ortho=True (orthogonal
    # when square — e.g. the Ux recurrences).
    return norm_weight(nin, nout, rng=rng)


def init_params(options, seed=None):
    """Create the full parameter OrderedDict (float32 numpy arrays).

    Mirrors init_params (nats.py:613-654). Distribution per tensor:
    square matrices are SVD-orthogonal; stacked gate matrices are two
    independently-drawn blocks; everything else N(0, 0.01); biases zero;
    the three middle readout projections use ortho=False.
    """
    rng = numpy.random.RandomState(seed) if seed is not None else numpy.random
    params = OrderedDict()
    for key, shape in param_shapes(options):
        params[key] = _init_one(key, shape, rng)
    return params
