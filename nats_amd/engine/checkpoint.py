"""Checkpoint I/O — exact ``.npz`` + options-pickle format parity.

Reference mechanism (nats.py:1427-1435, 1533-1535): periodic
``numpy.savez(saveto, history_errs=..., **params)`` plus a pickled options
dict at ``<saveto>.pkl``; the final save additionally embeds
``zipped_params`` (the best-validation snapshot). Resume restores options,
params (warning on missing keys, nats.py:81-89) and ``history_errs``.

We keep that layout bit-for-bit and add an OPTIONAL optimizer-state sidecar
``<saveto>.opt.npz`` (the reference restarts adadelta accumulators from
zero on resume; loading the sidecar is opt-in to preserve that behaviour).
"""

import os
import pickle
from collections import OrderedDict

import numpy


def save_options(saveto, options):
    with open("%s.pkl" % saveto, "wb") as f:
        pickle.dump(options, f)


def load_options(saveto):
    with open("%s.pkl" % saveto, "rb") as f:
        try:
            return pickle.load(f)
        except UnicodeDecodeError:
            f.seek(0)
            return pickle.load(f, encoding="latin-1")


def save_checkpoint(saveto, params, history_errs, zipped_params=None,
                    options=None):
    """numpy.savez with the canonical key set (nats.py:1433, 1533-1535)."""
    extra = {}
    if zipped_params is not None:
        # the reference stores the dict itself (pickled object array)
        extra["zipped_params"] = numpy.array(zipped_params, dtype=object)
    numpy.savez(saveto, history_errs=numpy.asarray(history_errs), **extra,
                **params)
    if options is not None:
        save_options(saveto, options)


def load_checkpoint(path):
    """Returns (params OrderedDict, history_errs list)."""
    if not os.path.exists(path) and os.path.exists(path + ".npz"):
        path = path + ".npz"
    archive = numpy.load(path, allow_pickle=True)
    params = OrderedDict()
    history = []
    for k in archive.files:
        if k == "history_errs":
            history = list(archive[k])
        elif k == "zipped_params":
            continue
        else:
            params[k] = archive[k]
    return params, history


def save_optimizer_state(saveto, optimizer):
    """Optional sidecar (not part of the reference format)."""
    sd = optimizer.state_dict()
    flat = {"__t__": numpy.asarray(sd["t"])}
    for k, st in sd["state"].items():
        for sk, sv in st.items():
            flat["%s::%s" % (k, sk)] = sv.detach().cpu().numpy()
    numpy.savez("%s.opt.npz" % saveto, **flat)


def load_optimizer_state(saveto, optimizer):
    import torch
    path = "%s.opt.npz" % saveto
    if not os.path.exists(path):
        return False
    archive = numpy.load(path)
    sd = {"t": int(archive["__t__"]), "state": {}}
    for key in archive.files:
        if key == "__t__":
            continue
        k, sk = key.split("::", 1)
        sd["state"].setdefault(k, {})[sk] = torch.from_numpy(archive[key])
    optimizer.load_state_dict(sd)
    return True
