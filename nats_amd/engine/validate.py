"""Validation scoring (pred_probs, nats.py:1080-1101)."""

import math

import numpy
import torch

from ..data.prepare import prepare_data


@torch.no_grad()
def pred_probs(model, iterator, device=None, verbose=False,
               rank=0, world=1, raise_on_nan=True):
    """Per-sequence NLL over a whole corpus iterator.

    Mirrors pred_probs: NO maxlen truncation (nats.py:1088), per-sequence
    masked-CE sums collected into one array. NaN raises (the reference
    drops into ipdb, nats.py:1096 — non-interactive here).

    With world>1 the valid set is sharded round-robin by batch index: each
    rank scores 1/world of the batches, and the caller all-reduces the
    (sum, count) pair to recover the global mean — no rank wastes step
    time re-scoring the full set. raise_on_nan=False defers the NaN check
    to the caller (a per-rank raise before a collective would hang the
    other ranks; NaN survives the sum-reduce, so every rank can check the
    reduced value together).
    """
    probs = []
    n_done = 0
    options = model.options
    for bidx, (xs, ys) in enumerate(iterator):
        if world > 1 and bidx % world != rank:
            continue
        n_done += len(xs)
        x, x_mask, y, y_mask = prepare_data(xs, ys, n_words=options["n_words"])
        x = torch.from_numpy(x)
        y = torch.from_numpy(y)
        x_mask = torch.from_numpy(x_mask)
        y_mask = torch.from_numpy(y_mask)
        if device is not None:
            x, x_mask = x.to(device), x_mask.to(device)
            y, y_mask = y.to(device), y_mask.to(device)
        cost = model(x, x_mask, y, y_mask)
        batch_costs = [float(c) for c in cost.cpu()]
        if raise_on_nan and not all(math.isfinite(p) for p in batch_costs):
            raise FloatingPointError("NaN/Inf in validation cost")
        probs.extend(batch_costs)
        if verbose:
            print("%d samples computed" % n_done)
    return numpy.array(probs, dtype="float32")
