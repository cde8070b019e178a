"""Minibatch padding/masking.

Behavioural contract (reference: scripts/nats.py:200-247):
  * sequences with len >= maxlen are TRUNCATED to their first maxlen-1
    tokens (kept, not dropped),
  * arrays are time-major ``(T, B)``; ``T = max(lengths) + 1`` so every
    sequence has at least one trailing 0 (= eos),
  * ``x``/``y`` are int64 token ids, zero-padded,
  * masks are float32 and cover ``length + 1`` positions — the extra step
    makes the model account for the EOS token,
  * returns ``(None, None, None, None)`` when truncation empties the batch
    (only possible for maxlen<=0 in practice; kept for parity).
"""

import numpy


def _truncate(seq, maxlen):
    return seq[: maxlen - 1] if len(seq) >= maxlen else seq


def _pack(seqs):
    """Column-per-sequence time-major pack: (T, B) int64 ids (zero = eos
    padding) and (T, B) float32 mask covering length+1 steps."""
    n_cols = len(seqs)
    n_rows = max(len(s) for s in seqs) + 1
    ids = numpy.zeros((n_rows, n_cols), dtype="int64")
    mask = numpy.zeros((n_rows, n_cols), dtype="float32")
    for col, seq in enumerate(seqs):
        ids[: len(seq), col] = seq
        mask[: len(seq) + 1, col] = 1.0
    return ids, mask


def prepare_data(seqs_x, seqs_y, maxlen=None, n_words=30000):
    if maxlen is not None:
        seqs_x = [_truncate(s, maxlen) for s in seqs_x]
        seqs_y = [_truncate(s, maxlen) for s in seqs_y]
        if not seqs_x or not seqs_y:
            return None, None, None, None

    x, x_mask = _pack(seqs_x)
    y, y_mask = _pack(seqs_y)
    return x, x_mask, y, y_mask
