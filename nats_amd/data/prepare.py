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


def prepare_data(seqs_x, seqs_y, maxlen=None, n_words=30000):
    lengths_x = [len(s) for s in seqs_x]
    lengths_y = [len(s) for s in seqs_y]

    if maxlen is not None:
        new_seqs_x, new_lengths_x = [], []
        new_seqs_y, new_lengths_y = [], []
        for l_x, s_x, l_y, s_y in zip(lengths_x, seqs_x, lengths_y, seqs_y):
            if l_x >= maxlen:
                new_seqs_x.append(s_x[: maxlen - 1])
                new_lengths_x.append(maxlen - 1)
            else:
                new_seqs_x.append(s_x)
                new_lengths_x.append(l_x)
            if l_y >= maxlen:
                new_seqs_y.append(s_y[: maxlen - 1])
                new_lengths_y.append(maxlen - 1)
            else:
                new_seqs_y.append(s_y)
                new_lengths_y.append(l_y)
        lengths_x, seqs_x = new_lengths_x, new_seqs_x
        lengths_y, seqs_y = new_lengths_y, new_seqs_y

        if len(lengths_x) < 1 or len(lengths_y) < 1:
            return None, None, None, None

    n_samples = len(seqs_x)
    maxlen_x = numpy.max(lengths_x) + 1
    maxlen_y = numpy.max(lengths_y) + 1

    x = numpy.zeros((maxlen_x, n_samples), dtype="int64")
    y = numpy.zeros((maxlen_y, n_samples), dtype="int64")
    x_mask = numpy.zeros((maxlen_x, n_samples), dtype="float32")
    y_mask = numpy.zeros((maxlen_y, n_samples), dtype="float32")
    for idx, (s_x, s_y) in enumerate(zip(seqs_x, seqs_y)):
        x[: lengths_x[idx], idx] = s_x
        x_mask[: lengths_x[idx] + 1, idx] = 1.0
        y[: lengths_y[idx], idx] = s_y
        y_mask[: lengths_y[idx] + 1, idx] = 1.0

    return x, x_mask, y, y_mask
