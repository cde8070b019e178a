"""Property-based tests (hypothesis) for the data layer.

Pin the invariants of prepare_data (nats.py:200-247), the dictionary
format (build_dictionary.py:9-35) and the iterator (data_iterator.py)
over generated corpora, beyond the hand-picked cases in test_data.py.
"""

import numpy
from hypothesis import given, settings, strategies as st

from nats_amd.data.dictionary import dictionary_from_freqs
from nats_amd.data.prepare import prepare_data

seqs = st.lists(
    st.lists(st.integers(min_value=2, max_value=40), min_size=1, max_size=20),
    min_size=1, max_size=8)


@given(xs=seqs, ys=seqs)
@settings(max_examples=60, deadline=None)
def test_prepare_data_shapes_and_masks(xs, ys):
    n = min(len(xs), len(ys))
    xs, ys = xs[:n], ys[:n]
    x, x_mask, y, y_mask = prepare_data(xs, ys)
    B = len(xs)
    assert x.shape == (max(len(s) for s in xs) + 1, B)
    assert y.shape == (max(len(s) for s in ys) + 1, B)
    assert x.dtype == numpy.int64 and x_mask.dtype == numpy.float32
    for j, (sx, sy) in enumerate(zip(xs, ys)):
        # tokens followed by the 0 (eos) pad; mask covers len+1 (the eos
        # step is TRAINED on — nats.py:243,245)
        assert list(x[:len(sx), j]) == sx
        assert (x[len(sx):, j] == 0).all()
        assert x_mask[:, j].sum() == len(sx) + 1
        assert list(y[:len(sy), j]) == sy
        assert y_mask[:, j].sum() == len(sy) + 1


@given(xs=seqs, ys=seqs, maxlen=st.integers(min_value=1, max_value=25))
@settings(max_examples=60, deadline=None)
def test_prepare_data_maxlen_truncates_not_drops(xs, ys, maxlen):
    """maxlen TRUNCATES over-long pairs (keep first maxlen-1 source tokens,
    nats.py:211-223) rather than dropping them."""
    n = min(len(xs), len(ys))
    xs, ys = xs[:n], ys[:n]
    out = prepare_data(xs, ys, maxlen=maxlen)
    if maxlen <= 1:
        # nothing fits below the cutoff -> empty batch sentinel
        if out[0] is None:
            return
    x, x_mask, y, y_mask = out
    assert x is not None
    assert len(xs) == x.shape[1]  # nothing dropped
    assert x.shape[0] <= maxlen + 1
    assert y.shape[0] <= maxlen + 1
    for j, sx in enumerate(xs):
        keep = min(len(sx), maxlen - 1) if len(sx) >= maxlen else len(sx)
        assert list(x[:keep, j]) == sx[:keep]


@given(words=st.lists(st.text(alphabet="abcdef", min_size=1, max_size=4),
                      min_size=1, max_size=60))
@settings(max_examples=60, deadline=None)
def test_dictionary_invariants(words):
    freqs = {}
    for w in words:
        freqs[w] = freqs.get(w, 0) + 1
    d = dictionary_from_freqs(freqs)
    assert d["eos"] == 0 and d["UNK"] == 1
    ids = [v for k, v in d.items() if k not in ("eos", "UNK")]
    # ids are 2..n+1, dense, assigned by descending frequency
    assert sorted(ids) == list(range(2, 2 + len(set(words) - {"eos", "UNK"})))
    by_id = sorted(((v, k) for k, v in d.items() if v >= 2))
    last = float("inf")
    for _, w in by_id:
        assert freqs[w] <= last
        last = freqs[w]


@given(data=st.data())
@settings(max_examples=30, deadline=None)
def test_iterator_epoch_roundtrip(data):
    import os
    import tempfile

    from nats_amd.data.iterator import TextIterator

    n_lines = data.draw(st.integers(min_value=1, max_value=12))
    batch = data.draw(st.integers(min_value=1, max_value=5))
    words = ["w%d" % i for i in range(10)]
    rng = numpy.random.RandomState(data.draw(st.integers(0, 1000)))
    d = tempfile.mkdtemp()
    src = os.path.join(d, "s.txt")
    tgt = os.path.join(d, "t.txt")
    with open(src, "w") as f1, open(tgt, "w") as f2:
        for _ in range(n_lines):
            f1.write(" ".join(rng.choice(words, size=3)) + "\n")
            f2.write(" ".join(rng.choice(words, size=2)) + "\n")
    wd = {w: i + 2 for i, w in enumerate(words)}
    wd["eos"] = 0
    wd["UNK"] = 1
    it = TextIterator(src, tgt, wd, batch_size=batch)
    for epoch in range(2):  # auto-reset yields identical epochs
        seen = [len(xs) for xs, ys in it]
        assert sum(seen) == n_lines
        # every batch full except possibly the last
        assert all(c == batch for c in seen[:-1])
