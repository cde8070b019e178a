import os
import pickle

import numpy

from nats_amd.data.dictionary import (build_dictionary, dictionary_from_freqs,
                                      invert_dictionary, load_dictionary)
from nats_amd.data.iterator import TextIterator
from nats_amd.data.prepare import prepare_data


def test_dictionary_ids(tmp_path):
    corpus = tmp_path / "c.txt"
    corpus.write_text("a b b c c c\nc a\n")
    d = build_dictionary(str(corpus))
    assert d["eos"] == 0 and d["UNK"] == 1
    # c (freq 4) must come before b (2) and a (2)
    assert d["c"] == 2
    assert set([d["a"], d["b"]]) == {3, 4}
    # pickle round-trips as OrderedDict
    loaded = load_dictionary(str(corpus) + ".pkl")
    assert list(loaded.items()) == list(d.items())


def test_invert_dictionary():
    d = dictionary_from_freqs({"x": 5, "y": 1})
    inv = invert_dictionary(d, with_specials=True)
    assert inv[0] == "<eos>" and inv[1] == "UNK"
    assert inv[2] == "x"


def _write_bitext(tmp_path, n):
    src = tmp_path / "s.txt"
    tgt = tmp_path / "t.txt"
    src.write_text("".join("w%d a\n" % i for i in range(n)))
    tgt.write_text("".join("a w%d\n" % i for i in range(n)))
    d = dictionary_from_freqs({"a": 100, **{"w%d" % i: 1 for i in range(n)}})
    dic = tmp_path / "d.pkl"
    with open(dic, "wb") as f:
        pickle.dump(d, f)
    return str(src), str(tgt), str(dic), d


def test_iterator_batches_and_reset(tmp_path):
    src, tgt, dic, d = _write_bitext(tmp_path, 7)
    it = TextIterator(src, tgt, dic, batch_size=3)
    batches = list(it)
    # 3 + 3 + 1 (partial final batch IS yielded)
    assert [len(b[0]) for b in batches] == [3, 3, 1]
    # auto-reset: second epoch identical
    batches2 = list(it)
    assert [len(b[0]) for b in batches2] == [3, 3, 1]
    assert batches[0][0] == batches2[0][0]


def test_iterator_unk_and_cutoff(tmp_path):
    src, tgt, dic, d = _write_bitext(tmp_path, 3)
    # cutoff: ids >= 4 -> UNK(1)
    it = TextIterator(src, tgt, dic, batch_size=10, n_words=4)
    s, t = next(it)
    for seq in s:
        assert all(tok < 4 for tok in seq)
    # unknown words map to 1
    with open(src, "w") as f:
        f.write("zzz a\n")
    it2 = TextIterator(src, tgt, dic, batch_size=10)
    s2, _ = next(it2)
    assert s2[0][0] == 1


def test_prepare_data_layout():
    x, x_mask, y, y_mask = prepare_data([[5, 6, 7], [8]], [[3], [4, 5]])
    # time-major, padded to max_len + 1 (eos slot)
    assert x.shape == (4, 2) and y.shape == (3, 2)
    assert x.dtype == numpy.int64 and x_mask.dtype == numpy.float32
    assert x[:, 0].tolist() == [5, 6, 7, 0]
    assert x[:, 1].tolist() == [8, 0, 0, 0]
    # mask covers length+1 (includes the EOS step)
    assert x_mask[:, 0].tolist() == [1, 1, 1, 1]
    assert x_mask[:, 1].tolist() == [1, 1, 0, 0]
    assert y_mask[:, 0].tolist() == [1, 1, 0]
    assert y_mask[:, 1].tolist() == [1, 1, 1]


def test_prepare_data_truncation():
    # sequences with len >= maxlen are cut to maxlen-1, not dropped
    x, x_mask, y, y_mask = prepare_data([[1] * 10], [[2] * 3], maxlen=5)
    assert x.shape[0] == 5  # 4 kept tokens + eos slot
    assert x_mask[:, 0].sum() == 5
    assert y.shape[0] == 4


def test_iterator_gzip(tmp_path):
    import gzip
    src = tmp_path / "s.txt.gz"
    tgt = tmp_path / "t.txt.gz"
    with gzip.open(src, "wt") as f:
        f.write("a b\nb a\n")
    with gzip.open(tgt, "wt") as f:
        f.write("b\na\n")
    d = dictionary_from_freqs({"a": 2, "b": 2})
    dic = tmp_path / "d.pkl"
    with open(dic, "wb") as f:
        pickle.dump(d, f)
    it = TextIterator(str(src), str(tgt), str(dic), batch_size=4)
    s, t = next(it)
    assert len(s) == 2 and len(t) == 2


def test_gen_map_line_char_level():
    from nats_amd.decode.driver import map_line
    d = {"a": 2, "b": 3, "ab": 4}
    # word level: "ab" is one token
    assert map_line("ab", d, 100) == [4, 0]
    # char level (-c): per-character ids, unknown -> 1
    assert map_line("ab", d, 100, chr_level=True) == [2, 3, 0]
    # n_words cutoff -> UNK
    assert map_line("b", d, 3) == [1, 0]


def test_synthetic_batch_layout():
    from nats_amd.data.synthetic import synthetic_batch
    rng = numpy.random.RandomState(0)
    x, xm, y, ym = synthetic_batch(rng, 3, 10, 4, 50)
    assert x.shape == (11, 3) and y.shape == (5, 3)
    assert (x[:10] >= 2).all() and (x[10] == 0).all()
    assert xm.dtype == numpy.float32 and (xm == 1).all()


def test_seqs2words_oov_ids_map_to_unk():
    """Model-emitted ids beyond the dictionary (n_words > corpus vocab)
    must render as UNK, not crash (found by the GPU e2e pipeline: a
    42-word toy corpus trained at n_words=64 emits ids 42..63)."""
    from nats_amd.decode.driver import seqs2words
    word_idict = {0: "<eos>", 1: "UNK", 2: "alpha", 3: "beta"}
    lines = seqs2words([[2, 45, 3, 0]], [[0, 1, 2, 3]], word_idict)
    assert lines == ["alpha [0] UNK [1] beta [2]"]
