"""Dictionary building/loading.

Format contract (reference: data/build_dictionary.py:9-35):
  * words are counted in first-seen order over whitespace-split lines,
  * sorted by frequency descending (ties broken by numpy argsort order over
    the first-seen sequence, reversed — reproduced exactly),
  * ids assigned: 'eos' -> 0, 'UNK' -> 1, the rest from 2,
  * pickled as a ``collections.OrderedDict`` to ``<corpus>.pkl``.

The pickle must round-trip with the reference's checkpoints, so we keep an
OrderedDict and protocol compatibility.
"""

import pickle
from collections import OrderedDict

import numpy


def count_words(lines):
    """Count word frequencies in first-seen order.

    Splits on single spaces like the reference (build_dictionary.py:14),
    i.e. ``line.strip().split(' ')`` — empty lines contribute one '' token,
    exactly as the reference does.
    """
    word_freqs = OrderedDict()
    for line in lines:
        for w in line.strip().split(" "):
            if w not in word_freqs:
                word_freqs[w] = 0
            word_freqs[w] += 1
    return word_freqs


def dictionary_from_freqs(word_freqs):
    """freq table -> OrderedDict word->id with eos=0, UNK=1, rest by freq desc.

    Tie-breaking mirrors build_dictionary.py:22-24: ``numpy.argsort(freqs)``
    (stable order not guaranteed by reference either — numpy's default
    quicksort — but for identical input the result is identical) reversed.
    """
    words = list(word_freqs.keys())
    freqs = list(word_freqs.values())
    sorted_idx = numpy.argsort(freqs)
    sorted_words = [words[ii] for ii in sorted_idx[::-1]]

    worddict = OrderedDict()
    worddict["eos"] = 0
    worddict["UNK"] = 1
    for ii, ww in enumerate(sorted_words):
        worddict[ww] = ii + 2
    return worddict


def build_dictionary(filename, saveto=None):
    """Build and pickle the dictionary for a corpus file.

    Returns the OrderedDict. Writes ``<filename>.pkl`` (or *saveto*).
    """
    with open(filename, "r", encoding="utf-8", errors="replace") as f:
        word_freqs = count_words(f)
    worddict = dictionary_from_freqs(word_freqs)
    out = saveto if saveto is not None else "%s.pkl" % filename
    with open(out, "wb") as f:
        pickle.dump(worddict, f)
    return worddict


def load_dictionary(path):
    """Load a pickled word->id dictionary (py2 or py3 pickles)."""
    with open(path, "rb") as f:
        try:
            return pickle.load(f)
        except UnicodeDecodeError:
            f.seek(0)
            return pickle.load(f, encoding="latin-1")


def invert_dictionary(worddict, with_specials=False):
    """id->word map (nats.py:1266-1268; gen.py:71-75 adds the specials)."""
    word_idict = {}
    for kk, vv in worddict.items():
        word_idict[vv] = kk
    if with_specials:
        word_idict[0] = "<eos>"
        word_idict[1] = "UNK"
    return word_idict
