"""Sequential bitext iterator.

Behavioural contract (reference: scripts/data_iterator.py:11-80):
  * reads source/target line-pairs sequentially, no shuffling,
  * maps words through the pickled dictionary, unknown -> 1 (UNK),
  * applies the ``n_words`` cutoff: id >= n_words -> 1,
  * yields ``(source, target)`` lists of token-id lists of up to
    ``batch_size`` pairs; the final partial batch IS yielded,
  * after EOF raises StopIteration once and auto-resets, so the same
    object can be iterated for the next epoch,
  * ``.gz`` files are transparently decompressed.
"""

import gzip

from .dictionary import load_dictionary


def fopen(filename, mode="r"):
    if filename.endswith(".gz"):
        return gzip.open(filename, mode + "t")
    return open(filename, mode)


class TextIterator:
    """Simple bitext iterator (API-compatible rebuild of data_iterator.py)."""

    def __init__(self, source, target, dict, batch_size=128, n_words=-1):
        self.source = fopen(source, "r")
        self.target = fopen(target, "r")
        if isinstance(dict, str):
            self.dict = load_dictionary(dict)
        else:
            self.dict = dict
        self.batch_size = batch_size
        self.n_words = n_words
        self.end_of_data = False

    def __iter__(self):
        return self

    def reset(self):
        self.source.seek(0)
        self.target.seek(0)

    def _map_line(self, line):
        toks = line.strip().split()
        ids = [self.dict[w] if w in self.dict else 1 for w in toks]
        if self.n_words > 0:
            ids = [w if w < self.n_words else 1 for w in ids]
        return ids

    def __next__(self):
        if self.end_of_data:
            self.end_of_data = False
            self.reset()
            raise StopIteration

        source = []
        target = []
        try:
            while True:
                ss = self.source.readline()
                if ss == "":
                    raise IOError
                tt = self.target.readline()
                if tt == "":
                    raise IOError
                source.append(self._map_line(ss))
                target.append(self._map_line(tt))
                if len(source) >= self.batch_size or len(target) >= self.batch_size:
                    break
        except IOError:
            self.end_of_data = True

        if len(source) <= 0 or len(target) <= 0:
            self.end_of_data = False
            self.reset()
            raise StopIteration

        return source, target

    # py2-style alias kept for API parity with the reference
    next = __next__
