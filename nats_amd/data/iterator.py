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


def _open_text(path):
    if path.endswith(".gz"):
        return gzip.open(path, "rt")
    return open(path, "r")


class TextIterator:
    """Simple bitext iterator (behavioural rebuild of data_iterator.py)."""

    def __init__(self, source, target, dict, batch_size=128, n_words=-1):
        self.source = _open_text(source)
        self.target = _open_text(target)
        self.dict = load_dictionary(dict) if isinstance(dict, str) else dict
        self.batch_size = batch_size
        self.n_words = n_words
        # Set when EOF was reached while a partial batch was still
        # delivered; the following __next__ ends the epoch.
        self._exhausted = False

    def __iter__(self):
        return self

    def reset(self):
        self.source.seek(0)
        self.target.seek(0)

    def _encode(self, line):
        """Token line -> id list: dictionary lookup, UNK=1, n_words cap."""
        ids = [self.dict.get(tok, 1) for tok in line.split()]
        if self.n_words > 0:
            ids = [tok_id if tok_id < self.n_words else 1 for tok_id in ids]
        return ids

    def _read_pair(self):
        """Next (source, target) id-list pair, or None at end of either
        file (a ragged tail line in one file is dropped, as in the
        reference's paired readline loop)."""
        src_line = self.source.readline()
        tgt_line = self.target.readline()
        if src_line == "" or tgt_line == "":
            return None
        return self._encode(src_line), self._encode(tgt_line)

    def __next__(self):
        def _end_epoch():
            self._exhausted = False
            self.reset()
            raise StopIteration

        if self._exhausted:
            _end_epoch()
        batch_src, batch_tgt = [], []
        while len(batch_src) < self.batch_size:
            pair = self._read_pair()
            if pair is None:
                self._exhausted = True
                break
            batch_src.append(pair[0])
            batch_tgt.append(pair[1])
        if not batch_src:
            _end_epoch()
        return batch_src, batch_tgt

    # py2-style alias kept for API parity with the reference
    next = __next__
