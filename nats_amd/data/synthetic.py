"""Synthetic corpora.

Two uses:
  * ``make_toy_corpus`` — a deterministic toy doc->summary corpus with the
    reference's file layout (200 train / 40 valid / 40 test pairs,
    ``<EOS>``-separated sentences — cf. data/toy_*.txt in the reference,
    regenerated rather than copied). The summary is the lead sentence with
    a fixed transformation, so a few hundred updates visibly reduce loss.
  * ``synthetic_batch`` — fixed-shape random-token batches for bench.py
    (no network: BASELINE.json prescribes synthetic LCSTS/CNN-DM-shaped
    data with random-init weights).
"""

import os

import numpy


_TOPICS = ["market", "storm", "team", "election", "study", "festival",
           "court", "factory", "river", "museum"]
_VERBS = ["reports", "announces", "confirms", "denies", "expands",
          "reviews", "wins", "loses", "opens", "closes"]
_NOUNS = ["officials", "results", "plans", "records", "figures", "workers",
          "visitors", "experts", "leaders", "students"]
_FILLER = ["the", "a", "new", "local", "major", "annual", "recent", "small",
           "large", "early"]


def _sentence(rng, topic):
    words = [topic, rng.choice(_VERBS), rng.choice(_FILLER),
             rng.choice(_NOUNS)]
    n_extra = rng.randint(2, 8)
    for _ in range(n_extra):
        words.append(rng.choice(_FILLER + _NOUNS))
    return " ".join(words)


def _pair(rng):
    topic = rng.choice(_TOPICS)
    lead = _sentence(rng, topic)
    n_body = rng.randint(2, 5)
    body = [_sentence(rng, rng.choice(_TOPICS)) for _ in range(n_body)]
    doc = " <EOS> ".join([lead] + body) + " <EOS>"
    summary = lead + " <EOS>"
    return doc, summary


def make_toy_corpus(outdir, n_train=200, n_valid=40, n_test=40, seed=1234):
    """Write toy_{train,validation,test}_{input,output}.txt + dictionary."""
    from .dictionary import build_dictionary
    rng = numpy.random.RandomState(seed)
    os.makedirs(outdir, exist_ok=True)
    splits = [("train", n_train), ("validation", n_valid), ("test", n_test)]
    for split, n in splits:
        src = os.path.join(outdir, "toy_%s_input.txt" % split)
        tgt = os.path.join(outdir, "toy_%s_output.txt" % split)
        with open(src, "w") as fs, open(tgt, "w") as ft:
            for _ in range(n):
                doc, summary = _pair(rng)
                fs.write(doc + "\n")
                ft.write(summary + "\n")
    build_dictionary(os.path.join(outdir, "toy_train_input.txt"))
    return outdir


def synthetic_batch(rng, batch, src_len, tgt_len, n_words):
    """Fixed-shape random batch: (x, x_mask, y, y_mask) numpy arrays in the
    prepare_data layout (time-major, eos slot included)."""
    T_s, T_t = src_len + 1, tgt_len + 1
    x = numpy.zeros((T_s, batch), dtype="int64")
    y = numpy.zeros((T_t, batch), dtype="int64")
    x[:src_len] = rng.randint(2, n_words, size=(src_len, batch))
    y[:tgt_len] = rng.randint(2, n_words, size=(tgt_len, batch))
    x_mask = numpy.ones((T_s, batch), dtype="float32")
    y_mask = numpy.ones((T_t, batch), dtype="float32")
    return x, x_mask, y, y_mask
