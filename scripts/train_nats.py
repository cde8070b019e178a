#!/usr/bin/env python3
"""Config driver for training — entrypoint parity with the reference's
scripts/train_nats.py:6-48 (same Spearmint-style params dict, same
hyperparameters: dim_word=120, dim=600, dim_att=100, n_words=25000,
adadelta, clip_c=100, batch=20, maxlen=500, toy corpus, aggressive
valid/save/sample freqs). Paths default to the repo's data/ directory
(override with NATS_DATA_DIR)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# pre-tuned hipBLASLt algorithms for the readout/weight-grad GEMMs
# (read-only; see bench.py)
_TUNED = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))),
                      "nats_amd", "ops", "tunableop_gfx950.csv")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ:
    # TunableOp substitutes the device ordinal into %d (or appends it
    # when absent), so stage per-ordinal copies of the committed table
    import shutil
    import tempfile
    _tdir = tempfile.mkdtemp(prefix="nats_tunableop")
    for _i in range(8):
        shutil.copy(_TUNED, os.path.join(_tdir, "tuned_%d.csv" % _i))
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(
        _tdir, "tuned_%d.csv")

from nats_amd.engine.trainer import train

DATA = os.environ.get(
    "NATS_DATA_DIR",
    os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                 "data"))
MODELS = os.environ.get("NATS_MODEL_DIR", os.path.join(DATA, "..", "models"))


def ensure_toy_corpus(data_dir=DATA):
    """Generate the deterministic toy corpus + dictionary on first run so
    the entrypoints work on a fresh clone (the reference ships its toy
    files in-repo, train_nats.py:22-26; ours are regenerated, seed=1234)."""
    if not os.path.exists(os.path.join(data_dir, "toy_train_input.txt")) or \
            not os.path.exists(os.path.join(data_dir,
                                            "toy_train_input.txt.pkl")):
        from nats_amd.data.synthetic import make_toy_corpus
        print("Bootstrapping toy corpus into %s" % data_dir)
        make_toy_corpus(data_dir)


def main(job_id, params):
    print(params)
    validerr = train(
        saveto=params["model"][0],
        reload_=params["reload"][0],
        dim_word=params["dim_word"][0],
        dim=params["dim"][0],
        dim_att=params["dim_att"][0],
        patience=params["patience"][0],
        n_words=params["n-words"][0],
        decay_c=params["decay-c"][0],
        clip_c=params["clip-c"][0],
        lrate=params["learning-rate"][0],
        optimizer=params["optimizer"][0],
        maxlen=500,
        batch_size=20,
        valid_batch_size=20,
        datasets=[os.path.join(DATA, "toy_train_input.txt"),
                  os.path.join(DATA, "toy_train_output.txt")],
        valid_datasets=[os.path.join(DATA, "toy_validation_input.txt"),
                        os.path.join(DATA, "toy_validation_output.txt")],
        dictionary=os.path.join(DATA, "toy_train_input.txt.pkl"),
        validFreq=10,
        dispFreq=1,
        saveFreq=10,
        sampleFreq=10,
    )
    return validerr


if __name__ == "__main__":
    os.makedirs(MODELS, exist_ok=True)
    ensure_toy_corpus()
    main(0, {
        "model": [os.path.join(MODELS, "model.npz")],
        "dim_word": [120],
        "dim": [600],
        "dim_att": [100],
        "n-words": [25000],
        "patience": [1],
        "optimizer": ["adadelta"],
        "decay-c": [0.0],
        "clip-c": [100.0],
        "use-dropout": [False],
        "learning-rate": [0.0001],
        "reload": [False],
    })
