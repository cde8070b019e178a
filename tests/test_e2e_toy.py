"""End-to-end: toy corpus trains, loss falls, checkpoints write, decode +
UNK-replace + ROUGE chain runs (the reference's integration test is exactly
this loop on its toy data — SURVEY §4)."""

import os

import numpy
import pytest
import torch

from nats_amd.data.iterator import TextIterator
from nats_amd.data.prepare import prepare_data
from nats_amd.engine.optim import build_optimizer
from nats_amd.engine.trainer import train
from nats_amd.models.distraction import NatsModel, default_options


def test_overfit_one_batch(toy_corpus, tiny_options):
    """Cost on a fixed batch must drop under adadelta (the default opt)."""
    it = TextIterator(os.path.join(toy_corpus, "toy_train_input.txt"),
                      os.path.join(toy_corpus, "toy_train_output.txt"),
                      os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
                      batch_size=4, n_words=tiny_options["n_words"])
    xs, ys = next(it)
    x, x_mask, y, y_mask = [torch.from_numpy(a) for a in prepare_data(
        xs, ys, maxlen=50, n_words=tiny_options["n_words"])]
    model = NatsModel(tiny_options, seed=11)
    opt = build_optimizer("adadelta", list(model.P.items()), clip_c=1.0)
    costs = []
    for step in range(40):
        opt.zero_grad()
        cost = model(x, x_mask, y, y_mask).mean()
        cost.backward()
        opt.step()
        costs.append(float(cost))
    assert costs[-1] < costs[0] * 0.98, costs[::8]


def test_train_entrypoint(toy_corpus, tmp_path, tiny_options):
    saveto = str(tmp_path / "model.npz")
    err = train(dim_word=12, dim=16, dim_att=8, n_words=64, maxlen=50,
                batch_size=8, valid_batch_size=8, saveto=saveto,
                datasets=[os.path.join(toy_corpus, "toy_train_input.txt"),
                          os.path.join(toy_corpus, "toy_train_output.txt")],
                valid_datasets=[
                    os.path.join(toy_corpus, "toy_validation_input.txt"),
                    os.path.join(toy_corpus, "toy_validation_output.txt")],
                dictionary=os.path.join(toy_corpus,
                                        "toy_train_input.txt.pkl"),
                validFreq=10, saveFreq=6, sampleFreq=1000, dispFreq=2,
                finish_after=12, clip_c=1.0, device="cpu", seed=5)
    assert numpy.isfinite(err)
    assert os.path.exists(saveto)
    assert os.path.exists(saveto + ".pkl")
    # checkpoint must contain the canonical schema
    from nats_amd.engine.checkpoint import load_checkpoint
    params, hist = load_checkpoint(saveto)
    assert "decoder_D_wei" in params and "Wemb" in params


def test_train_reload(toy_corpus, tmp_path):
    kw = dict(dim_word=10, dim=12, dim_att=6, n_words=64, maxlen=50,
              batch_size=8, valid_batch_size=8,
              datasets=[os.path.join(toy_corpus, "toy_train_input.txt"),
                        os.path.join(toy_corpus, "toy_train_output.txt")],
              valid_datasets=[
                  os.path.join(toy_corpus, "toy_validation_input.txt"),
                  os.path.join(toy_corpus, "toy_validation_output.txt")],
              dictionary=os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
              validFreq=100, saveFreq=4, sampleFreq=1000, dispFreq=100,
              finish_after=4, device="cpu", seed=6)
    saveto = str(tmp_path / "model.npz")
    train(saveto=saveto, **kw)
    # resume: options + params + history reload (nats.py:1271-1295)
    err = train(saveto=saveto, reload_=True, **kw)
    assert numpy.isfinite(err)


def test_decode_chain(toy_corpus, tmp_path):
    """train briefly -> gen (single process) -> replace_unk -> ROUGE."""
    from nats_amd.decode.driver import generate_file
    from nats_amd.decode.replace_unk import replace_unk_files
    from nats_amd.decode.rouge import score_files
    saveto = str(tmp_path / "model.npz")
    train(dim_word=10, dim=12, dim_att=6, n_words=64, maxlen=50,
          batch_size=8, valid_batch_size=8, saveto=saveto,
          datasets=[os.path.join(toy_corpus, "toy_train_input.txt"),
                    os.path.join(toy_corpus, "toy_train_output.txt")],
          valid_datasets=[
              os.path.join(toy_corpus, "toy_validation_input.txt"),
              os.path.join(toy_corpus, "toy_validation_output.txt")],
          dictionary=os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
          validFreq=1000, saveFreq=3, sampleFreq=1000, dispFreq=100,
          finish_after=3, device="cpu", seed=7)
    temp = str(tmp_path / "temp.txt")
    final = str(tmp_path / "final.txt")
    src = os.path.join(toy_corpus, "toy_test_input.txt")
    ref = os.path.join(toy_corpus, "toy_test_output.txt")
    generate_file(saveto, os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
                  src, temp, k=2, normalize=True, n_process=1,
                  devices=["cpu"], verbose=False)
    with open(temp) as f:
        lines = f.read().splitlines()
    assert len(lines) == 40
    # every line is "word [pos]" interleaved
    for ln in lines:
        toks = ln.split()
        assert len(toks) % 2 == 0
        for p in toks[1::2]:
            assert p.startswith("[") and p.endswith("]")
    replace_unk_files(src, temp, final)
    r, p, f = score_files(ref, final, 1, "N")
    assert 0.0 <= r <= 1.0 and 0.0 <= f <= 1.0


def test_train_profile_and_resume_optimizer(toy_corpus, tmp_path):
    """profile=True collects per-section timers; resume_optimizer=True
    writes the .opt.npz sidecar at saveFreq and restores it on reload."""
    kw = dict(dim_word=10, dim=12, dim_att=6, n_words=64, maxlen=50,
              batch_size=8, valid_batch_size=8,
              datasets=[os.path.join(toy_corpus, "toy_train_input.txt"),
                        os.path.join(toy_corpus, "toy_train_output.txt")],
              valid_datasets=[
                  os.path.join(toy_corpus, "toy_validation_input.txt"),
                  os.path.join(toy_corpus, "toy_validation_output.txt")],
              dictionary=os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
              validFreq=100, saveFreq=3, sampleFreq=1000, dispFreq=2,
              finish_after=6, device="cpu", seed=7,
              profile=True, resume_optimizer=True)
    saveto = str(tmp_path / "model.npz")
    train(saveto=saveto, **kw)
    assert os.path.exists(saveto + ".opt.npz")
    # restore must not blow up and must pick up the adadelta accumulators
    archive = numpy.load(saveto + ".opt.npz")
    assert any(k.endswith("::rg2") for k in archive.files)
    err = train(saveto=saveto, reload_=True, **kw)
    assert numpy.isfinite(err)


def test_gen_multiprocess_matches_single(toy_corpus, tmp_path):
    """gen.py's worker-pool path (gen.py:111-126): 2 spawned workers must
    produce byte-identical output to the single-process path."""
    from nats_amd.decode.driver import generate_file
    from nats_amd.engine.checkpoint import save_checkpoint

    opts = default_options(dim_word=10, dim=12, dim_att=6, n_words=43,
                           maxlen=50)
    model = NatsModel(opts, seed=13)
    saveto = str(tmp_path / "model.npz")
    save_checkpoint(saveto, model.get_params(), [], options=opts)
    src = str(tmp_path / "src.txt")
    with open(os.path.join(toy_corpus, "toy_test_input.txt")) as f:
        lines = f.read().splitlines()[:6]
    with open(src, "w") as f:
        f.write("\n".join(lines) + "\n")
    dic = os.path.join(toy_corpus, "toy_train_input.txt.pkl")
    out1 = str(tmp_path / "out1.txt")
    out2 = str(tmp_path / "out2.txt")
    generate_file(saveto, dic, src, out1, k=3, normalize=True, n_process=1,
                  verbose=False, maxlen=12, devices=["cpu"])
    generate_file(saveto, dic, src, out2, k=3, normalize=True, n_process=2,
                  verbose=False, maxlen=12, devices=["cpu"])
    assert open(out1).read() == open(out2).read()


def test_bench_contract(tmp_path):
    """The driver depends on bench.py's exact CLI + one-JSON-line output
    contract; pin it on the CPU-capable toy config."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--gpus", "1",
         "--steps", "2", "--warmup", "1", "--config", "toy"],
        capture_output=True, text=True, timeout=300, cwd=repo)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(lines) == 1, out.stdout
    r = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"] == "train_tokens_per_sec"
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["data"] == "synthetic" and r["scaling"] == "weak"
    assert r["value"] > 0 and numpy.isfinite(r["config"]["final_cost"])
    assert r["config"]["global_batch"] == r["config"]["batch_per_gpu"]


def test_nan_failure_detection(toy_corpus, tmp_path):
    """NaN/Inf training cost -> hard abort returning the
    (1., 1., 1.) sentinel (nats.py:1415-1417); NaN validation cost ->
    raise (the reference drops into ipdb, nats.py:1096)."""
    import torch

    from nats_amd.engine.validate import pred_probs

    kw = dict(dim_word=10, dim=12, dim_att=6, n_words=64, maxlen=50,
              batch_size=8, valid_batch_size=8,
              datasets=[os.path.join(toy_corpus, "toy_train_input.txt"),
                        os.path.join(toy_corpus, "toy_train_output.txt")],
              valid_datasets=[
                  os.path.join(toy_corpus, "toy_validation_input.txt"),
                  os.path.join(toy_corpus, "toy_validation_output.txt")],
              dictionary=os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
              validFreq=1000, saveFreq=1000, sampleFreq=1000, dispFreq=1000,
              finish_after=2, device="cpu", seed=3,
              saveto=str(tmp_path / "m.npz"))
    # inject the NaN via poisoned initial embeddings (adadelta ignores
    # lrate, so a divergence can't be provoked through the optimizer)
    import nats_amd.models.distraction as D
    orig = D.init_params

    def poisoned(options, seed=None):
        p = orig(options, seed=seed)
        p["Wemb"] = p["Wemb"] * numpy.nan
        return p

    D.init_params = poisoned
    try:
        out = train(**kw)
    finally:
        D.init_params = orig
    assert out == (1.0, 1.0, 1.0)

    # validation NaN raises
    opts = default_options(dim_word=10, dim=12, dim_att=6, n_words=64)
    model = NatsModel(opts, seed=3)
    with torch.no_grad():
        model.P["Wemb"].mul_(float("nan"))
    it = TextIterator(os.path.join(toy_corpus, "toy_validation_input.txt"),
                      os.path.join(toy_corpus, "toy_validation_output.txt"),
                      os.path.join(toy_corpus, "toy_train_input.txt.pkl"),
                      batch_size=4, n_words=64)
    import pytest as _pytest
    with _pytest.raises(FloatingPointError):
        pred_probs(model, it)
