"""Multi-process CPU (gloo) tests of the data-parallel gradient layer."""

import os

import numpy
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from nats_amd.data.prepare import prepare_data


def _make_batch(opts, B, seed):
    rng = numpy.random.RandomState(seed)
    xs = [list(rng.randint(2, opts["n_words"], size=6)) for _ in range(B)]
    ys = [list(rng.randint(2, opts["n_words"], size=4)) for _ in range(B)]
    return [torch.from_numpy(a) for a in prepare_data(xs, ys)]


def _worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nats_amd.models.distraction import NatsModel, default_options
        from nats_amd.parallel.ddp import DataParallelGrads

        opts = default_options(dim_word=8, dim=10, dim_att=6, n_words=32)
        model = NatsModel(opts, seed=3)
        dp = DataParallelGrads(model.parameters(), bucket_cap_mb=1)
        dp.broadcast_params()

        # full batch = shard0 + shard1 (equal sizes)
        full = _make_batch(opts, 4, seed=9)
        shard = [a[:, rank * 2:(rank + 1) * 2].contiguous() for a in full]

        cost = model(*shard).mean()
        cost.backward()
        dp.finish()

        # reference: single-process mean-of-shard-means on the full batch
        model_ref = NatsModel(opts, seed=3)
        model_ref.set_params(model.get_params())  # same (broadcast) weights
        for p in model_ref.parameters():
            p.grad = None
        c0 = model_ref(*[a[:, 0:2].contiguous() for a in full]).mean()
        c1 = model_ref(*[a[:, 2:4].contiguous() for a in full]).mean()
        ((c0 + c1) / 2).backward()

        for (k, p), (k2, p2) in zip(model.P.items(), model_ref.P.items()):
            assert k == k2
            torch.testing.assert_close(p.grad, p2.grad, rtol=1e-4, atol=1e-6)

        # scalar all-reduce
        v = dp.all_reduce_scalar(float(rank + 1), average=True)
        assert abs(v - 1.5) < 1e-9
        if rank == 0:
            open(os.path.join(tmpdir, "ok"), "w").write("ok")
    finally:
        dist.destroy_process_group()


def test_ddp_grad_allreduce_gloo(tmp_path):
    port = 29531
    mp.spawn(_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)
    assert os.path.exists(tmp_path / "ok")


def _ragged_worker(rank, world, port, tmpdir):
    """Full train() with a batch count per epoch NOT divisible by world and
    one maxlen-unsatisfiable batch: ranks must stay collective-aligned
    (ragged tail dropped, empty-batch skip agreed via all_agree)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    try:
        from nats_amd.engine.trainer import train

        src = os.path.join(tmpdir, "src.txt")
        tgt = os.path.join(tmpdir, "tgt.txt")
        words = ["w%d" % i for i in range(2, 28)]
        rng = numpy.random.RandomState(7)
        with open(src, "w") as fs, open(tgt, "w") as ft:
            for i in range(14):  # 7 batches of 2 -> odd, not % 2
                # one over-length source line makes batch 3 empty at maxlen=8
                n = 40 if i in (6, 7) else 5
                fs.write(" ".join(rng.choice(words, size=n)) + "\n")
                ft.write(" ".join(rng.choice(words, size=3)) + "\n")
        from nats_amd.data.dictionary import build_dictionary
        dpath = os.path.join(tmpdir, "d_r%d.pkl" % rank)
        build_dictionary(src, dpath)  # deterministic -> identical per rank
        err = train(dim_word=8, dim=10, dim_att=6, n_words=30, maxlen=8,
                    batch_size=2, valid_batch_size=2, max_epochs=3,
                    dispFreq=100, validFreq=100, saveFreq=100, sampleFreq=100,
                    datasets=[src, tgt], valid_datasets=[src, tgt],
                    dictionary=dpath,
                    saveto=os.path.join(tmpdir, "m_r%d.npz" % rank),
                    device="cpu", seed=1)
        assert numpy.isfinite(err)
        if rank == 0:
            open(os.path.join(tmpdir, "ok_ragged"), "w").write("ok")
    finally:
        for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE"):
            os.environ.pop(k, None)
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_ragged_batches_no_deadlock(tmp_path):
    port = 29541
    mp.spawn(_ragged_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
    assert os.path.exists(tmp_path / "ok_ragged")


@pytest.mark.timeout(600)
def test_bench_multirank_torchrun_contract(tmp_path):
    """The round-end driver launches bench.py via torch.distributed.run
    with N ranks; pin that exact invocation on CPU (gloo, toy config)."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29561", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--config", "toy"],
        capture_output=True, text=True, timeout=540, cwd=repo, env=env)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(lines) == 1, out.stdout  # rank 0 only
    r = json.loads(lines[0])
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["global_batch"] == 2 * r["config"]["batch_per_gpu"]


@pytest.mark.gpu
@pytest.mark.timeout(900)
def test_rccl_dp2_one_gpu():
    """Real-RCCL DP validation on hardware: 2 ranks sharing one MI355X
    (device mapping local_rank % device_count) running the full probe —
    init, broadcast, bucketed all-reduce grad exactness vs a
    single-process oracle, all_agree, scalar reduce. VERDICT r1 #1."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        env.pop(k, None)
    env["NATS_PROBE_FAST"] = "1"
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29573", "scripts/rccl_probe.py"],
        capture_output=True, text=True, timeout=840, cwd=repo, env=env)
    assert out.returncode == 0, (out.stdout[-800:], out.stderr[-1500:])
    import json as _json
    lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert lines, out.stdout
    r = _json.loads(lines[-1])
    assert r["world"] == 2
    # RCCL refuses 2 ranks on one physical GPU (Duplicate GPU detected);
    # the probe then proves the DP layer on device grads over gloo AND
    # real RCCL collectives through a 1-rank nccl communicator.
    if r["backend"] != "nccl":
        assert "refused" in r.get("nccl_shared_device", "")
        assert r.get("rccl_1rank_collectives") == "ok"
    assert r["grad_allreduce_max_rel_err"] < 1e-4
