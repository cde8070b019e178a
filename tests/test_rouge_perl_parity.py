"""Live parity: our ROUGE port vs the reference's ROUGE.pl run by perl
(skipped when perl or the reference checkout is unavailable)."""

import os
import shutil
import subprocess

import numpy
import pytest

from nats_amd.decode.rouge import format_report

ROUGE_PL = "/root/reference/scripts/ROUGE.pl"

pytestmark = pytest.mark.skipif(
    shutil.which("perl") is None or not os.path.exists(ROUGE_PL),
    reason="perl or reference ROUGE.pl unavailable")


def _mk_files(tmp_path, seed=0, n=25):
    rng = numpy.random.RandomState(seed)
    vocab = ["w%d" % i for i in range(30)]
    ref = tmp_path / "ref.txt"
    sys_ = tmp_path / "sys.txt"
    with open(ref, "w") as fr, open(sys_, "w") as fs:
        for _ in range(n):
            fr.write(" ".join(rng.choice(vocab, size=rng.randint(1, 15))) +
                     "\n")
            fs.write(" ".join(rng.choice(vocab, size=rng.randint(0, 15))) +
                     "\n")
    return str(ref), str(sys_)


@pytest.mark.parametrize("seed", [0, 1, 2])
@pytest.mark.parametrize("nsize,metric", [(1, "N"), (2, "N"), (3, "N"),
                                          (1, "L")])
def test_matches_perl(tmp_path, seed, nsize, metric):
    ref, sys_ = _mk_files(tmp_path, seed)
    perl_out = subprocess.run(
        ["perl", ROUGE_PL, str(nsize), metric, ref, sys_],
        capture_output=True, text=True, check=True).stdout
    ours = format_report(ref, sys_, nsize, metric)
    # compare the three printed averages
    perl_nums = perl_out.strip().splitlines()[-1].split()
    our_nums = ours.strip().splitlines()[-1].split()
    assert perl_nums == our_nums, (perl_out, ours)
