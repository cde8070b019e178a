"""In-tree build of the nats_amd_hip extension for gfx950.

Uses torch.utils.cpp_extension (drives hipcc for the .hip sources under
PYTORCH_ROCM_ARCH=gfx950) with build_directory inside the repo so the
built .so travels with repo snapshots to GPU machines. hipcc
cross-compiles fine on GPU-less boxes.
"""

import os

HERE = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(HERE, "build")

SOURCES = [
    os.path.join(HERE, "bind.cpp"),
    os.path.join(HERE, "gru_scan.hip"),
    os.path.join(HERE, "cond_gru.hip"),
    os.path.join(HERE, "softmax_ce.hip"),
    os.path.join(HERE, "mfma_test.hip"),
    os.path.join(HERE, "optim.hip"),
    os.path.join(HERE, "rerank.hip"),
    os.path.join(HERE, "embed.hip"),
    os.path.join(HERE, "pack.hip"),
]


def build(verbose=False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load
    mod = load(
        name="nats_amd_hip",
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        with_cuda=True,
    )
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built:", os.listdir(BUILD_DIR))
