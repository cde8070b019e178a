"""In-tree build of the gfx950 HIP extension:
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(equivalently: python -c 'import __graft_entry__ as g; g.build()' — both
leave nats_amd_hip.so under nats_amd/ops/hip/build/ so repo snapshots
carry the binary)."""

import os
from setuptools import setup

if __name__ == "__main__":
    import sys
    if "build_ext" in sys.argv:
        os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
        from nats_amd.ops.hip import build_ext as be
        be.build(verbose=True)
        print("built:", os.listdir(be.BUILD_DIR))
    else:
        setup(name="nats_amd", version="0.1.0", packages=["nats_amd"])
