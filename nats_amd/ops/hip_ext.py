"""Loader for the in-tree compiled HIP extension.

The extension is built IN-TREE (so the .so travels with the repo snapshot
to GPU machines) by ``__graft_entry__.build()`` or
``python setup.py build_ext --inplace``. This module only locates and
imports it; it never builds.
"""

import os


def load():
    """Import nats_amd_hip (the compiled extension) or raise ImportError."""
    # the .so lives next to the package root after an in-tree build
    try:
        import nats_amd_hip  # noqa: F401
        return nats_amd_hip
    except ImportError:
        pass
    # also look inside nats_amd/ops/hip/build
    here = os.path.dirname(os.path.abspath(__file__))
    import sys
    cand = os.path.join(here, "hip", "build")
    if os.path.isdir(cand) and cand not in sys.path:
        sys.path.insert(0, cand)
    import nats_amd_hip  # raises ImportError if truly absent
    return nats_amd_hip
