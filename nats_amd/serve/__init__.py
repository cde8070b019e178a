"""Online serving for the distraction summarizer (FastAPI).

The reference ships only offline batch decode (gen.py + test.sh); this
subsystem adds a production serving path on top of the same engine:
micro-batched beam search over the fused HIP decode kernels, one server
per GPU. See :mod:`nats_amd.serve.app`.
"""

from .app import SummarizerService, create_app

__all__ = ["SummarizerService", "create_app"]
