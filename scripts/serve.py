#!/usr/bin/env python3
"""Serve a trained summarizer over HTTP (one server per GPU).

Usage:
  python scripts/serve.py model.npz dict.pkl [--port 8000] [--k 10] ...

Endpoints: POST /summarize {"text": ...} | {"texts": [...]},
GET /healthz, GET /stats. See nats_amd/serve/app.py.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("model", help="checkpoint .npz (train_nats.py saveto)")
    ap.add_argument("dictionary", help="dictionary pickle")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--device", default=None, help="e.g. cuda:0 / cpu")
    ap.add_argument("-k", "--k", type=int, default=10, help="beam width")
    ap.add_argument("--maxlen", type=int, default=100)
    ap.add_argument("-n", "--normalize", action="store_true", default=True)
    ap.add_argument("--kl-factor", type=float, default=0.0)
    ap.add_argument("--ctx-factor", type=float, default=0.0)
    ap.add_argument("--state-factor", type=float, default=0.0)
    ap.add_argument("-c", "--chr-level", action="store_true")
    ap.add_argument("--max-batch", type=int, default=None)
    ap.add_argument("--max-wait-ms", type=float, default=5.0)
    args = ap.parse_args()

    import uvicorn

    from nats_amd.serve import SummarizerService, create_app

    service = SummarizerService(
        args.model, args.dictionary, device=args.device, k=args.k,
        maxlen=args.maxlen, normalize=args.normalize,
        kl_factor=args.kl_factor, ctx_factor=args.ctx_factor,
        state_factor=args.state_factor, chr_level=args.chr_level,
        max_batch=args.max_batch, max_wait_ms=args.max_wait_ms)
    app = create_app(service)
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
