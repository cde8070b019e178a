#!/usr/bin/env python3
"""Serving throughput benchmark: concurrent summarize requests through the
micro-batching service (nats_amd/serve) on one GPU.

Synthetic corpus + random-init weights (no network). Prints one JSON line
per concurrency level."""

import argparse
import json
import os
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=64, help="requests per level")
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--src-words", type=int, default=60)
    ap.add_argument("--maxlen", type=int, default=30)
    ap.add_argument("--dim", type=int, default=500)
    ap.add_argument("--vocab", type=int, default=4000)
    ap.add_argument("--concurrency", type=int, nargs="+", default=[1, 4, 16])
    args = ap.parse_args()

    from nats_amd.engine.checkpoint import save_checkpoint
    from nats_amd.models.distraction import NatsModel, default_options
    from nats_amd.serve import SummarizerService

    d = tempfile.mkdtemp()
    rng = numpy.random.RandomState(0)
    vocab_words = ["w%d" % i for i in range(args.vocab - 2)]
    corpus = os.path.join(d, "corpus.txt")
    with open(corpus, "w") as f:
        # every vocab word appears (freq-desc ids then cover the model's
        # full n_words range), plus random lines for realistic freq skew
        for i in range(0, len(vocab_words), 20):
            f.write(" ".join(vocab_words[i:i + 20]) + "\n")
        for _ in range(200):
            f.write(" ".join(rng.choice(vocab_words, size=10)) + "\n")
    from nats_amd.data.dictionary import build_dictionary
    build_dictionary(corpus, os.path.join(d, "dict.pkl"))

    opts = default_options(dim=args.dim, dim_word=100, dim_att=100,
                           n_words=args.vocab)
    model = NatsModel(opts, seed=0)
    saveto = os.path.join(d, "model.npz")
    save_checkpoint(saveto, model.get_params(), [], options=opts)
    del model

    device = "cuda" if torch.cuda.is_available() else "cpu"
    svc = SummarizerService(saveto, os.path.join(d, "dict.pkl"),
                            device=device, k=args.k, maxlen=args.maxlen)
    texts = [" ".join(rng.choice(vocab_words, size=args.src_words))
             for _ in range(args.n)]
    svc.summarize_many(texts[:4])  # warm

    for conc in args.concurrency:
        s_before = svc.stats()
        done = []
        lock = threading.Lock()
        it = iter(range(args.n))

        def client():
            while True:
                with lock:
                    i = next(it, None)
                if i is None:
                    return
                t0 = time.perf_counter()
                svc.summarize(texts[i])
                with lock:
                    done.append(time.perf_counter() - t0)

        t0 = time.perf_counter()
        threads = [threading.Thread(target=client) for _ in range(conc)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        dt = time.perf_counter() - t0
        lat = sorted(done)
        print(json.dumps({
            "metric": "serve_requests_per_sec", "value": args.n / dt,
            "concurrency": conc, "n": args.n, "beam": args.k,
            "p50_ms": 1000 * lat[len(lat) // 2],
            "p95_ms": 1000 * lat[int(len(lat) * 0.95)],
            "avg_batch": ((svc.stats()["batched_requests"]
                           - s_before["batched_requests"])
                          / max(1, svc.stats()["batches"]
                                - s_before["batches"])), "device": device,
            "dim": args.dim, "vocab": args.vocab, "data": "synthetic"}),
            flush=True)
    svc.close()


if __name__ == "__main__":
    main()
