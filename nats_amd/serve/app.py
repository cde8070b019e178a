"""HTTP summarization service.

A thin production-serving layer over the batch decode engine
(nats_amd.decode): requests are coalesced by a background worker into
micro-batches of up to ``64 // beam`` sentences (the fused decoder
kernels take 32 beam rows per launch, ops/hip/cond_gru.hip) and decoded
jointly with :func:`nats_amd.decode.batched.gen_sample_batched`, so
concurrent requests share kernel launches instead of queueing whole
beam searches behind each other.

The reference framework has no serving path (decode is offline only,
gen.py / test.sh); endpoints follow its decode semantics: beam search
with optional distraction rerank (nats.py:966-1011), UNK replacement by
attention argmax (replace_unk.py:14-37) done inline.

Run: ``python scripts/serve.py model.npz dict.pkl --port 8000``.
"""

import threading
import time
from collections import deque

import numpy
import torch

from ..data.dictionary import invert_dictionary, load_dictionary
from ..decode.batched import gen_sample_batched
from ..decode.driver import map_line
from ..engine.checkpoint import load_checkpoint, load_options
from ..models.distraction import NatsModel


class SummarizerService:
    """Loads one model replica and serves micro-batched beam decode.

    Thread-safe: ``summarize`` may be called from many request threads;
    a single worker thread owns the model/GPU and drains the queue in
    batches of up to ``max_batch`` (default: 64 // k beam rows fit the
    kernel launch), waiting at most ``max_wait_ms`` for co-batchable
    requests once one is pending.
    """

    def __init__(self, model_path, dictionary, device=None, k=10, maxlen=100,
                 normalize=True, kl_factor=0.0, ctx_factor=0.0,
                 state_factor=0.0, chr_level=False, max_batch=None,
                 max_wait_ms=5.0):
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.k = int(k)
        self.maxlen = int(maxlen)
        self.normalize = bool(normalize)
        self.kl_factor = float(kl_factor)
        self.ctx_factor = float(ctx_factor)
        self.state_factor = float(state_factor)
        self.chr_level = bool(chr_level)
        self.max_batch = int(max_batch or max(1, 64 // self.k))
        self.max_wait_s = float(max_wait_ms) / 1000.0
        self.model_path = str(model_path)

        self.options = load_options(model_path)
        self.word_dict = load_dictionary(dictionary)
        self.word_idict = invert_dictionary(self.word_dict, with_specials=True)
        params, _ = load_checkpoint(model_path)
        self.model = NatsModel(self.options, params=dict(params))
        self.model.eval()
        self.model.to(device)

        self._queue = deque()
        self._cv = threading.Condition()
        self._closed = False
        self._stats = {"requests": 0, "batches": 0, "batched_requests": 0,
                       "decode_s": 0.0}
        self._worker = threading.Thread(target=self._run, daemon=True,
                                        name="nats-serve-decode")
        self._worker.start()

    # ---- request side ----

    def summarize(self, text, timeout=120.0, maxlen=None):
        """Decode one source text; blocks until its micro-batch is done.

        maxlen overrides the service default for this request (requests
        with different caps still share a batch — the batched decoder
        takes per-sentence maxlens). Returns dict(summary, tokens,
        score, alignment).
        """
        item = {"text": text, "event": threading.Event(), "result": None,
                "error": None, "maxlen": int(maxlen or self.maxlen)}
        with self._cv:
            if self._closed:
                raise RuntimeError("service is shut down")
            self._queue.append(item)
            self._cv.notify()
        if not item["event"].wait(timeout):
            raise TimeoutError("decode timed out")
        if item["error"] is not None:
            raise item["error"]
        return item["result"]

    def summarize_many(self, texts, timeout=300.0, maxlen=None):
        """Enqueue several texts at once (they co-batch immediately)."""
        ml = int(maxlen or self.maxlen)
        items = [{"text": t, "event": threading.Event(), "result": None,
                  "error": None, "maxlen": ml} for t in texts]
        with self._cv:
            if self._closed:
                raise RuntimeError("service is shut down")
            self._queue.extend(items)
            self._cv.notify()
        deadline = time.monotonic() + timeout
        for it in items:
            if not it["event"].wait(max(0.0, deadline - time.monotonic())):
                raise TimeoutError("decode timed out")
            if it["error"] is not None:
                raise it["error"]
        return [it["result"] for it in items]

    def stats(self):
        with self._cv:
            s = dict(self._stats)
        s["avg_batch"] = (s["batched_requests"] / s["batches"]
                          if s["batches"] else 0.0)
        return s

    def close(self):
        with self._cv:
            self._closed = True
            self._cv.notify_all()
        self._worker.join(timeout=10)

    # ---- worker side ----

    def _collect(self):
        """Wait for >=1 request, then linger max_wait_s for co-batchables."""
        with self._cv:
            while not self._queue and not self._closed:
                self._cv.wait()
            if self._closed and not self._queue:
                return None
            deadline = time.monotonic() + self.max_wait_s
            while (len(self._queue) < self.max_batch and not self._closed):
                left = deadline - time.monotonic()
                if left <= 0:
                    break
                self._cv.wait(timeout=left)
            batch = [self._queue.popleft()
                     for _ in range(min(len(self._queue), self.max_batch))]
            return batch

    def _run(self):
        while True:
            batch = self._collect()
            if batch is None:
                return
            try:
                t0 = time.perf_counter()
                results = self._decode([it["text"] for it in batch],
                                       [it["maxlen"] for it in batch])
                dt = time.perf_counter() - t0
                with self._cv:
                    self._stats["requests"] += len(batch)
                    self._stats["batches"] += 1
                    self._stats["batched_requests"] += len(batch)
                    self._stats["decode_s"] += dt
                for it, res in zip(batch, results):
                    it["result"] = res
                    it["event"].set()
            except Exception as e:  # deliver the failure to every waiter
                for it in batch:
                    it["error"] = e
                    it["event"].set()

    def _decode(self, texts, maxlens=None):
        srcs = [(t.strip().split() if not self.chr_level
                 else list(t.strip())) for t in texts]
        seqs = [map_line(t, self.word_dict, self.options["n_words"],
                         self.chr_level) for t in texts]
        xs = [torch.tensor(s, dtype=torch.int64,
                           device=self.device).reshape(-1, 1) for s in seqs]
        with torch.no_grad():
            outs = gen_sample_batched(
                self.model, xs, k=self.k,
                maxlen=maxlens if maxlens else self.maxlen, use_unk=True,
                kl_factor=self.kl_factor, ctx_factor=self.ctx_factor,
                state_factor=self.state_factor)
        results = []
        for src_words, (sample, score, alphas) in zip(srcs, outs):
            score = numpy.array(score, dtype=numpy.float64)
            if self.normalize:
                lengths = numpy.array([max(len(s), 1) for s in sample])
                score = score / lengths
            sidx = int(numpy.argmin(score))
            ids = sample[sidx]
            align = [int(numpy.argmax(a)) for a in alphas[sidx]]
            words = []
            for w, p in zip(ids, align):
                if w == 0:  # eos
                    break
                # UNK (and any id outside the dictionary, e.g. a checkpoint
                # trained with n_words > dict size) -> aligned source word
                if (w == 1 or w not in self.word_idict) and p < len(src_words):
                    words.append(src_words[p])
                else:
                    words.append(self.word_idict.get(w, "UNK"))
            results.append({
                "summary": " ".join(words),
                "tokens": [int(w) for w in ids if w != 0],
                "score": float(score[sidx]),
                "alignment": align[:len(words)],
            })
        return results


def create_app(service):
    """Build the FastAPI app around a SummarizerService."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class SummarizeRequest(BaseModel):
        text: str = None
        texts: list = None
        maxlen: int = None

    app = FastAPI(title="nats_amd summarizer",
                  description="Distraction-based neural summarization "
                              "(MI355X-native)")
    app.state.service = service

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "device": service.device,
                "model": service.model_path, "beam": service.k,
                "max_batch": service.max_batch}

    @app.get("/stats")
    def stats():
        return service.stats()

    @app.post("/summarize")
    def summarize(req: SummarizeRequest):
        if req.text is None and not req.texts:
            raise HTTPException(status_code=422,
                                detail="provide 'text' or 'texts'")
        try:
            if req.text is not None:
                return service.summarize(req.text, maxlen=req.maxlen)
            return {"results": service.summarize_many(req.texts,
                                                      maxlen=req.maxlen)}
        except TimeoutError as e:
            raise HTTPException(status_code=504, detail=str(e))

    return app
