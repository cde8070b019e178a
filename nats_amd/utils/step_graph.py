"""Whole-training-step hipGraph capture.

The reference compiles its step into one Theano function
(f_grad_shared/f_update, nats.py:1112-1171) — one host call per
minibatch. Our step is ~2000 HIP kernel launches (two sequential scans x
T timesteps plus readout/softmax/optimizer); capturing forward +
backward + fused-optimizer into ONE torch.cuda.CUDAGraph (hipGraph on
ROCm) collapses every inter-kernel launch gap and all host launch
overhead into a single graph replay.

Requirements (all hold for fixed-shape steps):
  * static input shapes — one graph per (T_s, T_t, B) shape key,
  * gradients keep their storage across steps (grads are zeroed in-graph,
    never set to None),
  * the fused adadelta's device pointer table is cached (ops/optim.py).

Trainer integration caches a small number of shape keys and falls back
to eager for the long tail (real corpora have ragged batches; bench.py
has one shape). Capture failures (e.g. a non-capturable collective
backend) raise at capture time and disable graphing for the session.
"""

import torch


class GraphedTrainStep:
    """Replayable fwd+bwd+optimizer step for one input shape."""

    def __init__(self, model, opt, dp, inputs, use_amp=True, warmup=3):
        self.model = model
        self.opt = opt
        self.dp = dp
        self.use_amp = use_amp
        self.static = [t.clone() for t in inputs]
        self.graph = None
        self.cost = None
        self._capture(warmup)

    def _full_step(self):
        params = [p for p in self.model.parameters()]
        grads = [p.grad for p in params if p.grad is not None]
        if grads:
            torch._foreach_zero_(grads)
        if self.use_amp:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                cost = self.model(*self.static).mean()
        else:
            cost = self.model(*self.static).mean()
        cost.backward()
        if self.dp is not None:
            self.dp.finish()
        self.opt.step()
        return cost

    def _capture(self, warmup):
        # the warmup steps REALLY update parameters/optimizer state (they
        # must run the true kernels to steady-state the allocator), so
        # snapshot and restore both — training parity: capturing a graph
        # must not inject extra updates (nats.py trains one update per
        # minibatch, nothing else)
        with torch.no_grad():
            param_snap = [p.detach().clone() for p in
                          self.model.parameters()]
        opt_snap = self.opt.state_dict()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                self._full_step()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.cost = self._full_step()
        # self-check: one replay must MOVE the parameters (a capture that
        # silently drops the backward/optimizer dependency would
        # otherwise train nothing while reporting healthy throughput)
        with torch.no_grad():
            csum0 = float(torch.stack(
                [p.float().abs().sum() for p in
                 self.model.parameters()]).sum())
        self.graph.replay()
        torch.cuda.synchronize()
        with torch.no_grad():
            csum1 = float(torch.stack(
                [p.float().abs().sum() for p in
                 self.model.parameters()]).sum())
        with torch.no_grad():
            for p, snap in zip(self.model.parameters(), param_snap):
                p.copy_(snap)
        self.opt.load_state_dict(opt_snap)
        if csum1 == csum0:
            raise RuntimeError(
                "graph replay left parameters unchanged — capture did not "
                "wire the backward/optimizer update")

    def step(self, x, x_mask, y, y_mask):
        """Copy inputs into the static buffers and replay. Returns the
        (device) cost tensor — read it AFTER this call returns."""
        for dst, src in zip(self.static, (x, x_mask, y, y_mask)):
            dst.copy_(src, non_blocking=True)
        self.graph.replay()
        # replays run the fused optimizer kernel without re-entering
        # python, so bump the update epoch the decode/pack caches key on
        # (ops/optim.py does this on the eager path)
        for p in self.model.parameters():
            p._nats_update_epoch = getattr(p, "_nats_update_epoch", 0) + 1
        return self.cost


class GraphedStepCache:
    """Per-shape cache of GraphedTrainStep with eager fallback.

    max_shapes bounds capture memory; once a capture attempt fails the
    cache disables itself (returns None forever) so a non-capturable
    environment costs one attempt, not one per shape.
    """

    def __init__(self, model, opt, dp=None, use_amp=True, max_shapes=16):
        self.model = model
        self.opt = opt
        self.dp = dp
        self.use_amp = use_amp
        self.max_shapes = max_shapes
        self.graphs = {}
        self.disabled = False

    def get(self, x, x_mask, y, y_mask):
        if self.disabled or not x.is_cuda:
            return None
        key = (tuple(x.shape), tuple(y.shape))
        st = self.graphs.get(key)
        if st is not None:
            return st
        if len(self.graphs) >= self.max_shapes:
            return None
        try:
            st = GraphedTrainStep(self.model, self.opt, self.dp,
                                  (x, x_mask, y, y_mask),
                                  use_amp=self.use_amp)
        except Exception as e:  # non-capturable backend/op: go eager
            import sys
            print("step-graph capture failed (%s); falling back to eager"
                  % str(e)[:200], file=sys.stderr)
            self.disabled = True
            torch.cuda.synchronize()
            return None
        self.graphs[key] = st
        return st
