"""Data-parallel gradient synchronisation over RCCL/xGMI.

The reference has no distributed training at all (train.sh:7 pins
``device=gpu0``); SURVEY.md §2.3 specifies the MI355X-native design: one
process per GPU, ``torch.distributed`` with the nccl backend (= RCCL on
ROCm), bucketed gradient all-reduce overlapped with backward, parameter
broadcast at init, all-reduced validation loss.

Topology note (MI355X, 8 GPUs over xGMI): each GPU has 7 point-to-point
links at ~153 GB/s — ring collectives are per-link bound, so we use
moderately large buckets (default 25 MiB) issued asynchronously as soon as
a bucket's gradients are ready (autograd post-accumulate hooks), letting
RCCL spread channels across links while the tail of backward still runs.
The model here is small (~60-200 M params), so comm is mostly
latency-bound; fewer, larger buckets win.
"""

import os

import torch
import torch.distributed as dist


def distributed_info():
    """(rank, local_rank, world_size) from the torchrun env (or 0,0,1)."""
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    return rank, local_rank, world


def init_distributed(backend=None, device=None):
    """Initialise torch.distributed from torchrun env vars.

    backend: default nccl (=RCCL) on GPU, gloo on CPU.
    Returns (rank, local_rank, world_size); world_size==1 -> no-op.
    """
    rank, local_rank, world = distributed_info()
    if world <= 1:
        return rank, local_rank, world
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    if backend == "nccl":
        if device is None:
            # modulo mapping: identity on a full node (one rank per GPU);
            # lets >1 rank share one device for single-GPU RCCL testing
            device = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(device)
    return rank, local_rank, world


class DataParallelGrads:
    """Bucketed async all-reduce of gradients, overlapped with backward.

    Usage:
        dp = DataParallelGrads(model.parameters())
        ...
        loss.backward()          # hooks fire all_reduce per ready bucket
        dp.finish()              # wait + scale by 1/world
        optimizer.step()

    Buckets are assembled in REVERSE parameter order (gradients become
    ready roughly back-to-front during backward). Gradients are reduced in
    their native dtype (fp32 here — exact DP: sum then divide).
    """

    def __init__(self, params, bucket_cap_mb=25, process_group=None):
        self.params = [p for p in params if p.requires_grad]
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world > 1
        self._works = []
        self._flat = {}  # bucket id -> (flat tensor, [params])
        if not self.enabled:
            return
        # build buckets in reverse order
        cap = bucket_cap_mb * 1024 * 1024
        buckets, cur, cur_bytes = [], [], 0
        for p in reversed(self.params):
            sz = p.numel() * p.element_size()
            if cur and cur_bytes + sz > cap:
                buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            buckets.append(cur)
        self.buckets = buckets
        self._param_bucket = {}
        self._pending = []
        for bi, b in enumerate(buckets):
            for p in b:
                self._param_bucket[id(p)] = bi
        self._ready_count = [0] * len(buckets)
        self._hooks = []
        for p in self.params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready))

    def detach(self):
        """Remove the autograd hooks (e.g. before building a replacement
        DataParallelGrads over the same parameters — two live instances
        would each all-reduce every gradient)."""
        for h in getattr(self, "_hooks", []):
            h.remove()
        self._hooks = []

    def broadcast_params(self):
        """Broadcast initial parameters from rank 0 (SURVEY §2.4 note)."""
        if not self.enabled:
            return
        for p in self.params:
            dist.broadcast(p.data, src=0, group=self.group)

    def _on_grad_ready(self, p):
        bi = self._param_bucket[id(p)]
        self._ready_count[bi] += 1
        if self._ready_count[bi] == len(self.buckets[bi]):
            self._launch(bi)

    def _launch(self, bi):
        bucket = self.buckets[bi]
        grads = [p.grad for p in bucket]
        flat = torch._utils._flatten_dense_tensors(grads)
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group,
                               async_op=True)
        self._pending.append((work, flat, bucket, grads))

    def finish(self):
        """Wait for all reduces, scale by 1/world, scatter back to .grad."""
        if not self.enabled:
            return
        # launch any bucket that never became "fully ready" (params with no
        # grad this step get zeros so DP ranks stay consistent)
        for bi, b in enumerate(self.buckets):
            if self._ready_count[bi] != len(b) and any(
                    p.grad is not None for p in b):
                for p in b:
                    if p.grad is None:
                        p.grad = torch.zeros_like(p)
                self._launch(bi)
        for work, flat, bucket, grads in self._pending:
            work.wait()
            flat.div_(self.world)
            for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(
                    flat, grads)):
                g.copy_(synced)
        self._pending = []
        self._ready_count = [0] * len(self.buckets)

    def all_agree(self, ok):
        """True iff EVERY rank passes ok=True (MIN all-reduce). Used to
        make data-dependent step skips (empty minibatch after maxlen
        truncation) collective so per-rank collective counts stay equal."""
        if not self.enabled:
            return bool(ok)
        t = torch.tensor([1.0 if ok else 0.0], dtype=torch.float32,
                         device="cuda" if dist.get_backend(self.group) == "nccl"
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MIN, group=self.group)
        return bool(t.item() >= 0.5)

    def all_reduce_scalar(self, value, average=True):
        """All-reduce a python float (validation loss, token counts)."""
        if not self.enabled:
            return value
        t = torch.tensor([value], dtype=torch.float64,
                         device="cuda" if dist.get_backend(self.group) == "nccl"
                         else "cpu")
        dist.all_reduce(t, group=self.group)
        v = float(t.item())
        return v / self.world if average else v
