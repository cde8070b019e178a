"""Fused GPU optimizer step: global-norm clip + adadelta in two
multi-tensor HIP kernels (ops/hip/optim.hip). Falls back to torch._foreach
when the extension is unavailable. Formulas identical to
engine.optim.Adadelta + clip_grads_global_norm (nats.py:1145-1173,
1344-1356)."""

import torch

from ..engine.optim import Adadelta
from . import _hip_ext

CHUNK = 1 << 16


class FusedAdadelta(Adadelta):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._tables = None

    def _build_tables(self, device):
        sizes = [p.numel() for _, p in self.named]
        chunk_tensor, chunk_off = [], []
        for ti, n in enumerate(sizes):
            for off in range(0, n, CHUNK):
                chunk_tensor.append(ti)
                chunk_off.append(off)
        self._tables = (
            torch.tensor(sizes, dtype=torch.int64, device=device),
            torch.tensor(chunk_tensor, dtype=torch.int32, device=device),
            torch.tensor(chunk_off, dtype=torch.int64, device=device),
            torch.zeros(1, dtype=torch.float32, device=device),
        )

    @torch.no_grad()
    def step(self):
        ext = _hip_ext()
        ps = [p for _, p in self.named]
        if ext is None or not ps[0].is_cuda or any(
                p.grad is None for p in ps):
            return super().step()
        self.t += 1
        device = ps[0].device
        if self._tables is None:
            self._build_tables(device)
        sizes_t, chunk_tensor, chunk_off, g2 = self._tables
        g2.zero_()
        rg2 = [self.state[k]["rg2"] for k, _ in self.named]
        ru2 = [self.state[k]["ru2"] for k, _ in self.named]
        # pointer table: rebuilt only when a storage moved (grads keep
        # their buffers across steps under a steady allocator, and MUST
        # for hipGraph capture — an H2D upload of a fresh pageable tensor
        # is not capturable, and replays need stable addresses anyway)
        host_ptrs = ([p.data_ptr() for p in ps] +
                     [p.grad.data_ptr() for p in ps] +
                     [t.data_ptr() for t in rg2] +
                     [t.data_ptr() for t in ru2])
        cached = getattr(self, "_ptr_cache", None)
        if cached is not None and cached[0] == host_ptrs:
            ptrs = cached[1]
        else:
            ptrs = torch.tensor(host_ptrs, dtype=torch.int64, device=device)
            self._ptr_cache = (host_ptrs, ptrs)
        ext.fused_adadelta_step(ptrs, sizes_t, chunk_tensor, chunk_off, g2,
                                float(self.clip_c), self.rho, self.eps)
        # The fused kernel writes parameters through raw pointers, which
        # never bumps tensor._version — the invalidation key used by the
        # packed-weight cache (ops/cond_gru.py _step_packed) and the
        # hipGraph stepper cache (decode/graph.py). Bump an explicit
        # update epoch on each parameter so those caches see the change.
        for p in ps:
            p._nats_update_epoch = getattr(p, "_nats_update_epoch", 0) + 1
        return torch.sqrt(g2[0])  # pre-clip norm, no host sync

    @torch.no_grad()
    def _update(self):  # pragma: no cover - only used via super().step()
        Adadelta._update(self)
