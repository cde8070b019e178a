"""Fused GPU optimizer step (global-norm clip + adadelta) — wrapper.

Until the HIP kernel lands, this subclass runs the same math through
torch._foreach ops on GPU (one fused multi-tensor pass per state update);
the HIP path replaces it transparently when the extension is present.
Formula identical to engine.optim.Adadelta (nats.py:1145-1173).
"""

import torch

from ..engine.optim import Adadelta


class FusedAdadelta(Adadelta):
    @torch.no_grad()
    def _update(self):
        from . import _hip_ext
        ext = _hip_ext()
        ks = [k for k, p in self.named if p.grad is not None]
        ps = [p for _, p in self.named if p.grad is not None]
        gs = [p.grad for p in ps]
        rg2 = [self.state[k]["rg2"] for k in ks]
        ru2 = [self.state[k]["ru2"] for k in ks]
        if ext is not None and hasattr(ext, "adadelta_step"):
            ext.adadelta_step(ps, gs, rg2, ru2, self.rho, self.eps)
            return
        rho, eps = self.rho, self.eps
        torch._foreach_mul_(rg2, rho)
        g2 = torch._foreach_mul(gs, gs)
        torch._foreach_add_(rg2, g2, alpha=1.0 - rho)
        num = torch._foreach_sqrt(torch._foreach_add(ru2, eps))
        den = torch._foreach_sqrt(torch._foreach_add(rg2, eps))
        ud = torch._foreach_div(num, den)
        ud = torch._foreach_mul(ud, gs)
        torch._foreach_neg_(ud)
        torch._foreach_mul_(ru2, rho)
        ud2 = torch._foreach_mul(ud, ud)
        torch._foreach_add_(ru2, ud2, alpha=1.0 - rho)
        torch._foreach_add_(ps, ud)
