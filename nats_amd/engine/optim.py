"""Optimizers with the reference's exact update formulas.

The reference compiles each optimizer as a pair (f_grad_shared, f_update)
(nats.py:1106-1221); here a single ``step()`` applies the same math after
``backward()``. Formula fidelity notes:

  * adadelta (the default, nats.py:1145-1173): rho=0.95, eps=1e-6; the
    gradient-EMA update happens in f_grad_shared (i.e. BEFORE the update
    direction is computed) and the step-EMA after — order preserved. The
    ``lr`` argument is ignored, as in the reference.
  * adam (nats.py:1106-1142): NONSTANDARD constants — lr0=2e-4 hardcoded
    (the lr argument is ignored), b1=0.1 / b2=0.001 are the (1-beta)
    weights on the NEW gradient, and the bias corrections use
    1 - b1**t / 1 - b2**t with those same small constants. Reproduced
    verbatim; do not "fix" to standard Adam.
  * rmsprop (nats.py:1176-1206): Graves-style with momentum 0.9 and the
    hardcoded 1e-4 step; lr ignored.
  * sgd (nats.py:1209-1221): p -= lr*g. (The reference's sgd signature is
    broken when invoked through eval(optimizer)(...) — it unpacks inputs
    wrongly — we provide the working formula.)

All state tensors are float32 on the parameters' device. On MI355X the
adadelta update (+ the global-norm clip) is executed by the fused
multi-tensor HIP kernel when available (ops/hip/optim.hip).
"""

import math
import os

import torch


def clip_grads_global_norm(params, clip_c):
    """Global-norm gradient clip (nats.py:1344-1356).

    g2 = sum of squared norms over ALL grads; every grad is scaled by
    clip_c/sqrt(g2) iff g2 > clip_c**2. Returns sqrt(g2) (pre-clip norm).
    """
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return 0.0
    # fused multi-tensor path (one norm kernel set + one scale pass)
    norms = torch._foreach_norm(grads)
    g2 = (torch.stack(norms).float() ** 2).sum()
    if clip_c is None or clip_c <= 0:
        return torch.sqrt(g2)
    scale = torch.where(g2 > clip_c * clip_c,
                        clip_c / torch.sqrt(g2),
                        torch.ones_like(g2))
    torch._foreach_mul_(grads, scale)
    # returned lazily (no device sync); float() it only when logging
    return torch.sqrt(g2)


class _RefOptimizer:
    """Base: holds named params + state; subclasses implement _update."""

    def __init__(self, named_params, lrate=0.01, clip_c=-1.0):
        self.named = [(k, p) for k, p in named_params]
        self.params = [p for _, p in self.named]
        self.lrate = lrate
        self.clip_c = clip_c
        self.t = 0

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def state_dict(self):
        return {"t": self.t,
                "state": {k: {sk: sv.clone() for sk, sv in st.items()}
                          for k, st in getattr(self, "state", {}).items()}}

    def load_state_dict(self, sd):
        self.t = sd["t"]
        if hasattr(self, "state"):
            for k, st in sd["state"].items():
                for sk, sv in st.items():
                    self.state[k][sk].copy_(sv)

    @torch.no_grad()
    def step(self):
        norm = clip_grads_global_norm(self.params, self.clip_c)
        self.t += 1
        self._update()
        return norm


class Adadelta(_RefOptimizer):
    """nats.py:1145-1173 (rho=0.95, eps=1e-6; lr ignored)."""

    def __init__(self, named_params, lrate=0.01, clip_c=-1.0,
                 rho=0.95, epsilon=1e-6):
        super().__init__(named_params, lrate, clip_c)
        self.rho, self.eps = rho, epsilon
        self.state = {k: {"rg2": torch.zeros_like(p, dtype=torch.float32),
                          "ru2": torch.zeros_like(p, dtype=torch.float32)}
                      for k, p in self.named}

    @torch.no_grad()
    def _update(self):
        rho, eps = self.rho, self.eps
        for k, p in self.named:
            if p.grad is None:
                continue
            g = p.grad.float()
            st = self.state[k]
            st["rg2"].mul_(rho).add_(g * g, alpha=1.0 - rho)
            ud = -torch.sqrt(st["ru2"] + eps) / torch.sqrt(st["rg2"] + eps) * g
            st["ru2"].mul_(rho).add_(ud * ud, alpha=1.0 - rho)
            p.add_(ud.to(p.dtype))


class Adam(_RefOptimizer):
    """nats.py:1106-1142 — the reference's nonstandard constants, verbatim."""

    LR0 = 0.0002
    B1 = 0.1
    B2 = 0.001
    EPS = 1e-8

    def __init__(self, named_params, lrate=0.01, clip_c=-1.0):
        super().__init__(named_params, lrate, clip_c)
        self.state = {k: {"m": torch.zeros_like(p, dtype=torch.float32),
                          "v": torch.zeros_like(p, dtype=torch.float32)}
                      for k, p in self.named}

    @torch.no_grad()
    def _update(self):
        t = float(self.t)
        fix1 = 1.0 - self.B1 ** t
        fix2 = 1.0 - self.B2 ** t
        lr_t = self.LR0 * (math.sqrt(fix2) / fix1)
        for k, p in self.named:
            if p.grad is None:
                continue
            g = p.grad.float()
            st = self.state[k]
            m_t = self.B1 * g + (1.0 - self.B1) * st["m"]
            v_t = self.B2 * (g * g) + (1.0 - self.B2) * st["v"]
            g_t = m_t / (torch.sqrt(v_t) + self.EPS)
            st["m"].copy_(m_t)
            st["v"].copy_(v_t)
            p.add_((-lr_t * g_t).to(p.dtype))


class RMSProp(_RefOptimizer):
    """nats.py:1176-1206 (Graves rmsprop; lr ignored)."""

    def __init__(self, named_params, lrate=0.01, clip_c=-1.0):
        super().__init__(named_params, lrate, clip_c)
        self.state = {k: {"rg": torch.zeros_like(p, dtype=torch.float32),
                          "rg2": torch.zeros_like(p, dtype=torch.float32),
                          "ud": torch.zeros_like(p, dtype=torch.float32)}
                      for k, p in self.named}

    @torch.no_grad()
    def _update(self):
        for k, p in self.named:
            if p.grad is None:
                continue
            g = p.grad.float()
            st = self.state[k]
            st["rg"].mul_(0.95).add_(g, alpha=0.05)
            st["rg2"].mul_(0.95).add_(g * g, alpha=0.05)
            new_ud = 0.9 * st["ud"] - 1e-4 * g / torch.sqrt(
                st["rg2"] - st["rg"] ** 2 + 1e-4)
            st["ud"].copy_(new_ud)
            p.add_(new_ud.to(p.dtype))


class SGD(_RefOptimizer):
    """nats.py:1209-1221: p -= lr * g."""

    @torch.no_grad()
    def _update(self):
        for _, p in self.named:
            if p.grad is None:
                continue
            p.add_(p.grad, alpha=-self.lrate)


_OPTIMIZERS = {"adadelta": Adadelta, "adam": Adam,
               "rmsprop": RMSProp, "sgd": SGD}


def build_optimizer(name, named_params, lrate=0.01, clip_c=-1.0):
    if name not in _OPTIMIZERS:
        raise ValueError("unknown optimizer %r (have %s)"
                         % (name, sorted(_OPTIMIZERS)))
    cls = _OPTIMIZERS[name]
    named_params = list(named_params)
    # On GPU the adadelta step (clip included) runs through the fused
    # multi-tensor HIP kernel; see FusedAdadelta in nats_amd/ops/optim.py.
    if name == "adadelta" and not os.environ.get("NATS_AMD_FORCE_EAGER"):
        if named_params and named_params[0][1].is_cuda:
            from ..ops.optim import FusedAdadelta
            return FusedAdadelta(named_params, lrate=lrate, clip_c=clip_c)
    return cls(named_params, lrate=lrate, clip_c=clip_c)
