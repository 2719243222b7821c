"""Flat parameter arena + fused optimizers.

All trainable parameters of a model live in ONE flat allocation per role:

    master  fp32   (the authoritative weights)
    mirror  bf16   (the compute copy the GEMM kernels read)
    grad    fp32   (what backward accumulates / RCCL all-reduces)
    + optimizer state (momentum / adam m1,m2), same flat shape

so an optimizer step is ONE kernel over the whole model (elementwise.hip
sgd/adam), a DDP step is ONE bucketed all-reduce over ``grad`` (SURVEY §2.5:
"DP via bucketed RCCL all-reduce over xGMI — first-class"), and everything is
hipGraph-capture friendly (no allocations after ``finalize``).

Each parameter is 8-element padded so every view is 16-byte aligned for the
vectorized kernels.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

import torch

from ..ops import functional as F


def _pad8(n: int) -> int:
    return (n + 7) // 8 * 8


class ParamArena:
    def __init__(self, device="cpu"):
        self.device = torch.device(device)
        self._specs: List[Tuple[str, Tuple[int, ...], object]] = []
        self._offsets: Dict[str, Tuple[int, int]] = {}
        self.master: Optional[torch.Tensor] = None
        self.mirror: Optional[torch.Tensor] = None
        self.grad: Optional[torch.Tensor] = None
        self.numel = 0

    def add(self, name: str, shape: Tuple[int, ...], init) -> str:
        """Register a parameter before finalize(). ``init``: tensor, callable
        (shape)->tensor, or float std for randn*std."""
        assert self.master is None, "arena already finalized"
        assert name not in self._offsets
        n = int(math.prod(shape))
        self._offsets[name] = (self.numel, n)
        self._specs.append((name, tuple(shape), init))
        self.numel += _pad8(n)
        return name

    def finalize(self, seed: int = 0) -> None:
        dev = self.device
        self.master = torch.zeros(self.numel, dtype=torch.float32, device=dev)
        self.grad = torch.zeros(self.numel, dtype=torch.float32, device=dev)
        g = torch.Generator(device="cpu").manual_seed(seed)
        for name, shape, init in self._specs:
            off, n = self._offsets[name]
            if callable(init):
                t = init(shape)
            elif isinstance(init, torch.Tensor):
                t = init
            else:
                t = torch.randn(shape, generator=g, dtype=torch.float32) * float(init)
            self.master[off:off + n] = t.reshape(-1).to(dev, torch.float32)
        self.mirror = self.master.to(torch.bfloat16)

    # -- views --------------------------------------------------------------
    def _view(self, flat: torch.Tensor, name: str) -> torch.Tensor:
        off, n = self._offsets[name]
        shape = next(s for nm, s, _ in self._specs if nm == name)
        return flat[off:off + n].view(shape)

    def p(self, name: str) -> torch.Tensor:
        """bf16 compute view (what kernels read)."""
        return self._view(self.mirror, name)

    def pf(self, name: str) -> torch.Tensor:
        """fp32 master view."""
        return self._view(self.master, name)

    def g(self, name: str) -> torch.Tensor:
        """fp32 grad view (backward accumulates here)."""
        return self._view(self.grad, name)

    def grad_slice(self, names) -> Optional[torch.Tensor]:
        """Contiguous flat-grad slice covering ``names`` (params are laid
        out in registration order, so a layer's params are contiguous) —
        the unit of the DDP overlap all-reduce."""
        offs = [self._offsets[n] for n in names if n in self._offsets]
        if not offs:
            return None
        start = min(o for o, _ in offs)
        end = max(o + _pad8(n) for o, n in offs)
        return self.grad[start:end]

    def zero_grad(self) -> None:
        self.grad.zero_()

    # -- checkpoint (real in-training checkpoints) --------------
    def state_dict(self) -> Dict[str, torch.Tensor]:
        return {name: self.pf(name).detach().cpu().clone()
                for name, _, _ in self._specs}

    def load_state_dict(self, sd: Dict[str, torch.Tensor]) -> None:
        for name, _, _ in self._specs:
            self.pf(name).copy_(sd[name].to(self.device))
        self.mirror.copy_(self.master.to(torch.bfloat16))


class SGD:
    """Fused momentum SGD over the whole arena — one kernel per step."""

    def __init__(self, arena: ParamArena, lr: float = 0.01, momentum: float = 0.9,
                 weight_decay: float = 0.0):
        self.arena = arena
        self.lr, self.momentum, self.weight_decay = lr, momentum, weight_decay
        self.mom = torch.zeros_like(arena.master)

    def step(self, gscale: float = 1.0) -> None:
        a = self.arena
        F.sgd_step(a.master, a.grad, self.mom, a.mirror, self.lr, self.momentum,
                   self.weight_decay, gscale)

    def state_dict(self):
        return {"mom": self.mom.cpu().clone(), "lr": self.lr,
                "momentum": self.momentum, "weight_decay": self.weight_decay}

    def load_state_dict(self, sd):
        self.mom.copy_(sd["mom"].to(self.mom.device))
        self.lr, self.momentum = sd["lr"], sd["momentum"]
        self.weight_decay = sd["weight_decay"]


class Adam:
    def __init__(self, arena: ParamArena, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        self.arena = arena
        self.lr, self.betas, self.eps, self.weight_decay = lr, betas, eps, weight_decay
        self.m1 = torch.zeros_like(arena.master)
        self.m2 = torch.zeros_like(arena.master)
        if arena.master.is_cuda:
            # device step counter: the increment and the bias correction both
            # happen on-device, so a hipGraph-captured step stays correct
            self.t = torch.zeros(1, dtype=torch.int32, device=arena.master.device)
        else:
            self.t = 0

    def step(self, gscale: float = 1.0) -> None:
        if torch.is_tensor(self.t):
            self.t += 1
        else:
            self.t += 1
        a = self.arena
        F.adam_step(a.master, a.grad, self.m1, self.m2, a.mirror, self.lr,
                    self.betas[0], self.betas[1], self.eps, self.weight_decay,
                    self.t, gscale)
