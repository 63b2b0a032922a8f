"""Updaters: fused Adam (north-star) and RMSProp (reference semantics).

Reference updater stack per layer (Java:123-127):
  gradient -> elementwise |g|<=clip (ClipElementWiseAbsoluteValue 1.0)
           -> + l2 * w   (L2 0.0001, coupled)
           -> RMSProp(lr, rmsDecay, eps)   [north-star swaps in fused Adam]
Frozen layers use lr=0.0 (gradients still flow; Java:187-216).

On GPU the whole chain (clip + L2 + moment update + master-weight fp32 +
bf16 write-back) is ONE fused HIP kernel per parameter; bf16 params keep
fp32 masters. On CPU it is vectorized torch (numerics reference).
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch


class ParamSlot:
    def __init__(self, param: torch.Tensor, lr: float):
        self.param = param
        self.lr = lr
        self.master: Optional[torch.Tensor] = None  # fp32 master for bf16
        self.m: Optional[torch.Tensor] = None
        self.v: Optional[torch.Tensor] = None

    def _lazy_state(self, kind: str):
        p = self.param
        if p.dtype != torch.float32 and self.master is None:
            self.master = p.detach().float().clone()
        dev = p.device
        n = p.numel()
        if self.v is None:
            self.v = torch.zeros(n, dtype=torch.float32, device=dev).view_as(p)
        if kind == "adam" and self.m is None:
            self.m = torch.zeros(n, dtype=torch.float32, device=dev).view_as(p)


class Updater:
    """Per-layer-LR updater over a ComputationGraph (DL4J-updater analog)."""

    def __init__(
        self,
        slots: list[ParamSlot],
        kind: str = "adam",
        beta1: float = 0.5,
        beta2: float = 0.999,
        rms_decay: float = 0.95,
        eps: float = 1e-8,
        grad_clip: float = 1.0,
        l2: float = 1e-4,
    ):
        self.slots = slots
        self.kind = kind
        self.beta1, self.beta2 = beta1, beta2
        self.rms_decay = rms_decay
        self.eps = eps
        self.grad_clip = grad_clip
        self.l2 = l2
        self.t = 0
        # global LR multiplier (schedules; trainers set it per step).
        # NOTE: reaches the fused kernel as a scalar argument, so it is
        # frozen inside a captured graph — trainers disable capture
        # when a schedule is active.
        self.lr_scale = 1.0
        # device-side step counter (hipGraph-replayable bias correction)
        self._t_dev: Optional[torch.Tensor] = None

    # ------------------------------------------------------------------
    @classmethod
    def for_graph(cls, graph, optim_cfg) -> "Updater":
        """Build from a ComputationGraph + OptimConfig (per-layer lr)."""
        slots: list[ParamSlot] = []
        for name, layer in graph.layers.items():
            lr = layer.lr
            if lr is None:
                lr = optim_cfg.dis_learning_rate
            if layer.frozen:
                lr = 0.0
            for p in layer.parameters(recurse=False):
                slots.append(ParamSlot(p, lr))
        return cls(
            slots,
            kind=optim_cfg.optimizer,
            beta1=optim_cfg.beta1,
            beta2=optim_cfg.beta2,
            rms_decay=optim_cfg.rms_decay,
            eps=optim_cfg.epsilon,
            grad_clip=optim_cfg.grad_clip,
            l2=optim_cfg.l2,
        )

    def zero_grad(self):
        for s in self.slots:
            if s.param.grad is not None:
                s.param.grad = None

    @torch.no_grad()
    def step(self):
        self.t += 1
        if self.kind == "adam" and any(s.param.is_cuda for s in self.slots):
            if self._t_dev is None:
                dev = next(s.param.device for s in self.slots
                           if s.param.is_cuda)
                # seed from the host count so bias correction is right
                # after a checkpoint resume (self.t was already bumped)
                self._t_dev = torch.full((1,), self.t - 1,
                                         dtype=torch.int32, device=dev)
            self._t_dev += 1
        for s in self.slots:
            g = s.param.grad
            if g is None or s.lr == 0.0:
                continue
            s._lazy_state(self.kind)
            if s.param.is_cuda:
                self._step_gpu(s, g)
                # the fused kernel writes params behind torch's back: the
                # version counter does not move, so drop any packed-weight
                # cache (gpu_ops._packed) explicitly
                if hasattr(s.param, "_gdlj_cache"):
                    del s.param._gdlj_cache
            else:
                self._step_cpu(s, g)

    # ------------------------------------------------------------- paths
    def _step_cpu(self, s: ParamSlot, g: torch.Tensor):
        w = s.master if s.master is not None else s.param.data
        g = g.float()
        if self.grad_clip > 0:
            g = g.clamp(-self.grad_clip, self.grad_clip)
        if self.l2 > 0:
            g = g + self.l2 * w
        if self.kind == "adam":
            s.m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            s.v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            mhat = s.m / (1 - self.beta1 ** self.t)
            vhat = s.v / (1 - self.beta2 ** self.t)
            w.add_(-s.lr * self.lr_scale * mhat / (vhat.sqrt() + self.eps))
        elif self.kind == "rmsprop":
            s.v.mul_(self.rms_decay).addcmul_(g, g, value=1 - self.rms_decay)
            w.add_(-s.lr * self.lr_scale * g / (s.v.sqrt() + self.eps))
        else:
            raise KeyError(self.kind)
        if s.master is not None:
            s.param.data.copy_(w.to(s.param.dtype))

    def _step_gpu(self, s: ParamSlot, g: torch.Tensor):
        from . import gpu_ops

        gpu_ops.fused_update(
            kind=self.kind,
            param=s.param.data,
            grad=g,
            master=s.master,
            m=s.m,
            v=s.v,
            lr=s.lr * self.lr_scale,
            beta1=self.beta1,
            beta2=self.beta2,
            rms_decay=self.rms_decay,
            eps=self.eps,
            clip=self.grad_clip,
            l2=self.l2,
            t=self.t,
            t_dev=self._t_dev,
        )

    # -------------------------------------------------------- state io
    def state_dict(self) -> dict:
        # under hipGraph capture, replays advance only the device counter
        # _t_dev; read it back so checkpoints never store a stale t (Adam
        # bias correction would be skewed after resume)
        if self._t_dev is not None:
            self.t = int(self._t_dev.item())
        return {
            "kind": self.kind,
            "t": self.t,
            "slots": [
                {
                    "lr": s.lr,
                    "m": None if s.m is None else s.m.cpu(),
                    "v": None if s.v is None else s.v.cpu(),
                    "master": None if s.master is None else s.master.cpu(),
                }
                for s in self.slots
            ],
        }

    def load_state_dict(self, sd: dict):
        self.t = sd["t"]
        # drop any device-side step counter: it reseeds from the restored
        # host count on the next step (a stale one would skew Adam bias
        # correction after an in-process resume)
        self._t_dev = None
        for s, ss in zip(self.slots, sd["slots"]):
            s.lr = ss["lr"]
            dev = s.param.device
            s.m = None if ss["m"] is None else ss["m"].to(dev)
            s.v = None if ss["v"] is None else ss["v"].to(dev)
            s.master = None if ss["master"] is None else ss["master"].to(dev)
