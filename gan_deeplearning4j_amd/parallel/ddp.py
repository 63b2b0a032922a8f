"""Data parallelism: bucketed gradient all-reduce overlapped with backward.

MI355X-native replacement for the reference's ParameterAveragingTrainingMaster
(Java:325-330; math in gan.ipynb cell 3 "Synchronous Parameter Averaging"):

- `GradReducer`: flat-bucket RCCL all-reduce launched from
  post-accumulate-grad hooks, so communication overlaps the rest of
  backward. xGMI is point-to-point (7 links x ~153 GB/s); GAN gradients
  total only tens of MB, so the default is FEW, LARGE buckets
  (latency-dominated regime — SURVEY.md §5 'distributed backend').
- `average_parameters`: explicit parameter averaging for DL4J semantic
  parity (`averaging_frequency`-style local steps between averaging
  rounds).
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch
import torch.distributed as dist


class _Bucket:
    def __init__(self, params: list[torch.Tensor], comm_dtype: torch.dtype):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.comm_dtype = comm_dtype
        self.flat: Optional[torch.Tensor] = None
        self.ready = 0
        self.work = None

    def lazy_flat(self):
        if self.flat is None:
            p0 = self.params[0]
            self.flat = torch.zeros(
                self.numel, dtype=self.comm_dtype, device=p0.device
            )
        return self.flat


class GradReducer:
    """Bucketed, overlapped gradient all-reduce over module parameters.

    Usage per step:
        reducer.prepare()        # before backward
        loss.backward()          # hooks fire, buckets reduce async
        reducer.finish()         # wait + write averaged grads back
    With world_size==1 (or dist uninitialized) everything is a no-op,
    preserving the reference's single-process property.
    """

    def __init__(
        self,
        modules: Iterable[torch.nn.Module],
        bucket_cap_mb: int = 40,
        local_steps: int = 1,
        sync_buffers: bool = True,
        buffer_modules: Optional[Iterable[torch.nn.Module]] = None,
        comm: str = "fp32",
    ):
        modules = list(modules)
        self.params = [
            p
            for m in modules
            for p in m.parameters()
            if p.requires_grad
        ]
        # BatchNorm running statistics update from PER-RANK batch stats in
        # forward; without syncing them the replicas' eval paths (and
        # checkpoints — DL4J's flat params() vector includes mean/var)
        # drift apart. They are a few KB per model: average them every
        # reduction round. buffer_modules widens the set beyond `modules`
        # (e.g. the step's LAST reducer also owns the other model's stats,
        # which keep updating in later phases of the same step).
        self.sync_buffers = sync_buffers
        self._buffers = [
            b for m in (list(buffer_modules) if buffer_modules is not None
                        else modules)
            for b in m.buffers() if b.dtype.is_floating_point
        ]
        self.local_steps = max(1, local_steps)
        self._step = 0
        self.enabled = dist.is_initialized() and dist.get_world_size() > 1
        self._active = False
        self.buckets: list[_Bucket] = []
        self._param_bucket: dict[int, _Bucket] = {}
        if self.enabled:
            # Gradient sums communicate in fp32 by default: an 8-rank bf16
            # ring-SUM loses ~log2(8) mantissa bits on every gradient, and
            # at these sizes (GAN grads total 16-60 MB) the fp32 all-reduce
            # costs ~0.4 ms overlapped with backward — noise at 70 ms
            # steps. bf16 comms remain an explicit opt-in for
            # bandwidth-bound regimes (comm="bf16" or GDLJ_COMM_DTYPE=bf16;
            # gloo lacks bf16 so it always uses fp32).
            import os as _os

            comm = _os.environ.get("GDLJ_COMM_DTYPE", comm)
            self.comm_dtype = (
                torch.bfloat16
                if comm == "bf16"
                and dist.get_backend() == "nccl"
                and all(p.dtype == torch.bfloat16 for p in self.params)
                else torch.float32
            )
            self._build_buckets(bucket_cap_mb)
            for p in self.params:
                p.register_post_accumulate_grad_hook(self._hook)

    def _build_buckets(self, cap_mb: int):
        elt = 2 if self.comm_dtype == torch.bfloat16 else 4
        cap = cap_mb * (1 << 20) // elt
        cur: list[torch.Tensor] = []
        size = 0
        # reverse order ~ backward completion order (last layers first)
        for p in reversed(self.params):
            cur.append(p)
            size += p.numel()
            if size >= cap:
                self.buckets.append(_Bucket(cur, self.comm_dtype))
                cur, size = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, self.comm_dtype))
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[id(p)] = b

    # ------------------------------------------------------------ hooks
    def _hook(self, p: torch.Tensor):
        if not self._active:
            return
        b = self._param_bucket[id(p)]
        b.ready += 1
        if b.ready == len(b.params):
            self._reduce_bucket(b)

    def _reduce_bucket(self, b: _Bucket):
        flat = b.lazy_flat()
        off = 0
        for p in b.params:
            n = p.numel()
            flat[off : off + n].copy_(
                p.grad.detach().reshape(-1).to(b.comm_dtype)
            )
            off += n
        b.work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)

    # ------------------------------------------------------------- api
    def prepare(self):
        self._step += 1
        self._active = (
            self.enabled and (self._step % self.local_steps == 0)
        )
        for b in self.buckets:
            b.ready = 0
            b.work = None

    def finish(self):
        if not self._active:
            return
        world = dist.get_world_size()
        for b in self.buckets:
            if b.work is None and b.ready == len(b.params):
                self._reduce_bucket(b)
            if b.work is not None:
                b.work.wait()
                off = 0
                for p in b.params:
                    n = p.numel()
                    p.grad.detach().reshape(-1).copy_(
                        (b.flat[off : off + n] / world).to(p.grad.dtype)
                    )
                    off += n
        if self.sync_buffers and self._buffers:
            # one flat collective (a dozen KB-sized all-reduces would be
            # pure latency on xGMI)
            with torch.no_grad():
                flat = torch.cat([b.detach().reshape(-1).float()
                                  for b in self._buffers])
                dist.all_reduce(flat, op=dist.ReduceOp.SUM)
                flat /= world
                off = 0
                for b in self._buffers:
                    n = b.numel()
                    b.copy_(flat[off:off + n].view_as(b).to(b.dtype))
                    off += n
        self._active = False


@torch.no_grad()
def average_parameters(module: torch.nn.Module):
    """Synchronous parameter averaging (exact DL4J TrainingMaster semantics:
    broadcast/average the PARAMETER vector, not gradients). BatchNorm
    running statistics are averaged too — they are part of DL4J's flat
    params() vector (the reference syncs mean/var by name, Java:445-455),
    and leaving them per-rank would diverge the replicas' eval paths."""
    if not (dist.is_initialized() and dist.get_world_size() > 1):
        return
    world = dist.get_world_size()
    for p in module.parameters():
        t = p.data.float()
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        p.data.copy_((t / world).to(p.dtype))
        if hasattr(p, "_gdlj_cache"):
            del p._gdlj_cache
    for b in module.buffers():
        if not b.dtype.is_floating_point:
            continue
        t = b.data.float()
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        b.data.copy_((t / world).to(b.dtype))


@torch.no_grad()
def broadcast_parameters(module: torch.nn.Module, src: int = 0):
    """Broadcast rank-0 params (initial replica sync)."""
    if not (dist.is_initialized() and dist.get_world_size() > 1):
        return
    for p in module.parameters():
        dist.broadcast(p.data, src=src)
    for b in module.buffers():
        dist.broadcast(b.data, src=src)
