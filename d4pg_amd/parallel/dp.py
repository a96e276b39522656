"""Learner data parallelism over torch.distributed (RCCL on ROCm).

Two flavors, both re-expressing the reference's shared-memory parameter
coupling (/root/reference/ddpg.py:104-120) as explicit
collectives over xGMI:

* ``DPEngine`` — SYNCHRONOUS gradient data parallelism for the fused HIP
  engine: every rank owns a full engine + its own replay shard and samples
  its own batch; per train step the gradient slabs are all-reduce-AVERAGED
  between the backward and Adam phases (engine split-step API,
  ops/hip/engine.hip PH_* masks), so N ranks at batch B are numerically a
  single learner at batch N*B (means of per-rank batch means == the global
  batch mean).  The reference's update ORDER is preserved: critic grads →
  (all-reduce) → critic Adam → policy forward/backward → (all-reduce) →
  actor Adam (ddpg.py:229-244).  This is the multi-GPU mode for the
  MFMA-bound wide config (BASELINE config 5).

* ``LocalSGDSync`` — HogWild re-expression for the latency-bound flagship
  config: every rank trains independently for ``sync_every`` steps (one
  persistent-megakernel launch), then the PARAMETER slabs (actor, critic
  and both targets) are all-reduce-averaged.  This is the reference's
  asynchronous shared-parameter scheme (every worker steps a shared model,
  main.py:303-307) with the unbounded HogWild staleness replaced by a
  bounded, chosen window — per-step gradient sync would serialize a 0.3 ms
  step behind two collective latencies, exactly the tradeoff SURVEY §2b's
  xGMI note warns about.

Both use zero-copy device views of the engine slabs (FusedEngine.device_slab
-> torch.from_blob on HBM) when the process group speaks RCCL, and stage
through the host when the group is gloo (CPU test topology).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..ops import FusedEngine


def _comm_is_device(group=None) -> bool:
    """True when the process-group backend can reduce CUDA tensors in
    place (nccl == RCCL on ROCm); gloo stages through the host."""
    backend = dist.get_backend(group)
    return str(backend) == "nccl" and torch.cuda.is_available()


class _SlabComm:
    """All-reduce-average engine slabs over the process group, zero-copy
    on RCCL, host-staged on gloo."""

    def __init__(self, engine: FusedEngine, names, group=None):
        self.engine = engine
        self.names = list(names)
        self.group = group
        self.world = dist.get_world_size(group)
        self.on_device = _comm_is_device(group)
        if self.on_device:
            self.views = [engine.device_slab(n) for n in self.names]

    def allreduce_average(self):
        if self.on_device:
            # RCCL reduces the HBM slabs in place; sync orders the NCCL
            # stream against the engine's private stream before the next
            # engine launch consumes the result.
            for v in self.views:
                dist.all_reduce(v, group=self.group)
                v.div_(self.world)
            torch.cuda.synchronize()
        else:
            for n in self.names:
                t = self.engine.store_slab(n)
                dist.all_reduce(t, group=self.group)
                t.div_(self.world)
                self.engine.load_slab(n, t)

    def broadcast(self, src=0):
        if self.on_device:
            for v in self.views:
                dist.broadcast(v, src=src, group=self.group)
            torch.cuda.synchronize()
        else:
            for n in self.names:
                t = self.engine.store_slab(n)
                dist.broadcast(t, src=src, group=self.group)
                self.engine.load_slab(n, t)


class DPEngine:
    """Synchronous gradient-DP wrapper around a FusedEngine (see module
    docstring).  train_steps(n) == n data-parallel train steps."""

    def __init__(self, engine: FusedEngine, group=None):
        self.engine = engine
        self.group = group
        self.world = dist.get_world_size(group)
        self._gc = _SlabComm(engine, ["g_critic"], group)
        self._ga = _SlabComm(engine, ["g_actor"], group)

    def train_steps(self, n: int = 1):
        E = self.engine
        for _ in range(int(n)):
            E.step_part(E.PH_CRITIC_GRADS)
            self._gc.allreduce_average()
            E.step_part(E.PH_CRITIC_APPLY)
            E.step_part(E.PH_ACTOR_GRADS)
            self._ga.allreduce_average()
            E.step_part(E.PH_ACTOR_APPLY)


class LocalSGDSync:
    """Parameter-averaging sync for independent engines (see module
    docstring).  Call average() every sync_every local steps."""

    PARAM_SLABS = ("actor", "actor_target", "critic", "critic_target")

    def __init__(self, engine: FusedEngine, group=None):
        self.comm = _SlabComm(engine, self.PARAM_SLABS, group)

    def broadcast_initial(self, src=0):
        """Start all ranks from rank-src's exact parameters (the reference
        starts every worker from the one shared global model,
        main.py:382-388)."""
        self.comm.broadcast(src=src)

    def average(self):
        self.comm.allreduce_average()


def eager_grad_sync(group=None):
    """grad_sync hook for the eager (torch) backend: all-reduce-average a
    module's .grad tensors — assign to DDPG.grad_sync to turn N eager
    agents into one synchronous-DP learner (the CPU/gloo twin of DPEngine,
    used by the parity tests)."""
    world = dist.get_world_size(group)

    def hook(module: torch.nn.Module):
        for p in module.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad, group=group)
                p.grad.div_(world)
    return hook
