"""Data-parallel PPO over RCCL/xGMI (BASELINE config #3).

The reference has no distributed communication at all; this
module is the MI355X-native design: one process per GPU, torch.distributed
backend "nccl" (RCCL on ROCm), gradients in ONE flat bucket per model
(~0.4 MB for the MLP, ~1.3 MB for the LSTM) so the all-reduce is
latency-bound on the fully-connected xGMI octet — a single fused bucket per
minibatch beats any bucketing scheme at this size.

Overlap: the all-reduce is issued async after backward; the NEXT minibatch's
Feistel gather (parameter-independent) runs on the compute stream underneath
it, and only the optimizer step waits on the collective
(PPOTrainer.update).  Episode metrics reduce over the same communicator.

Rendezvous uses MASTER_ADDR/MASTER_PORT (always 127.0.0.1 for single-node
xGMI work — container hostnames may not resolve).
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch


def init_from_env(device: torch.device) -> Tuple[int, int, Optional[object]]:
    """Initialize torch.distributed from torchrun env vars.

    Returns (rank, world_size, process_group); world_size==1 -> no init.
    Backend: nccl (=RCCL) on GPU, gloo on CPU.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world_size <= 1:
        return 0, 1, None
    import torch.distributed as dist

    if not dist.is_initialized():
        backend = "nccl" if device.type == "cuda" else "gloo"
        dist.init_process_group(backend=backend)
    return rank, world_size, dist.group.WORLD


class GradAllReducer:
    """Async all-reduce of one flat gradient bucket; mean over ranks.

    start() issues the collective (RCCL kernel on the PG stream, ordered
    after everything queued on the current stream); finish() makes the
    current stream wait and applies the 1/world_size scale.  Work enqueued
    between the two overlaps the collective.
    """

    def __init__(self, world_size: int, process_group=None,
                 timeout_s: float = 300.0):
        self.world_size = world_size
        self.pg = process_group
        self.timeout_s = timeout_s
        self._work = None

    def start(self, grads: torch.Tensor) -> None:
        if self.world_size <= 1:
            return
        import torch.distributed as dist

        self._work = dist.all_reduce(grads, op=dist.ReduceOp.SUM,
                                     group=self.pg, async_op=True)
        self._grads = grads

    def finish(self) -> None:
        """Wait with a watchdog: a dead peer rank surfaces as a clear error
        instead of an indefinite hang (failure-detection stance of
        SURVEY.md §5.3 — fail loudly at the collective boundary)."""
        if self.world_size <= 1 or self._work is None:
            return
        from datetime import timedelta

        ok = self._work.wait(timedelta(seconds=self.timeout_s))
        if ok is False:  # gloo returns False on timeout
            raise RuntimeError(
                f"gradient all-reduce timed out after {self.timeout_s}s "
                "(peer rank dead or desynchronized)")
        self._grads.mul_(1.0 / self.world_size)
        self._work = None


def allreduce_mean_(t: torch.Tensor, world_size: int, process_group=None) -> torch.Tensor:
    """In-place mean over ranks (episode metrics / eval scalars)."""
    if world_size <= 1:
        return t
    import torch.distributed as dist

    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=process_group)
    t.mul_(1.0 / world_size)
    return t


def barrier(world_size: int) -> None:
    if world_size <= 1:
        return
    import torch.distributed as dist

    dist.barrier()
