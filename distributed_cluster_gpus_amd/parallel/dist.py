"""Distributed helpers: one process per GPU, torch.distributed over RCCL/xGMI.

The reference has NO distributed backend (SURVEY §2 "Parallelism ... NONE");
these are the new framework's first-class distributed components, mandated by
BASELINE.json:

* ``init_distributed``  — rendezvous from torchrun env vars; backend "nccl"
  (= RCCL on ROCm) for GPU runs, "gloo" for CPU tests.
* ``allreduce_gradients`` — ONE fused flat all-reduce of all grads.  The
  CHSAC-AF nets total ~0.6 M fp32 params (~2.3 MB), so the collective is
  latency-bound on the fully-connected xGMI mesh (7 p2p links/GPU at
  ~153 GB/s); a single flat buffer minimizes launch/collective count —
  bucketed overlap would only pay for multi-hundred-MB gradients.
* ``allreduce_tensor_sum`` / ``allreduce_scalar`` — replica-shard metric
  reductions (energy totals, latency histograms) for the batched engine.
"""
import os
from typing import List, Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None, device: Optional[torch.device] = None):
    """Initialize from torchrun/driver env (RANK/WORLD_SIZE/MASTER_*).
    No-op when WORLD_SIZE is absent or 1.  Returns (rank, world_size)."""
    if dist.is_available() and not dist.is_initialized() and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)
    return rank(), world_size()


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def barrier():
    if is_distributed():
        dist.barrier()


def allreduce_gradients(params: List[torch.Tensor]):
    """Average gradients across ranks with one flat fused all-reduce."""
    if not is_distributed():
        return
    grads = [p.grad for p in params if p is not None and p.grad is not None]
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    dist.all_reduce(flat, op=dist.ReduceOp.SUM)
    flat.div_(world_size())
    for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)


def allreduce_tensor_sum(t: torch.Tensor) -> torch.Tensor:
    if is_distributed():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def allreduce_scalar(x: float, device=None) -> float:
    if not is_distributed():
        return x
    dev = device
    if dev is None:
        dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    t = torch.tensor([x], dtype=torch.float64, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return float(t.item())


def _collective_device() -> torch.device:
    """Tensor device for small control collectives: RCCL needs device
    tensors, gloo wants host tensors."""
    if is_distributed() and dist.get_backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def dp_sync_step(err: int, local_done: bool, replay_size: int, n_new: int):
    """One per-launch control synchronization for data-parallel RL training.

    Every rank of the batched chsac_af engine calls this exactly once per
    advance-launch, so the training collectives that follow are structurally
    paired across ranks: the per-launch SAC step count is derived from the
    GLOBAL transition count and the warmup gate from the global MIN replay
    size, so no rank can run a different number of all-reduces (the round-1
    advisor's mispairing/hang finding).

    Returns (err_any, all_done, min_replay_size, total_new_transitions).
    No-op passthrough when torch.distributed is not initialized.
    """
    if not is_distributed():
        return err, local_done, replay_size, n_new
    dev = _collective_device()
    mx = torch.tensor([float(err), 0.0 if local_done else 1.0,
                       -float(replay_size)], dtype=torch.float64, device=dev)
    dist.all_reduce(mx, op=dist.ReduceOp.MAX)
    sm = torch.tensor([float(n_new)], dtype=torch.float64, device=dev)
    dist.all_reduce(sm, op=dist.ReduceOp.SUM)
    vals = mx.cpu().tolist()
    return int(vals[0]), vals[1] == 0.0, int(-vals[2]), int(sm.cpu().item())


def broadcast_module(module: torch.nn.Module, src: int = 0):
    """Broadcast parameters+buffers from src so all DP replicas start equal."""
    if not is_distributed():
        return
    for t in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(t.data, src=src)
