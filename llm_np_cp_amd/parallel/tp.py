"""Tensor-parallel process-group helpers (RCCL over xGMI).

One process per GPU; ``torch.distributed`` backend "nccl" IS RCCL on
ROCm.  World size 1 means every collective is a no-op, so the TP=1 path
is the same code with collectives skipped (doubles as the CPU-CI "fake
backend" — SURVEY §4 distributed test strategy).

Sharding scheme (SURVEY §7 stage 4):
  q/k/v projections  : head-sharded rows      (column-parallel)
  o_proj             : column-sharded         (row-parallel, all-reduce)
  gate/up            : row-sharded            (column-parallel)
  down               : column-sharded         (row-parallel, all-reduce)
  lm_head            : vocab-row-sharded      (all-gather of logits)
  embeddings, norms  : replicated
  KV cache           : kv-head-sharded (per-rank pool)
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np


def init_distributed(backend: Optional[str] = None):
    """Initialize torch.distributed from torchrun env vars if present.
    Returns (rank, world_size). Safe to call with no env (1 GPU)."""
    import torch
    import torch.distributed as dist

    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world == 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    # keep RCCL capture-friendly: the watchdog/monitoring threads make
    # collectives refuse hipGraph capture on some stacks (the one-shot
    # xGMI kernels are the decode-path primary; this is fallback
    # insurance for RCCL-in-graph)
    os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "0")
    os.environ.setdefault("TORCH_NCCL_ENABLE_MONITORING", "0")
    dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


# ----------------------------------------------------------------------
# collective routing: small (decode-sized) payloads go through the
# one-shot xGMI peer-memory kernel when available (graph-capturable,
# ~1 hop); everything else through RCCL (prefill-sized payloads are
# bandwidth-bound — the ring is right for those).
# ----------------------------------------------------------------------

_XGMI = None


def init_xgmi(device, slot_bytes: int) -> bool:
    """Try to bring up the one-shot communicator (validated on hardware
    inside XgmiComm.create; all ranks agree on the outcome)."""
    global _XGMI
    import torch
    import torch.distributed as dist

    if _XGMI is not None:
        return True
    if not (dist.is_initialized() and dist.get_world_size() > 1
            and torch.cuda.is_available()):
        return False
    from .xgmi import XgmiComm

    _XGMI = XgmiComm.create(dist.get_rank(), dist.get_world_size(),
                            device, slot_bytes)
    return _XGMI is not None


def xgmi_comm():
    return _XGMI


def shutdown_xgmi():
    global _XGMI
    if _XGMI is not None:
        _XGMI.close()
        _XGMI = None


def all_reduce(t):
    if _XGMI is not None and _XGMI.fits(t):
        _XGMI.all_reduce(t)
        return
    import torch.distributed as dist

    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t)


def all_gather_into(out, t):
    if _XGMI is not None and _XGMI.fits(t) and out.is_contiguous():
        _XGMI.all_gather_into(out, t)
        return
    import torch.distributed as dist

    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_gather_into_tensor(out, t)
    else:
        out.copy_(t.view(out.shape))


def broadcast(t, src: int = 0):
    import torch.distributed as dist

    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.broadcast(t, src)


# ----------------------------------------------------------------------
# Weight sharding (numpy, done at load time; SURVEY §5 checkpoint scope:
# per-rank TP sharding happens on the host before the single H2D copy)
# ----------------------------------------------------------------------

def shard_rows(w: np.ndarray, rank: int, world: int) -> np.ndarray:
    n = w.shape[0]
    assert n % world == 0, f"cannot row-shard {w.shape} over {world}"
    c = n // world
    return np.ascontiguousarray(w[rank * c:(rank + 1) * c])


def shard_cols(w: np.ndarray, rank: int, world: int) -> np.ndarray:
    n = w.shape[1]
    assert n % world == 0, f"cannot col-shard {w.shape} over {world}"
    c = n // world
    return np.ascontiguousarray(w[:, rank * c:(rank + 1) * c])
