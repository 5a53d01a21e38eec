"""Distributed helpers: RCCL process-group init, rank guards, DDP wrap.

Replaces the reference's scattered torch.distributed usage
(src/utils.py:29-51, run_pretraining.py:185,270) with one module.
On ROCm the ``nccl`` backend IS RCCL; gradient all-reduce runs over
xGMI (7 point-to-point links/GPU). DDP buckets are sized at 100 MB —
larger than torch's 25 MB default because with ~86 accumulation
micro-steps per update the single boundary all-reduce is latency-
amortized and fewer, larger buckets keep all 7 xGMI links busy via
RCCL's multi-channel rings; gradient_as_bucket_view avoids a second
1.3 GB gradient copy.
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


class WorkerInitObj:
    """Per-DataLoader-worker seeding (reference: src/utils.py:22-27)."""

    def __init__(self, seed: int):
        self.seed = seed

    def __call__(self, worker_id: int) -> None:
        import numpy as np  # noqa: PLC0415
        import random  # noqa: PLC0415

        np.random.seed(self.seed + worker_id)
        random.seed(self.seed + worker_id)
        info = torch.utils.data.get_worker_info()
        if info is not None and hasattr(info.dataset, "_rng"):
            import numpy as np  # noqa: PLC0415

            info.dataset._rng = np.random.default_rng(self.seed + worker_id)


def get_rank() -> int:
    return dist.get_rank() if dist.is_available() and dist.is_initialized() else 0


def get_world_size() -> int:
    return (
        dist.get_world_size()
        if dist.is_available() and dist.is_initialized()
        else 1
    )


def is_main_process() -> bool:
    return get_rank() == 0


def barrier() -> None:
    if dist.is_available() and dist.is_initialized():
        dist.barrier()


def mkdir_by_main_process(path: str) -> None:
    if is_main_process():
        os.makedirs(path, exist_ok=True)
    barrier()


def init_distributed(
    backend: Optional[str] = None, timeout_minutes: int = 30
) -> tuple[int, int, int]:
    """Initialize torch.distributed from torchrun env vars.

    Returns (rank, local_rank, world_size). No-op single-process when
    WORLD_SIZE is absent/1.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        dist.init_process_group(
            backend=backend,
            init_method="env://",
            timeout=datetime.timedelta(minutes=timeout_minutes),
        )
    elif torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world_size


def wrap_ddp(
    model: torch.nn.Module,
    local_rank: int,
    bucket_cap_mb: int = 100,
    find_unused_parameters: bool = False,
    grad_compress: Optional[str] = None,
) -> torch.nn.Module:
    """DDP wrap over RCCL (or gloo on CPU).

    ``grad_compress`` in {"bf16", "fp16"} registers the corresponding
    built-in compression comm hook: gradients are cast down for the
    xGMI all-reduce and restored after — halves inter-GPU bytes per
    update. Default off (the reference all-reduces fp32 grads for
    pretraining: run_pretraining.py:270; its SQuAD path used fp16 flat
    all-reduce via apex DDP).
    """
    if get_world_size() == 1:
        return model
    # Key off the module's actual device, not cuda.is_available(): a
    # CPU/gloo module on a GPU box must not get device_ids.
    on_cuda = any(p.is_cuda for p in model.parameters())
    device_ids = [local_rank] if on_cuda else None
    ddp = torch.nn.parallel.DistributedDataParallel(
        model,
        device_ids=device_ids,
        bucket_cap_mb=bucket_cap_mb,
        gradient_as_bucket_view=True,
        find_unused_parameters=find_unused_parameters,
    )
    if grad_compress:
        from torch.distributed.algorithms.ddp_comm_hooks import (  # noqa: PLC0415
            default_hooks,
        )

        hook = {
            "bf16": default_hooks.bf16_compress_hook,
            "fp16": default_hooks.fp16_compress_hook,
        }[grad_compress]
        ddp.register_comm_hook(dist.group.WORLD, hook)
    return ddp
