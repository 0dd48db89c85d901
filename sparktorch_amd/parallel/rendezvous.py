"""Process-group rendezvous for barrier workers.

Mirrors the reference's flow (distributed.py:35-50,95-110): rank 0 binds a free
port, broadcasts it through the barrier context's ``allGather``, every rank
sets MASTER_* env and joins ``torch.distributed``.  MI355X-native difference:
the backend is **RCCL** (``"nccl"`` on ROCm) whenever the worker drives a GPU,
gloo only for CPU-only runs; each worker pins exactly one GPU
(rank % visible devices) before the process group is created so RCCL
communicators bind to the right device.
"""

from __future__ import annotations

import datetime
import os
import socket
from typing import Optional

import torch
import torch.distributed as dist


def get_available_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def cleanup_stale_process_group() -> None:
    """Tear down a leaked process group from a retried task
    (reference distributed.py:95-96)."""
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()


def pick_device(device: str, rank: int) -> str:
    """Resolve a worker's device; pins one GPU per worker."""
    if device and device.startswith("cuda"):
        if not torch.cuda.is_available():
            raise RuntimeError(
                "device=%r requested but torch.cuda.is_available() is False; "
                "refusing to fall back silently" % device
            )
        if ":" in device:
            local = int(device.split(":")[1])
        else:
            local = rank % torch.cuda.device_count()
        torch.cuda.set_device(local)
        return "cuda:%d" % local
    return "cpu"


def select_backend(
    device: str,
    backend: Optional[str] = None,
    world_size: int = 1,
    local_ranks: Optional[int] = None,
) -> str:
    """Pick the collective backend.

    RCCL wants exactly one rank per GPU per host ("Duplicate GPU detected"
    otherwise), so the comparison is ranks-on-THIS-host vs visible GPUs —
    not global world size, which would wrongly demote any multi-node run
    (e.g. 2x8 ranks on 8-GPU hosts) to gloo.  ``local_ranks`` comes from
    LOCAL_WORLD_SIZE or from counting barrier-task addresses sharing this
    host; absent both, assume the whole world is local (single-node).
    """
    if backend:
        return backend
    if device.startswith("cuda"):
        n_gpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
        if local_ranks is None:
            env = os.environ.get("LOCAL_WORLD_SIZE")
            local_ranks = int(env) if env else world_size
        if local_ranks <= n_gpu:
            return "nccl"
        import warnings

        warnings.warn(
            "falling back to gloo collectives: %d ranks share this host's %d "
            "visible GPU(s); give each rank its own GPU for RCCL over xGMI"
            % (local_ranks, n_gpu)
        )
        return "gloo"
    return "gloo"


def init_process_group_from_barrier(
    ctx,
    rank: int,
    device: str = "cpu",
    backend: Optional[str] = None,
    timeout_s: float = 300.0,
) -> int:
    """Rendezvous via the barrier context and join torch.distributed.

    Returns the world size.  Reference flow: distributed.py:98-110.
    """
    infos = ctx.getTaskInfos()
    world_size = len(infos)

    port = str(get_available_port()) if rank == 0 else ""
    ports = ctx.allGather(port)
    master_port = next(p for p in ports if p)

    master_host = infos[0].address.split(":")[0] if infos else "127.0.0.1"
    if master_host in ("", "0.0.0.0", "localhost"):
        master_host = "127.0.0.1"

    # ranks co-located on this host = tasks whose executor address shares
    # our host (drives the nccl-vs-gloo choice in select_backend)
    hosts = [i.address.split(":")[0] for i in infos]
    local_ranks = hosts.count(hosts[rank]) if rank < len(hosts) else world_size

    os.environ["MASTER_ADDR"] = master_host
    os.environ["MASTER_PORT"] = master_port
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)

    dist.init_process_group(
        select_backend(device, backend, world_size, local_ranks=local_ranks),
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    return world_size
