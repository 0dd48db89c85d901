"""Barrier-task context abstraction.

The sync engine needs, per worker: its rank, the world size, the host list,
``allGather(str) -> [str]`` and ``barrier()``.  On Spark this is
``pyspark.BarrierTaskContext`` (reference distributed.py:98-110); on the local
executor it is a TCPStore-backed equivalent installed by
:mod:`sparktorch_amd.compat.local` before the worker function runs.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional

_LOCAL_CTX = None


class LocalBarrierContext:
    """TCPStore-backed stand-in for pyspark.BarrierTaskContext.

    One parent-owned ``torch.distributed.TCPStore`` master; every worker holds
    a client.  ``allGather``/``barrier`` run over store keys with a per-context
    sequence number so repeated calls never collide.
    """

    def __init__(self, rank: int, world_size: int, host: str, port: int, timeout_s: float = 120.0):
        from torch.distributed import TCPStore

        self._rank = rank
        self._world = world_size
        self._seq = 0
        self._store = TCPStore(
            host,
            port,
            world_size + 1,
            is_master=False,
            timeout=datetime.timedelta(seconds=timeout_s),
        )

    def partitionId(self) -> int:
        return self._rank

    def getTaskInfos(self):
        class _Info:
            def __init__(self, address):
                self.address = address

        return [_Info("127.0.0.1:0") for _ in range(self._world)]

    def allGather(self, message: str = "") -> List[str]:
        seq = self._seq
        self._seq += 1
        self._store.set("ag_%d_%d" % (seq, self._rank), message)
        out = []
        for r in range(self._world):
            out.append(self._store.get("ag_%d_%d" % (seq, r)).decode("utf-8"))
        return out

    def barrier(self) -> None:
        self.allGather("")


def install_local_context(ctx: LocalBarrierContext) -> None:
    global _LOCAL_CTX
    _LOCAL_CTX = ctx


def get_barrier_context(use_barrier: bool = True):
    """Return the active barrier context: Spark's if running inside a Spark
    barrier task, otherwise the installed local one."""
    if _LOCAL_CTX is not None:
        return _LOCAL_CTX
    try:  # Spark barrier tasks (real pyspark or the vendored double)
        from pyspark import BarrierTaskContext

        return BarrierTaskContext.get()
    except Exception:
        pass
    raise RuntimeError(
        "no barrier context available: not inside a Spark barrier task and no "
        "local context installed"
    )
