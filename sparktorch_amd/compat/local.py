"""Local partitioned execution engine (the no-JVM stand-in for Spark).

Gives the framework a Spark-shaped data path when pyspark is unavailable:

* :class:`LocalRDD` — partitioned rows + a lazy mapPartitions chain; ``collect``
  runs every partition **concurrently, one OS process per partition** (barrier
  semantics, like Spark barrier execution — reference distributed.py:53-63),
  with a :class:`~sparktorch_amd.compat.barrier.LocalBarrierContext` installed
  in each worker so engines can rendezvous (allGather) and init
  torch.distributed exactly as they do under Spark.
* :class:`LocalDataFrame` — a minimal column-store with ``.rdd``,
  ``withColumn``, ``repartition``, ``collect``.

This is a *test/CPU vehicle and single-node driver*; under a real Spark
cluster the same engine code runs inside Spark barrier tasks instead.
"""

from __future__ import annotations

import multiprocessing
import socket
from typing import Any, Callable, Dict, Iterable, List, Optional, Sequence

import dill
import numpy as np


def free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _apply_chain(index: int, partition: Iterable, chain) -> Iterable:
    data = iter(partition)
    for kind, f in chain:
        if kind == "with_index":
            data = f(index, data)
        else:
            data = f(data)
    return data


def _worker_main(index: int, payload: bytes, result_q, store_host: str, store_port: int, world: int):
    import traceback

    try:
        from sparktorch_amd.compat.barrier import LocalBarrierContext, install_local_context

        chain, partition = dill.loads(payload)
        ctx = LocalBarrierContext(index, world, store_host, store_port)
        install_local_context(ctx)
        out = list(_apply_chain(index, partition, chain))
        result_q.put((index, True, dill.dumps(out)))
    except BaseException:
        result_q.put((index, False, traceback.format_exc()))


class LocalRDD:
    def __init__(self, partitions: List[list], chain=(), barrier: bool = False):
        self._partitions = partitions
        self._chain = tuple(chain)
        self._barrier = barrier

    # --- structure -----------------------------------------------------------
    def getNumPartitions(self) -> int:
        return len(self._partitions)

    def repartition(self, n: int) -> "LocalRDD":
        import random as _random

        flat: list = []
        for i, part in enumerate(self._partitions):
            flat.extend(_apply_chain(i, part, self._chain))
        # Spark's repartition is a randomizing shuffle — re-randomize row
        # placement so partitionShuffles>1 actually re-mixes data
        # (reference hogwild.py:175-177 relies on this)
        _random.Random().shuffle(flat)
        parts: List[list] = [[] for _ in range(n)]
        for i, row in enumerate(flat):
            parts[i % n].append(row)
        return LocalRDD(parts, chain=(), barrier=self._barrier)

    def barrier(self) -> "LocalRDD":
        return LocalRDD(self._partitions, self._chain, barrier=True)

    # --- transformations -----------------------------------------------------
    def mapPartitions(self, f: Callable, preservesPartitioning: bool = False) -> "LocalRDD":
        return LocalRDD(self._partitions, self._chain + (("plain", f),), self._barrier)

    def mapPartitionsWithIndex(self, f: Callable, preservesPartitioning: bool = False) -> "LocalRDD":
        return LocalRDD(self._partitions, self._chain + (("with_index", f),), self._barrier)

    # --- actions -------------------------------------------------------------
    def _materialize_serial(self) -> list:
        out: list = []
        for i, part in enumerate(self._partitions):
            out.extend(_apply_chain(i, part, self._chain))
        return out

    def collect(self, timeout_s: float = 600.0) -> list:
        if not self._chain:
            return [r for p in self._partitions for r in p]
        return self._run_parallel(timeout_s)

    def foreach(self, f: Callable) -> None:
        for item in self.collect():
            f(item)

    def count(self) -> int:
        return len(self.collect())

    def _run_parallel(self, timeout_s: float) -> list:
        """Run every partition concurrently in its own process (barrier mode)."""
        world = len(self._partitions)
        if world == 1:
            # Single partition: run inline with an in-process context.
            from sparktorch_amd.compat.barrier import LocalBarrierContext, install_local_context

            host, port = "127.0.0.1", free_port()
            store = _make_master_store(host, port, 2)
            ctx = LocalBarrierContext(0, 1, host, port)
            install_local_context(ctx)
            try:
                return list(_apply_chain(0, self._partitions[0], self._chain))
            finally:
                install_local_context(None)  # type: ignore[arg-type]
                del store

        mp = multiprocessing.get_context("spawn")
        host, port = "127.0.0.1", free_port()
        store = _make_master_store(host, port, world + 1)
        result_q = mp.Queue()
        procs = []
        for i, part in enumerate(self._partitions):
            payload = dill.dumps((self._chain, part))
            p = mp.Process(
                target=_worker_main, args=(i, payload, result_q, host, port, world), daemon=True
            )
            p.start()
            procs.append(p)

        results: Dict[int, list] = {}
        errors: List[str] = []
        try:
            for _ in range(world):
                idx, ok, blob = result_q.get(timeout=timeout_s)
                if ok:
                    results[idx] = dill.loads(blob)
                else:
                    errors.append("partition %d failed:\n%s" % (idx, blob))
                    break
        finally:
            for p in procs:
                p.join(timeout=5 if not errors else 0.5)
                if p.is_alive():
                    p.terminate()
        del store
        if errors:
            raise RuntimeError("local barrier job failed\n" + "\n".join(errors))
        out: list = []
        for i in range(world):
            out.extend(results.get(i, []))
        return out


def _make_master_store(host: str, port: int, world: int):
    from torch.distributed import TCPStore

    return TCPStore(host, port, world, is_master=True, wait_for_workers=False)


class Row(dict):
    """Dict-backed row with attribute access, like pyspark.sql.Row."""

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError as e:
            raise AttributeError(name) from e


class LocalDataFrame:
    """Columnar-enough local DataFrame: list of Rows + partition count."""

    def __init__(self, rows: List[Row], num_partitions: int = 1):
        self._rows = [r if isinstance(r, Row) else Row(r) for r in rows]
        self._num_partitions = max(1, num_partitions)

    # construction helpers
    @classmethod
    def from_arrays(
        cls,
        features: np.ndarray,
        labels: Optional[Sequence] = None,
        feature_col: str = "features",
        label_col: str = "label",
        num_partitions: int = 1,
    ) -> "LocalDataFrame":
        rows = []
        for i in range(len(features)):
            r = {feature_col: np.asarray(features[i])}
            if labels is not None:
                r[label_col] = labels[i]
            rows.append(Row(r))
        return cls(rows, num_partitions)

    @property
    def rdd(self) -> LocalRDD:
        n = self._num_partitions
        parts: List[list] = [[] for _ in range(n)]
        for i, row in enumerate(self._rows):
            parts[i % n].append(row)
        return LocalRDD(parts)

    @property
    def columns(self) -> List[str]:
        return list(self._rows[0].keys()) if self._rows else []

    def repartition(self, n: int) -> "LocalDataFrame":
        return LocalDataFrame(self._rows, n)

    def withColumn(self, name: str, values: Sequence) -> "LocalDataFrame":
        if len(values) != len(self._rows):
            raise ValueError("column length mismatch")
        rows = []
        for row, v in zip(self._rows, values):
            nr = Row(row)
            nr[name] = v
            rows.append(nr)
        return LocalDataFrame(rows, self._num_partitions)

    def select(self, *cols: str) -> "LocalDataFrame":
        return LocalDataFrame([Row({c: r[c] for c in cols}) for r in self._rows], self._num_partitions)

    def collect(self) -> List[Row]:
        return list(self._rows)

    def count(self) -> int:
        return len(self._rows)

    def take(self, n: int) -> List[Row]:
        return self._rows[:n]
