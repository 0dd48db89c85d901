"""Helper module for tests/test_ps_stress.py::test_barrier_fail_fast_on_dead_worker.

Spawn-based multiprocessing pickles the mapped function by reference, so the
crash function must live in an importable module (not a test-local lambda).
"""

import os


def crash_partition_one(idx, it):
    rows = list(it)
    if idx == 1:
        os._exit(3)  # die without reporting — simulates an OOM-killed task
    return iter([(idx, len(rows))])
