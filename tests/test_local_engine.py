"""LocalRDD / LocalDataFrame / barrier context tests (multi-process, gloo)."""

import numpy as np
import pytest
import torch

from sparktorch_amd.compat.local import LocalDataFrame, LocalRDD, Row


def test_local_rdd_map_collect_serial():
    rdd = LocalRDD([[1, 2], [3, 4]])
    assert rdd.collect() == [1, 2, 3, 4]
    assert rdd.getNumPartitions() == 2


def test_local_rdd_repartition():
    rdd = LocalRDD([[1, 2, 3, 4, 5, 6]])
    r2 = rdd.repartition(3)
    assert r2.getNumPartitions() == 3
    assert sorted(r2.collect()) == [1, 2, 3, 4, 5, 6]


def test_local_df_roundtrip():
    feats = np.random.randn(10, 4)
    labels = list(range(10))
    df = LocalDataFrame.from_arrays(feats, labels, num_partitions=2)
    assert df.count() == 10
    assert df.columns == ["features", "label"]
    df2 = df.withColumn("pred", [float(i) for i in range(10)])
    assert df2.collect()[3]["pred"] == 3.0
    assert df2.collect()[3].label == 3


def _barrier_worker(index, iterator):
    """Runs inside a spawned partition process: allGather ranks."""
    from sparktorch_amd.compat.barrier import get_barrier_context

    ctx = get_barrier_context()
    got = ctx.allGather(str(index))
    yield (index, sorted(got))


def test_local_barrier_allgather_two_partitions():
    rdd = LocalRDD([[0], [0]])
    out = rdd.mapPartitionsWithIndex(_barrier_worker).collect(timeout_s=120)
    assert len(out) == 2
    for _idx, gathered in out:
        assert gathered == ["0", "1"]
