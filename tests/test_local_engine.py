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


def test_sync_engine_empty_partition_raises_clearly():
    """A partition with no rows must fail with the actionable repartition
    message, not a shape error deep in torch (reference leaves this as an
    obscure crash)."""
    import numpy as np
    import pytest
    import torch
    import torch.nn as nn

    from sparktorch_amd.compat.local import LocalDataFrame
    from sparktorch_amd.models.simple_net import Net
    from sparktorch_amd.parallel.sync import train_distributed
    from sparktorch_amd.utils.data import handle_data
    from sparktorch_amd.utils.serialize import serialize_torch_obj

    rng = np.random.RandomState(0)
    # 3 partitions but only 2 rows -> one partition is empty
    df = LocalDataFrame.from_arrays(rng.rand(2, 10), [0.0, 1.0], num_partitions=3)
    obj = serialize_torch_obj(Net(), nn.MSELoss(), torch.optim.Adam, lr=0.01)
    rdd = df.rdd.mapPartitions(handle_data("features", "label"))
    with pytest.raises(Exception, match="repartition|empty"):
        train_distributed(rdd, obj, iters=1, device="cpu", backend="gloo")
