"""SparkTorch estimator and SparkTorchModel transformer.

API parity with reference sparktorch/torch_distributed.py:59-358 — same 17
Params (names, defaults, semantics; torch_distributed.py:143-201), same fit ->
transform flow, same dill/base64 ``modStr`` carrying the trained net.

MI355X-native differences:
* sync mode rendezvouses RCCL over xGMI (one barrier task = one GPU) through
  :mod:`sparktorch_amd.parallel.sync`;
* inference batches partition rows into large tensors (HIP-graph-captured
  forward on GPU) instead of a batch-1 scalar UDF per row
  (reference torch_distributed.py:106-128);
* ``partitionShuffles`` is honored in sync mode (the reference hardcodes 1,
  torch_distributed.py:309);
* ``compileMode`` actually compiles the model (the reference's call is
  broken, distributed.py:117-118) and adds mode ``"hipgraph"``.

Works against a real pyspark DataFrame (when pyspark is installed) or the
local no-JVM DataFrame (:mod:`sparktorch_amd.compat.local`).
"""

from __future__ import annotations

from typing import Any, List, Optional

import numpy as np
import torch

from sparktorch_amd.compat.local import LocalDataFrame, LocalRDD
from sparktorch_amd.compat.params import (
    HAS_PYSPARK,
    Estimator,
    HasInputCol,
    HasLabelCol,
    HasPredictionCol,
    Identifiable,
    MLReadable,
    MLWritable,
    Model,
    Param,
    Params,
    TypeConverters,
    keyword_only,
)
from sparktorch_amd.pipeline_util import PysparkReaderWriter
from sparktorch_amd.utils.codec import b64_to_obj, obj_to_b64
from sparktorch_amd.utils.data import handle_data, handle_features
from sparktorch_amd.utils.serialize import load_torch_model


def _is_local_df(dataset) -> bool:
    return isinstance(dataset, LocalDataFrame)


class SparkTorchModel(
    Model,
    HasInputCol,
    HasPredictionCol,
    PysparkReaderWriter,
    MLReadable,
    MLWritable,
    Identifiable,
):
    """Fitted transformer holding the trained net as dill/base64 ``modStr``
    (reference torch_distributed.py:59-130)."""

    modStr = Param(Params._dummy(), "modStr", "serialized pytorch model", TypeConverters.toString)
    useVectorOut = Param(Params._dummy(), "useVectorOut", "vector output", TypeConverters.toBoolean)
    batchSize = Param(Params._dummy(), "batchSize", "inference batch size", TypeConverters.toInt)
    device = Param(Params._dummy(), "device", "inference device", TypeConverters.toString)

    @keyword_only
    def __init__(self, inputCol=None, predictionCol=None, modStr=None, useVectorOut=None,
                 batchSize=None, device=None):
        super().__init__()
        self._setDefault(
            inputCol="features",
            predictionCol="predicted",
            modStr="",
            useVectorOut=False,
            batchSize=8192,
            device=None,
        )
        kwargs = self._input_kwargs
        self.setParams(**kwargs)

    @keyword_only
    def setParams(self, inputCol=None, predictionCol=None, modStr=None, useVectorOut=None,
                  batchSize=None, device=None):
        kwargs = self._input_kwargs
        return self._set(**kwargs)

    def getPytorchModel(self) -> torch.nn.Module:
        return b64_to_obj(self.getOrDefault(self.modStr))

    # ------------------------------------------------------------------
    def _resolve_device(self) -> str:
        dev = self.getOrDefault(self.device)
        if dev:
            return dev
        return "cuda:0" if torch.cuda.is_available() else "cpu"

    def _predict_batches(self, feats: np.ndarray, use_vector: bool, device: str) -> List[Any]:
        """Batched forward over packed rows — the mapPartitions replacement for
        the reference's row-at-a-time UDF.  On GPU the forward is HIP-graph
        captured per batch shape."""
        model = self.getPytorchModel().to(device)
        model.eval()
        bs = int(self.getOrDefault(self.batchSize))
        runner = None
        if device.startswith("cuda"):
            from sparktorch_amd.ops.graph import GraphedForward

            runner = GraphedForward(model, device=device, batch_size=bs)
        out: List[Any] = []
        with torch.no_grad():
            for s in range(0, len(feats), bs):
                xb = torch.from_numpy(feats[s : s + bs]).float()
                if runner is not None:
                    pred = runner(xb)
                else:
                    pred = model(xb.to(device))
                pred = pred.detach().cpu().numpy()
                if use_vector:
                    out.extend([row.astype(np.float64) for row in pred])
                else:
                    if pred.ndim > 1 and pred.shape[1] > 1:
                        out.extend(np.argmax(pred, axis=1).astype(np.float64).tolist())
                    else:
                        out.extend(pred.reshape(-1).astype(np.float64).tolist())
        return out

    def _transform(self, dataset):
        inp = self.getOrDefault(self.inputCol)
        out_col = self.getOrDefault(self.predictionCol)
        use_vector = bool(self.getOrDefault(self.useVectorOut))

        if _is_local_df(dataset):
            rows = dataset.collect()
            if not rows:
                return dataset
            feats = np.stack(
                [np.asarray(r[inp].toArray() if hasattr(r[inp], "toArray") else r[inp], dtype=np.float32)
                 for r in rows]
            )
            preds = self._predict_batches(feats, use_vector, self._resolve_device())
            return dataset.withColumn(out_col, preds)

        # pyspark path: batched mapPartitions, then rebuild the DataFrame with
        # the appended prediction column.  Exercised against real pyspark or
        # the vendored double (vendor/pyspark).
        if not HAS_PYSPARK:
            raise RuntimeError("unsupported dataset type: %r" % type(dataset))
        from pyspark.ml.linalg import Vectors, VectorUDT
        from pyspark.sql.types import DoubleType, StructField

        bs = int(self.getOrDefault(self.batchSize))
        dev = self._resolve_device()
        # broadcast the deserialized net once (reference torch_distributed.py:104)
        # instead of re-deserializing the dill payload in every partition task
        bc_model = dataset.sparkSession.sparkContext.broadcast(self.getPytorchModel())

        def map_parts(partition):
            import numpy as _np
            import torch as _torch

            model = bc_model.value.to(dev)
            model.eval()
            rows = list(partition)
            if not rows:
                return
            feats = _np.stack(
                [_np.asarray(r[inp].toArray(), dtype=_np.float32) for r in rows]
            )
            with _torch.no_grad():
                for s in range(0, len(rows), bs):
                    xb = _torch.from_numpy(feats[s : s + bs]).to(dev)
                    pred = model(xb).detach().cpu().numpy()
                    for j, row in enumerate(rows[s : s + bs]):
                        if use_vector:
                            yield row + (Vectors.dense(pred[j].astype(_np.float64).flatten()),)
                        elif pred.ndim > 1 and pred.shape[1] > 1:
                            yield row + (float(_np.argmax(pred[j])),)
                        else:
                            yield row + (float(pred[j].reshape(-1)[0]),)

        schema = dataset.schema.add(
            StructField(out_col, VectorUDT() if use_vector else DoubleType())
        )
        return dataset.rdd.mapPartitions(map_parts).toDF(schema)


class SparkTorch(
    Estimator,
    HasInputCol,
    HasPredictionCol,
    HasLabelCol,
    PysparkReaderWriter,
    MLReadable,
    MLWritable,
    Identifiable,
):
    """Spark ML Estimator for distributed PyTorch training on MI355X nodes.

    The 17 Params mirror reference torch_distributed.py:143-201 exactly."""

    torchObj = Param(Params._dummy(), "torchObj", "serialized torch object", TypeConverters.toString)
    mode = Param(Params._dummy(), "mode", "training mode", TypeConverters.toString)
    device = Param(Params._dummy(), "device", "", TypeConverters.toString)
    iters = Param(Params._dummy(), "iters", "", TypeConverters.toInt)
    partitions = Param(Params._dummy(), "partitions", "", TypeConverters.toInt)
    verbose = Param(Params._dummy(), "verbose", "", TypeConverters.toInt)
    acquireLock = Param(Params._dummy(), "acquireLock", "", TypeConverters.toBoolean)
    partitionShuffles = Param(Params._dummy(), "partitionShuffles", "", TypeConverters.toInt)
    port = Param(Params._dummy(), "port", "", TypeConverters.toInt)
    useBarrier = Param(Params._dummy(), "useBarrier", "", TypeConverters.toBoolean)
    useVectorOut = Param(Params._dummy(), "useVectorOut", "", TypeConverters.toBoolean)
    earlyStopPatience = Param(Params._dummy(), "earlyStopPatience", "", TypeConverters.toInt)
    miniBatch = Param(Params._dummy(), "miniBatch", "", TypeConverters.toInt)
    validationPct = Param(Params._dummy(), "validationPct", "", TypeConverters.toFloat)
    compileMode = Param(Params._dummy(), "compileMode", "", TypeConverters.toString)
    # MI355X tunables beyond the reference's 17 Params ("adds
    # device/RCCL tunables as new Params")
    bucketCapMb = Param(Params._dummy(), "bucketCapMb",
                        "gradient all-reduce bucket size (MB)", TypeConverters.toFloat)
    backend = Param(Params._dummy(), "backend",
                    "torch.distributed backend override (nccl=RCCL / gloo)",
                    TypeConverters.toString)

    @keyword_only
    def __init__(
        self,
        inputCol=None,
        labelCol=None,
        torchObj=None,
        iters=None,
        predictionCol=None,
        partitions=None,
        acquireLock=None,
        verbose=None,
        partitionShuffles=None,
        port=None,
        useBarrier=None,
        useVectorOut=None,
        earlyStopPatience=None,
        miniBatch=None,
        validationPct=None,
        mode=None,
        device=None,
        compileMode=None,
        bucketCapMb=None,
        backend=None,
    ):
        super().__init__()
        self._setDefault(
            inputCol="features",
            labelCol=None,
            torchObj="",
            iters=10,
            predictionCol="predicted",
            partitions=-1,
            acquireLock=True,
            verbose=0,
            partitionShuffles=1,
            port=3000,
            useBarrier=False,
            useVectorOut=False,
            earlyStopPatience=-1,
            miniBatch=-1,
            validationPct=0.0,
            mode="synchronous",
            device="cpu",
            compileMode=None,
            bucketCapMb=25.0,
            backend=None,
        )
        kwargs = self._input_kwargs
        self.setParams(**kwargs)

    @keyword_only
    def setParams(
        self,
        inputCol=None,
        labelCol=None,
        torchObj=None,
        iters=None,
        predictionCol=None,
        partitions=None,
        acquireLock=None,
        verbose=None,
        partitionShuffles=None,
        port=None,
        useBarrier=None,
        useVectorOut=None,
        earlyStopPatience=None,
        miniBatch=None,
        validationPct=None,
        mode=None,
        device=None,
        compileMode=None,
        bucketCapMb=None,
        backend=None,
    ):
        kwargs = self._input_kwargs
        return self._set(**kwargs)

    # getters (parity incl. the reference's misspelled getAqcuireLock,
    # torch_distributed.py:230-273)
    def getTorchObj(self):
        return self.getOrDefault(self.torchObj)

    def getMode(self):
        return self.getOrDefault(self.mode)

    def getDevice(self):
        return self.getOrDefault(self.device)

    def getIters(self):
        return self.getOrDefault(self.iters)

    def getPartitions(self):
        return self.getOrDefault(self.partitions)

    def getVerbose(self):
        return self.getOrDefault(self.verbose)

    def getAcquireLock(self):
        return self.getOrDefault(self.acquireLock)

    getAqcuireLock = getAcquireLock  # reference public-surface typo preserved

    def getPartitionShuffles(self):
        return self.getOrDefault(self.partitionShuffles)

    def getPort(self):
        return self.getOrDefault(self.port)

    def getUseBarrier(self):
        return self.getOrDefault(self.useBarrier)

    def getUseVectorOut(self):
        return self.getOrDefault(self.useVectorOut)

    def getEarlyStopPatience(self):
        return self.getOrDefault(self.earlyStopPatience)

    def getMiniBatch(self):
        return self.getOrDefault(self.miniBatch)

    def getValidationPct(self):
        return self.getOrDefault(self.validationPct)

    def getCompileMode(self):
        return self.getOrDefault(self.compileMode)

    def getBucketCapMb(self):
        return self.getOrDefault(self.bucketCapMb)

    def getBackend(self):
        return self.getOrDefault(self.backend)

    # ------------------------------------------------------------------
    def _fit(self, dataset) -> SparkTorchModel:
        inp = self.getOrDefault(self.inputCol)
        label = self.getOrDefault(self.labelCol)
        pred_col = self.getOrDefault(self.predictionCol)
        torch_obj = self.getOrDefault(self.torchObj)
        mode = self.getOrDefault(self.mode)
        device = self.getOrDefault(self.device)
        iters = self.getOrDefault(self.iters)
        partitions = self.getOrDefault(self.partitions)
        verbose = self.getOrDefault(self.verbose)
        mini_batch = self.getOrDefault(self.miniBatch)
        validation_pct = self.getOrDefault(self.validationPct)
        early_stop = self.getOrDefault(self.earlyStopPatience)
        shuffles = self.getOrDefault(self.partitionShuffles)
        use_barrier = self.getOrDefault(self.useBarrier)
        compile_mode = self.getOrDefault(self.compileMode)

        rdd = dataset.rdd.mapPartitions(handle_data(inp, label))
        if partitions and partitions > 0:
            rdd = rdd.repartition(partitions)

        if mode == "synchronous":
            from sparktorch_amd.parallel.sync import train_distributed

            # sync mode always runs under barrier scheduling (reference
            # distributed.py:53-63 builds PipelinedRDD(isFromBarrier=True));
            # train_distributed applies rdd.barrier() per shuffle round so
            # repartition still happens on the plain RDD.
            state = train_distributed(
                rdd,
                torch_obj,
                iters=iters,
                partition_shuffles=shuffles,
                verbose=verbose,
                mini_batch=mini_batch,
                validation_pct=validation_pct,
                device=device,
                early_stop_patience=early_stop,
                compile_mode=compile_mode,
                backend=self.getOrDefault(self.backend),
                bucket_cap_mb=self.getOrDefault(self.bucketCapMb),
            )
        elif mode == "hogwild":
            from sparktorch_amd.parallel import hogwild
            from sparktorch_amd.parallel.server import Server, determine_master

            port = self.getOrDefault(self.port)
            # partition count BEFORE any barrier wrap: pyspark's RDDBarrier
            # exposes only mapPartitions* (no getNumPartitions)
            n_parts = rdd.getNumPartitions()
            if use_barrier:
                rdd = rdd.barrier() if hasattr(rdd, "barrier") else rdd
            master_url = _driver_url(dataset, port)
            server = Server(
                torch_obj,
                master_url=master_url,
                port=port,
                acquire_lock=self.getOrDefault(self.acquireLock),
                early_stop_patience=early_stop,
                window_len=n_parts,
            )
            server.start_server()
            _wait_for_server(master_url)
            state = hogwild.train(
                rdd,
                torch_obj,
                server,
                iters=iters,
                partition_shuffles=shuffles,
                verbose=verbose,
                mini_batch=mini_batch,
                validation_pct=validation_pct,
                device=device,
                early_stop_patience=early_stop,
            )
        else:
            raise ValueError("mode must be 'synchronous' or 'hogwild', got %r" % mode)

        loaded = load_torch_model(torch_obj, from_json=True)
        loaded.model.load_state_dict(state)
        mod_str = obj_to_b64(loaded.model.cpu())

        return SparkTorchModel(
            inputCol=inp,
            predictionCol=pred_col,
            modStr=mod_str,
            useVectorOut=self.getOrDefault(self.useVectorOut),
        )


def _driver_url(dataset, port: int) -> str:
    if _is_local_df(dataset):
        return "127.0.0.1:%d" % port
    if HAS_PYSPARK:
        try:
            from pyspark.sql import SparkSession

            host = SparkSession.getActiveSession().sparkContext.getConf().get("spark.driver.host")
            return "%s:%d" % (host, port)
        except Exception:
            pass
    from sparktorch_amd.parallel.server import determine_master

    return determine_master(port)


def _wait_for_server(master_url: str, timeout_s: float = 30.0) -> None:
    import time

    from sparktorch_amd.parallel.hogwild import get_main

    deadline = time.time() + timeout_s
    while True:
        try:
            get_main(master_url)
            return
        except Exception:
            if time.time() > deadline:
                raise RuntimeError("parameter server did not come up at %s" % master_url)
            time.sleep(0.25)
