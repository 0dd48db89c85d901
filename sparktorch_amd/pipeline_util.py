"""Pipeline persistence — the checkpoint format.

Parity with reference sparktorch/pipeline_util.py:27-131: a fitted sparktorch
stage is persisted as ``zlib(dill(stage))`` encoded as a comma-joined decimal
byte string, carried in the ``stopWords`` of a ``StopWordsRemover`` with the
magic GUID ``4c1740b00d3c4ff6806a1402321572cb`` appended
(:mod:`sparktorch_amd.utils.codec` implements the byte format).

Two carriers share that format:

* With pyspark: a JVM ``StopWordsRemover`` inside the standard Spark ML
  pipeline save (same mechanism as the reference; ``PysparkReaderWriter``
  mixin + ``PysparkPipelineWrapper.unwrap``).
* Without pyspark (this sandbox): :class:`LocalPipeline` /
  :class:`LocalPipelineModel` write the identical stopWords payload into a
  JSON file per stage, so save -> load -> unwrap round-trips with the same
  encoding.
"""

from __future__ import annotations

import json
import os
from typing import List

from sparktorch_amd.compat.params import HAS_PYSPARK
from sparktorch_amd.utils.codec import (
    MAGIC_GUID,
    is_sparktorch_stopwords,
    obj_to_stopwords,
    stopwords_to_obj,
)


class PysparkReaderWriter:
    """Mixin giving sparktorch stages Spark-ML save/load via the carrier
    format (reference pipeline_util.py:81-131)."""

    def write(self):
        if HAS_PYSPARK:
            from pyspark.ml.util import JavaMLWriter

            return JavaMLWriter(self._to_carrier())
        return _LocalStageWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def read(cls):
        if HAS_PYSPARK:
            from pyspark.ml.feature import StopWordsRemover
            from pyspark.ml.util import JavaMLReader

            return JavaMLReader(StopWordsRemover)
        return _LocalStageReader()

    @classmethod
    def load(cls, path: str):
        loaded = cls.read().load(path)
        return PysparkPipelineWrapper.unwrap(loaded)

    def _to_carrier(self):
        from pyspark.ml.feature import StopWordsRemover

        return StopWordsRemover(stopWords=obj_to_stopwords(self))

    # pyspark's Pipeline persistence converts python stages to their JVM
    # counterpart via _to_java (reference pipeline_util.py:113-131); aliasing
    # the carrier builder makes sparktorch stages saveable inside a plain
    # pyspark Pipeline/PipelineModel.
    _to_java = _to_carrier


class _LocalStageWriter:
    def __init__(self, stage):
        self._stage = stage
        self._overwrite = False

    def overwrite(self):
        self._overwrite = True
        return self

    def save(self, path: str) -> None:
        if os.path.exists(path) and not self._overwrite:
            raise IOError("path %s already exists (use .overwrite())" % path)
        os.makedirs(path, exist_ok=True)
        payload = obj_to_stopwords(self._stage)
        with open(os.path.join(path, "stage.json"), "w") as f:
            json.dump({"class": "StopWordsRemover", "stopWords": payload}, f)


class _CarrierStage:
    """Loaded-but-not-yet-unwrapped local carrier (mirrors the JVM
    StopWordsRemover the reference sees on load)."""

    def __init__(self, stop_words: List[str]):
        self._stop_words = stop_words

    def getStopWords(self) -> List[str]:
        return self._stop_words


class _LocalStageReader:
    def load(self, path: str):
        with open(os.path.join(path, "stage.json")) as f:
            d = json.load(f)
        return _CarrierStage(d["stopWords"])


class PysparkPipelineWrapper:
    """Walk a loaded pipeline, replacing GUID-marked carrier stages with the
    deserialized sparktorch stage (reference pipeline_util.py:58-78)."""

    @staticmethod
    def unwrap(pipeline):
        if isinstance(pipeline, _CarrierStage):
            return stopwords_to_obj(pipeline.getStopWords())
        if isinstance(pipeline, (LocalPipeline, LocalPipelineModel)):
            stages = [PysparkPipelineWrapper.unwrap(s) for s in pipeline.stages]
            pipeline.stages = stages
            return pipeline

        if HAS_PYSPARK:
            from pyspark.ml import Pipeline, PipelineModel
            from pyspark.ml.feature import StopWordsRemover

            # a bare carrier loaded via PysparkReaderWriter.load (not nested
            # in a pipeline) decodes directly
            if isinstance(pipeline, StopWordsRemover) and is_sparktorch_stopwords(
                pipeline.getStopWords()
            ):
                return stopwords_to_obj(pipeline.getStopWords())

            if isinstance(pipeline, (Pipeline, PipelineModel)):
                stages = (
                    pipeline.getStages() if isinstance(pipeline, Pipeline) else pipeline.stages
                )
                out = []
                for stage in stages:
                    if isinstance(stage, (Pipeline, PipelineModel)):
                        out.append(PysparkPipelineWrapper.unwrap(stage))
                    elif isinstance(stage, StopWordsRemover) and is_sparktorch_stopwords(
                        stage.getStopWords()
                    ):
                        out.append(stopwords_to_obj(stage.getStopWords()))
                    else:
                        out.append(stage)
                if isinstance(pipeline, Pipeline):
                    pipeline.setStages(out)
                else:
                    pipeline.stages = out
                return pipeline
        return pipeline


# ----------------------------------------------------------------------------
# Local pipeline (no-JVM stand-in for pyspark.ml.Pipeline)
# ----------------------------------------------------------------------------


class LocalPipeline:
    def __init__(self, stages: List):
        self.stages = list(stages)

    def getStages(self):
        return self.stages

    def fit(self, df):
        fitted = []
        cur = df
        for stage in self.stages:
            if hasattr(stage, "fit"):
                model = stage.fit(cur)
                fitted.append(model)
                cur = model.transform(cur)
            else:
                fitted.append(stage)
                cur = stage.transform(cur)
        return LocalPipelineModel(fitted)


class LocalPipelineModel:
    def __init__(self, stages: List):
        self.stages = list(stages)

    def transform(self, df):
        cur = df
        for stage in self.stages:
            cur = stage.transform(cur)
        return cur

    def write(self):
        return _LocalPipelineWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        with open(os.path.join(path, "pipeline.json")) as f:
            meta = json.load(f)
        stages = []
        for i in range(meta["num_stages"]):
            sd = os.path.join(path, "stage_%d" % i)
            with open(os.path.join(sd, "stage.json")) as f:
                d = json.load(f)
            sw = d["stopWords"]
            obj = stopwords_to_obj(sw)
            # sparktorch stages stay wrapped until PysparkPipelineWrapper.unwrap
            # (same observable flow as the reference's JVM carrier on load);
            # ordinary stages come back ready to use.
            if isinstance(obj, PysparkReaderWriter):
                stages.append(_CarrierStage(sw))
            else:
                stages.append(obj)
        return cls(stages)


class _LocalPipelineWriter:
    def __init__(self, model: LocalPipelineModel):
        self._model = model
        self._overwrite = False

    def overwrite(self):
        self._overwrite = True
        return self

    def save(self, path: str) -> None:
        if os.path.exists(path) and not self._overwrite:
            raise IOError("path %s already exists (use .overwrite())" % path)
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "pipeline.json"), "w") as f:
            json.dump({"num_stages": len(self._model.stages)}, f)
        for i, stage in enumerate(self._model.stages):
            sd = os.path.join(path, "stage_%d" % i)
            os.makedirs(sd, exist_ok=True)
            with open(os.path.join(sd, "stage.json"), "w") as f:
                json.dump(
                    {"class": "StopWordsRemover", "stopWords": obj_to_stopwords(stage)}, f
                )
