"""Spark ML API compatibility layer.

When pyspark is installed, this module re-exports the real Spark ML classes so
``SparkTorch`` is a genuine ``pyspark.ml.Estimator`` usable inside a Spark
``Pipeline``.  When pyspark is absent (as in this build sandbox — no network),
a behavior-compatible shim of the small surface the reference uses
(``Param``/``Params``/``keyword_only``/``TypeConverters``, the Has*Col mixins,
``Estimator``/``Model``/``Transformer``, ``Identifiable``) is provided, so the
whole framework runs and is testable without a JVM.

Reference usage being mirrored: sparktorch/torch_distributed.py:133-201
(Param declarations, _setDefault, keyword_only setParams).
"""

from __future__ import annotations

import uuid

try:  # exercised under vendor/pyspark (tests/test_pyspark_double.py) or real pyspark
    import pyspark  # noqa: F401

    HAS_PYSPARK = True
except ImportError:
    HAS_PYSPARK = False


if HAS_PYSPARK:  # exercised under vendor/pyspark or real pyspark
    from pyspark import keyword_only
    from pyspark.ml.base import Estimator, Model, Transformer
    from pyspark.ml.param import Param, Params, TypeConverters
    from pyspark.ml.param.shared import HasInputCol, HasLabelCol, HasPredictionCol
    from pyspark.ml.util import Identifiable, MLReadable, MLWritable
else:
    import functools

    def keyword_only(func):
        """Capture explicitly-passed kwargs in self._input_kwargs (pyspark semantics)."""

        @functools.wraps(func)
        def wrapper(self, *args, **kwargs):
            if len(args) > 0:
                raise TypeError("Method %s only takes keyword arguments." % func.__name__)
            self._input_kwargs = kwargs
            return func(self, **kwargs)

        return wrapper

    class TypeConverters:
        @staticmethod
        def toString(v):
            return str(v)

        @staticmethod
        def toInt(v):
            return int(v)

        @staticmethod
        def toFloat(v):
            return float(v)

        @staticmethod
        def toBoolean(v):
            return bool(v)

        @staticmethod
        def identity(v):
            return v

    class Param:
        def __init__(self, parent, name, doc="", typeConverter=None):
            self.parent = parent
            self.name = name
            self.doc = doc
            self.typeConverter = typeConverter

        def __repr__(self):
            return "Param(%s)" % self.name

    class Params:
        """Minimal param store: user map overrides default map."""

        @staticmethod
        def _dummy():
            return "undefined"

        def __init__(self):
            self._paramMap = {}
            self._defaultParamMap = {}
            self.uid = type(self).__name__ + "_" + uuid.uuid4().hex[:12]

        def _param_by_name(self, name):
            p = getattr(type(self), name, None)
            if not isinstance(p, Param):
                raise AttributeError("no param named %r on %s" % (name, type(self).__name__))
            return p

        def _resolve(self, param):
            return param if isinstance(param, Param) else self._param_by_name(param)

        def _set(self, **kwargs):
            for name, value in kwargs.items():
                p = self._param_by_name(name)
                if value is not None and p.typeConverter is not None:
                    value = p.typeConverter(value)
                self._paramMap[p.name] = value
            return self

        def _setDefault(self, **kwargs):
            for name, value in kwargs.items():
                p = self._param_by_name(name)
                self._defaultParamMap[p.name] = value
            return self

        def getOrDefault(self, param):
            p = self._resolve(param)
            if p.name in self._paramMap:
                return self._paramMap[p.name]
            return self._defaultParamMap.get(p.name)

        def isDefined(self, param):
            p = self._resolve(param)
            return p.name in self._paramMap or p.name in self._defaultParamMap

        def copy(self, extra=None):
            import copy as _copy

            other = _copy.copy(self)
            other._paramMap = dict(self._paramMap)
            other._defaultParamMap = dict(self._defaultParamMap)
            return other

    class Identifiable:
        pass

    class MLReadable:
        pass

    class MLWritable:
        pass

    class Transformer(Params):
        def transform(self, dataset):
            return self._transform(dataset)

    class Estimator(Params):
        def fit(self, dataset):
            return self._fit(dataset)

    class Model(Transformer):
        pass

    class _HasColMixin(Params):
        pass

    class HasInputCol(_HasColMixin):
        inputCol = Param(Params._dummy(), "inputCol", "input column name", TypeConverters.toString)

        def getInputCol(self):
            return self.getOrDefault(self.inputCol)

    class HasLabelCol(_HasColMixin):
        labelCol = Param(Params._dummy(), "labelCol", "label column name", TypeConverters.toString)

        def getLabelCol(self):
            return self.getOrDefault(self.labelCol)

    class HasPredictionCol(_HasColMixin):
        predictionCol = Param(
            Params._dummy(), "predictionCol", "prediction column name", TypeConverters.toString
        )

        def getPredictionCol(self):
            return self.getOrDefault(self.predictionCol)
