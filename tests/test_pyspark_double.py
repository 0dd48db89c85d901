"""Run the reference-parity scenario matrix against the vendored pyspark.

Each test launches tests/pyspark_scenarios.py in a subprocess with
``PYTHONPATH=vendor`` prepended, so ``import pyspark`` resolves to the double
and the framework's real-pyspark branches (compat/params.py HAS_PYSPARK,
torch_distributed mapPartitions/toDF/broadcast, pipeline_util
JavaMLWriter/JavaMLReader carriers, barrier scheduling through
pyspark.BarrierTaskContext) execute genuinely.  See the scenario module's
docstring for the reference test each scenario mirrors.
"""

from __future__ import annotations

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCENARIOS = os.path.join(REPO, "tests", "pyspark_scenarios.py")


def _run_group(group: str, timeout: float = 600.0) -> str:
    env = dict(os.environ)
    env["PYTHONPATH"] = os.pathsep.join(
        [os.path.join(REPO, "vendor"), REPO, env.get("PYTHONPATH", "")]
    ).rstrip(os.pathsep)
    proc = subprocess.run(
        [sys.executable, SCENARIOS, group],
        capture_output=True,
        text=True,
        timeout=timeout,
        env=env,
        cwd=REPO,
    )
    out = proc.stdout + "\n" + proc.stderr
    assert proc.returncode == 0, "scenario group %r failed:\n%s" % (group, out)
    assert "FAIL" not in proc.stdout, out
    return proc.stdout


def test_vendored_pyspark_importable():
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.join(REPO, "vendor")
    proc = subprocess.run(
        [sys.executable, "-c",
         "import pyspark; assert 'vendor' in pyspark.__file__; "
         "from pyspark import BarrierTaskContext, SparkContext, keyword_only; "
         "from pyspark.sql import SparkSession, Row; "
         "from pyspark.ml import Pipeline, PipelineModel; "
         "from pyspark.ml.feature import StopWordsRemover, VectorAssembler; "
         "from pyspark.ml.linalg import Vectors, VectorUDT; "
         "from pyspark.ml.util import JavaMLReader, JavaMLWriter; "
         "print('ok')"],
        capture_output=True,
        text=True,
        timeout=120,
        env=env,
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr


def test_scenarios_core():
    out = _run_group("core")
    assert out.count("PASS") == 6


def test_scenarios_modes():
    out = _run_group("modes")
    assert out.count("PASS") == 7


def test_scenarios_pipeline():
    out = _run_group("pipeline")
    assert out.count("PASS") == 4


def test_scenarios_hogwild():
    out = _run_group("hogwild")
    assert out.count("PASS") == 3


def test_scenarios_unit():
    out = _run_group("unit")
    assert out.count("PASS") == 5
