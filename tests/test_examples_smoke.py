"""Examples stay runnable (subprocess smoke; the heavier examples are
exercised implicitly through the engine test suites)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_spark_pipeline_example_runs_on_vendored_pyspark(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.pathsep.join([os.path.join(REPO, "vendor"), REPO])
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "spark_pipeline_dnn.py")],
        capture_output=True,
        text=True,
        timeout=420,
        env=env,
        cwd=str(tmp_path),  # the example saves its pipeline into cwd
    )
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "Train accuracy" in proc.stdout
    assert (tmp_path / "spark_pipeline_dnn_saved" / "metadata" / "part-00000").exists()
