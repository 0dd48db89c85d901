"""Guard the bench.py driver contract: one JSON line on stdout with the
agreed keys, runnable with no flags on CPU, all three modes intact."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


def _run(*args):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), *args],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    return json.loads(line)


def test_train_contract_cpu():
    d = _run("--steps", "2", "--warmup", "1", "--batch", "256")
    for k in REQUIRED:
        assert k in d, k
    assert d["metric"] == "samples_per_sec"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["config"]["global_batch"] == 256
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["model"].startswith("mnist_mlp")


def test_infer_contract_cpu():
    d = _run("--mode", "infer", "--steps", "2", "--warmup", "1", "--batch", "128")
    assert d["metric"] == "inference_samples_per_sec"
    assert d["value"] > 0


def test_time_to_loss_contract_cpu():
    d = _run("--mode", "time_to_loss", "--steps", "2", "--warmup", "1",
             "--batch", "256", "--target-loss", "5.0", "--max-steps", "3")
    assert d["metric"] == "time_to_loss_s"
    assert d["higher_is_better"] is False
    assert d["scaling"] == "strong"
    assert "reached" in d["config"]


def test_graft_entry_contract():
    """__graft_entry__ must expose build() and smoke() (driver contract);
    build() must be idempotent-cheap when the extension is current."""
    sys.path.insert(0, REPO)
    try:
        import __graft_entry__ as g
    finally:
        sys.path.pop(0)
    assert callable(g.build) and callable(g.smoke)
    g.build()  # no-op rebuild when sources unchanged; must not raise on CPU
    from sparktorch_amd import ops
    assert ops.available()


def test_fit_contract_cpu():
    """--mode fit: estimator-path metric with ingest inside the timed region."""
    out = _run("--mode", "fit", "--steps", "2", "--warmup", "1", "--batch", "512")
    for k in REQUIRED:
        assert k in out, k
    assert out["metric"] == "fit_samples_per_sec"
    assert out["higher_is_better"] is True
    assert out["config"]["ingest_included_in_timed_region"] is True
    assert out["config"]["ingest_ms"] > 0
    assert out["value"] > 0
