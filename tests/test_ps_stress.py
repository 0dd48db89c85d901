"""Parameter-server concurrency stress (CPU).

The reference PS (reference server.py:95-149) serves concurrent Flask
threads: readers pull the full state_dict while writers push gradients and
``optimizer.step()`` runs — with ``acquireLock`` either serializing updates
through the RWLock or leaving them genuinely lock-free (hogwild).  This
hammers our HTTP PS with the same mixed load from many threads and checks
it neither deadlocks, corrupts the wire, nor loses the model.
"""

import threading
import time
import urllib.request

import pytest
import torch
import torch.nn as nn

from sparktorch_amd.compat.local import free_port
from sparktorch_amd.models.simple_net import Net
from sparktorch_amd.parallel.hogwild import (
    get_main,
    get_state_dict,
    put_deltas_to_server,
    put_early_stop,
)
from sparktorch_amd.parallel.server import Server
from sparktorch_amd.utils.serialize import serialize_torch_obj


def _wait_up(base: str) -> None:
    for _ in range(150):
        try:
            with urllib.request.urlopen(base + "/", timeout=2):
                return
        except Exception:
            time.sleep(0.1)
    raise TimeoutError("PS never came up at " + base)


@pytest.mark.parametrize("acquire_lock", [True, False])
def test_ps_concurrent_pull_push(acquire_lock):
    obj = serialize_torch_obj(Net(), nn.MSELoss(), torch.optim.Adam, lr=0.01)
    port = free_port()
    srv = Server(obj, port=port, acquire_lock=acquire_lock,
                 early_stop_patience=-1, window_len=4)
    srv.start_server()
    base = "http://127.0.0.1:%d" % port
    ps = "127.0.0.1:%d" % port  # hogwild helpers take host:port (no scheme)
    errors: list = []
    try:
        _wait_up(base)
        sd0 = get_state_dict(ps)
        shapes = [v.shape for v in sd0.values()]
        n_workers, n_iters = 6, 15
        start = threading.Barrier(n_workers)

        def worker(seed: int):
            try:
                start.wait(timeout=30)
                g = torch.Generator().manual_seed(seed)
                for _ in range(n_iters):
                    sd = get_state_dict(ps)
                    assert list(v.shape for v in sd.values()) == shapes
                    assert all(torch.isfinite(v).all() for v in sd.values())
                    deltas = [torch.randn(s, generator=g) * 1e-3 for s in shapes]
                    put_deltas_to_server(ps, deltas)
            except Exception as e:  # pragma: no cover - surfaced below
                errors.append(e)

        threads = [threading.Thread(target=worker, args=(i,), daemon=True)
                   for i in range(n_workers)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=120)
            assert not t.is_alive(), "PS worker thread hung (deadlock?)"
        assert not errors, errors

        # the model moved and is still finite + wire-decodable
        sd1 = get_state_dict(ps)
        moved = any(not torch.equal(sd0[k], sd1[k]) for k in sd0)
        assert moved
        assert all(torch.isfinite(v).all() for v in sd1.values())
    finally:
        srv.stop_server()


def test_ps_windowed_early_stop_under_concurrency():
    """/losses averages a window of #partitions losses (reference
    server.py:104-125): concurrent posts of a decreasing-then-flat series
    must eventually flip the stop flag exactly once, never crash."""
    obj = serialize_torch_obj(Net(), nn.MSELoss(), torch.optim.Adam, lr=0.01)
    port = free_port()
    srv = Server(obj, port=port, acquire_lock=True,
                 early_stop_patience=2, window_len=3)
    srv.start_server()
    base = "http://127.0.0.1:%d" % port
    ps = "127.0.0.1:%d" % port
    try:
        _wait_up(base)
        stop_seen = []

        def poster(losses):
            for lv in losses:
                r = put_early_stop(ps, lv)
                stop_seen.append(bool(r.get("stop")))

        flat = [1.0] * 30  # no improvement -> patience must trip
        threads = [threading.Thread(target=poster, args=(flat,), daemon=True)
                   for _ in range(3)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
            assert not t.is_alive()
        assert any(stop_seen), "early stop never tripped on a flat loss series"
    finally:
        srv.stop_server()


def test_http_helpers_retry_exactly_once():
    """The hogwild HTTP helpers retry exactly once (reference
    hogwild.py:31-57): one transient failure is absorbed, two are fatal —
    except put_deltas_to_server, which silently tolerates a lost update
    (reference hogwild.py:44-49)."""
    import json
    from http.server import BaseHTTPRequestHandler, HTTPServer

    from sparktorch_amd.compat.local import free_port

    calls = {"n": 0, "fail_first": 0}

    class Flaky(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def _go(self):
            calls["n"] += 1
            if calls["n"] <= calls["fail_first"]:
                self.send_response(503)
                self.end_headers()
                self.wfile.write(b"down")
                return
            self.send_response(200)
            self.end_headers()
            self.wfile.write(json.dumps({"stop": False}).encode())

        def do_GET(self):
            self._go()

        def do_POST(self):
            self.rfile.read(int(self.headers.get("Content-Length", 0) or 0))
            self._go()

    port = free_port()
    httpd = HTTPServer(("127.0.0.1", port), Flaky)
    t = threading.Thread(target=httpd.serve_forever, daemon=True)
    t.start()
    ps = "127.0.0.1:%d" % port
    try:
        # one transient failure -> absorbed by the single retry
        calls.update(n=0, fail_first=1)
        assert get_main(ps) is not None
        assert calls["n"] == 2

        # two failures -> raises (no third attempt)
        calls.update(n=0, fail_first=2)
        with pytest.raises(RuntimeError):
            get_main(ps)
        assert calls["n"] == 2

        # put_deltas tolerates total failure silently
        calls.update(n=0, fail_first=99)
        put_deltas_to_server(ps, [torch.zeros(2)])
        assert calls["n"] == 2
    finally:
        httpd.shutdown()
        httpd.server_close()


def test_barrier_fail_fast_on_dead_worker():
    """A barrier task that dies without reporting (OOM-kill shape) must fail
    the stage within seconds — not hang until the stage timeout.  Guards the
    dead-worker detection in vendor/pyspark/rdd.py:_collect_barrier."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["PYTHONPATH"] = os.pathsep.join(
        [os.path.join(repo, "vendor"), os.path.join(repo, "tests"), repo]
        + [p for p in env.get("PYTHONPATH", "").split(os.pathsep) if p]
    )
    code = (
        "import time\n"
        "from pyspark import SparkContext\n"
        "from barrier_crash_worker import crash_partition_one\n"
        "sc = SparkContext()\n"
        "rdd = sc.parallelize(list(range(30)), 3)\n"
        "t0 = time.time()\n"
        "try:\n"
        "    rdd.barrier().mapPartitionsWithIndex(crash_partition_one).collect(timeout_s=120)\n"
        "except Exception as e:\n"
        "    took = time.time() - t0\n"
        "    msg = str(e)\n"
        "    assert 'died' in msg or 'exit' in msg.lower(), msg\n"
        "    assert took < 60, took\n"
        "    print('FAILFAST_OK %.1fs' % took)\n"
        "else:\n"
        "    raise SystemExit('barrier stage did not fail')\n"
    )
    r = subprocess.run([sys.executable, "-c", code], env=env, cwd=repo,
                       capture_output=True, text=True, timeout=150)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "FAILFAST_OK" in r.stdout, r.stdout + r.stderr
