"""Asynchronous (hogwild) parameter server.

Re-design of the reference's Flask+dill PS (server.py:35-151) with the same
observable behaviors and a faster transport:

* stdlib ``ThreadingHTTPServer`` in its own process (no Flask dependency,
  no per-request JSON), binary tensor wire format
  (:mod:`sparktorch_amd.parallel.wire`) instead of dill — the reference
  re-pickles the full state_dict per worker-iteration;
* routes: ``GET /`` health, ``GET /parameters`` (state_dict pull),
  ``POST /update`` (gradient push + ``optimizer.step()``),
  ``POST /losses`` (windowed-average early stopping, window = #partitions,
  reference server.py:104-125);
* ``acquireLock=True`` serializes updates through the writer-priority
  :class:`~sparktorch_amd.utils.rw_lock.RWLock` (parameter reads take the
  READ lock — knowingly fixing the reference quirk where ``get_parameters``
  takes the write lock, server.py:97-98); ``acquireLock=False`` is genuine
  lock-free HOGWILD;
* tolerates up to 10 bad updates before surfacing errors
  (reference server.py:141-144).
"""

from __future__ import annotations

import json
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

import torch
import torch.multiprocessing as mp

from sparktorch_amd.parallel.wire import decode_tensors, encode_state_dict
from sparktorch_amd.utils.early_stopper import EarlyStopping
from sparktorch_amd.utils.rw_lock import RWLock
from sparktorch_amd.utils.serialize import load_torch_model


def determine_master(port: int) -> str:
    """Resolve this driver's address for workers (reference server.py:65-71)."""
    try:
        host = socket.gethostbyname(socket.gethostname())
    except Exception:
        host = "127.0.0.1"
    return "%s:%d" % (host, port)


class _PSState:
    def __init__(self, torch_obj: str, acquire_lock: bool, early_stop_patience: int, window_len: int):
        loaded = load_torch_model(torch_obj, from_json=torch_obj.lstrip().startswith("{"))
        self.model = loaded.model
        self.criterion = loaded.criterion
        self.optimizer = loaded.optimizer
        self.model.share_memory()
        self.lock: Optional[RWLock] = RWLock() if acquire_lock else None
        # aux state (error budget / loss window / stop flag) is mutated from
        # ThreadingHTTPServer handler threads regardless of acquireLock —
        # guard it with its own mutex so a read-modify-reset of the loss
        # window can't drop or double-count entries under concurrency
        self.aux_lock = threading.Lock()
        self.error_count = 0
        self.loss_window = []
        self.window_len = max(1, window_len)
        self.es = EarlyStopping(patience=early_stop_patience) if early_stop_patience > 0 else None
        self.should_stop = False


def _make_handler(state: _PSState):
    class Handler(BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, *args):  # silence, like reference server.py:31-32
            pass

        def _send(self, code: int, body: bytes, ctype="application/octet-stream"):
            self.send_response(code)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_GET(self):
            if self.path == "/":
                self._send(200, b"sparktorch_amd parameter server", "text/plain")
            elif self.path == "/parameters":
                if state.lock is not None:
                    state.lock.acquire_read()
                try:
                    body = encode_state_dict(self.server._ps_state.model.state_dict())
                finally:
                    if state.lock is not None:
                        state.lock.release()
                self._send(200, body)
            else:
                self._send(404, b"not found", "text/plain")

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            data = self.rfile.read(n)
            if self.path == "/update":
                try:
                    grads = decode_tensors(data)
                    if state.lock is not None:
                        state.lock.acquire_write()
                    try:
                        for p, g in zip(state.model.parameters(), grads):
                            p.grad = g.to(p.dtype)
                        state.optimizer.step()
                    finally:
                        if state.lock is not None:
                            state.lock.release()
                    self._send(200, b"ok", "text/plain")
                except Exception as e:  # error budget, reference server.py:141-144
                    with state.aux_lock:
                        state.error_count += 1
                        over = state.error_count > 10
                    if over:
                        self._send(500, str(e).encode(), "text/plain")
                    else:
                        self._send(200, b"tolerated", "text/plain")
            elif self.path == "/losses":
                d = json.loads(data.decode("utf-8"))
                with state.aux_lock:
                    stop = state.should_stop
                    if state.es is not None and not stop:
                        state.loss_window.append(float(d["loss"]))
                        if len(state.loss_window) >= state.window_len:
                            avg = sum(state.loss_window) / len(state.loss_window)
                            state.loss_window = []
                            if state.es.step(avg):
                                state.should_stop = True
                                stop = True
                self._send(200, json.dumps({"stop": bool(stop)}).encode(), "application/json")
            else:
                self._send(404, b"not found", "text/plain")

    return Handler


def _serve(torch_obj: str, port: int, acquire_lock: bool, early_stop_patience: int, window_len: int):
    state = _PSState(torch_obj, acquire_lock, early_stop_patience, window_len)
    httpd = ThreadingHTTPServer(("0.0.0.0", port), _make_handler(state))
    httpd._ps_state = state  # type: ignore[attr-defined]
    httpd.serve_forever()


class Server:
    """Driver-side handle: starts/stops the PS process
    (reference server.py:35-79)."""

    def __init__(
        self,
        torch_obj: str,
        master_url: Optional[str] = None,
        port: int = 3000,
        acquire_lock: bool = False,
        early_stop_patience: int = -1,
        window_len: int = 4,
    ):
        self.torch_obj = torch_obj
        self.port = port
        self.master_url = master_url or determine_master(port)
        self.acquire_lock = acquire_lock
        self.early_stop_patience = early_stop_patience
        self.window_len = window_len
        self.server: Optional[mp.Process] = None

    def start_server(self) -> None:
        ctx = mp.get_context("spawn")
        self.server = ctx.Process(
            target=_serve,
            args=(
                self.torch_obj,
                self.port,
                self.acquire_lock,
                self.early_stop_patience,
                self.window_len,
            ),
            daemon=True,
        )
        self.server.start()

    def stop_server(self) -> None:
        if self.server is not None and self.server.is_alive():
            self.server.terminate()
            self.server.join(timeout=5)
        self.server = None
