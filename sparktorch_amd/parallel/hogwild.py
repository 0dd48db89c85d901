"""Hogwild (async parameter-server) worker engine.

Parity with reference sparktorch/hogwild.py:31-186, re-based on the binary
wire format: per iteration a worker pulls the full state_dict
(``GET /parameters``), runs a local forward/backward, pushes raw gradients
(``POST /update``), and optionally polls the loss/early-stop route.  All HTTP
helpers retry exactly once with a 10 s timeout, and the gradient push
swallows a second failure silently (stale-gradient semantics tolerate a lost
update) — reference hogwild.py:44-49.

On a GPU worker the pulled parameters are staged through pinned host tensors
so the H2D copies run async on a side stream (`hipMemcpyAsync` under torch's
``copy_(non_blocking=True)``), and gradients are gathered D2H the same way.
"""

from __future__ import annotations

import http.client
import json
import time
from typing import Iterable, List, Optional

import numpy as np
import torch

from sparktorch_amd.parallel.sync import compute_loss
from sparktorch_amd.parallel.wire import decode_state_dict, encode_tensors
from sparktorch_amd.utils.data import handle_features, handle_features_device
from sparktorch_amd.utils.trace import StepMetrics, trace_range
from sparktorch_amd.utils.serialize import load_torch_model


def _request(master_url: str, method: str, path: str, body: Optional[bytes] = None, timeout: float = 10.0):
    host, _, port = master_url.partition(":")
    last = None
    for _attempt in range(2):  # retry exactly once (reference hogwild.py:31-57)
        try:
            conn = http.client.HTTPConnection(host, int(port or 80), timeout=timeout)
            try:
                conn.request(method, path, body=body)
                resp = conn.getresponse()
                data = resp.read()
                if resp.status != 200:
                    raise RuntimeError("PS %s %s -> %d" % (method, path, resp.status))
                return data
            finally:
                conn.close()
        except Exception as e:
            last = e
            time.sleep(0.1)
    raise last  # type: ignore[misc]


def get_main(master_url: str) -> bytes:
    return _request(master_url, "GET", "/")


def get_state_dict(master_url: str) -> dict:
    return decode_state_dict(_request(master_url, "GET", "/parameters"))


def put_deltas_to_server(master_url: str, deltas: List[torch.Tensor]) -> None:
    try:
        _request(master_url, "POST", "/update", body=encode_tensors(deltas))
    except Exception:
        # silently tolerate a lost update (reference hogwild.py:44-49)
        pass


def put_early_stop(master_url: str, loss: float) -> dict:
    data = _request(
        master_url, "POST", "/losses", body=json.dumps({"loss": float(loss)}).encode("utf-8")
    )
    return json.loads(data.decode("utf-8"))


def handle_model(
    data,
    torch_obj: str,
    master_url: str,
    iters: int = 10,
    verbose: int = 0,
    mini_batch: int = -1,
    validation_pct: float = 0.0,
    device: str = "cpu",
    early_stop_patience: int = -1,
):
    """Per-partition hogwild worker loop (reference hogwild.py:65-142)."""
    if device.startswith("cuda") and not torch.cuda.is_available():
        raise RuntimeError("device=%r requested but no GPU is visible" % device)

    if device.startswith("cuda"):
        # pinned staging + device-side cast; fp32 activations (the hogwild
        # model runs eager fp32 — its per-iteration cost is the PS pull/push)
        import torch as _torch

        feats = handle_features_device(data, validation_pct, device=device,
                                       dtype=_torch.float32)
    else:
        feats = handle_features(data, validation_pct)
    if feats.x_train is None:
        return iter([])

    loaded = load_torch_model(torch_obj, from_json=torch_obj.lstrip().startswith("{"))
    model = loaded.model.to(device)
    criterion = loaded.criterion

    x_train = feats.x_train if feats.x_train.is_cuda else feats.x_train.to(device)
    y_train = (
        (feats.y_train if feats.y_train.is_cuda else feats.y_train.to(device))
        if feats.y_train is not None else x_train
    )
    x_val = (feats.x_val if feats.x_val.is_cuda else feats.x_val.to(device)) if feats.x_val is not None else None
    y_val = (
        (feats.y_val if feats.y_val.is_cuda else feats.y_val.to(device))
        if feats.y_val is not None else (x_val if x_val is not None else None)
    )

    pinned: Optional[dict] = None
    side_stream = None
    if device.startswith("cuda"):
        pinned = {k: torch.empty_like(v, device="cpu").pin_memory() for k, v in model.state_dict().items()}
        side_stream = torch.cuda.Stream()  # hipMemcpyAsync H2D off the compute stream

    n = x_train.shape[0]
    metrics = StepMetrics()
    for i in range(iters):
        metrics.start()
        sd = get_state_dict(master_url)
        if pinned is not None:
            # the previous iteration's async H2D must have drained before the
            # pinned buffers are rewritten — explicit, not implied by some
            # later blocking call on the compute stream
            side_stream.synchronize()
            for k, v in sd.items():
                pinned[k].copy_(v)
            # async H2D on a side stream; the compute stream waits on it
            with torch.cuda.stream(side_stream):
                sd = {k: v.to(device, non_blocking=True) for k, v in pinned.items()}
            torch.cuda.current_stream().wait_stream(side_stream)
        model.load_state_dict(sd)

        if 0 < mini_batch < n:
            idx = torch.from_numpy(np.random.choice(n, mini_batch, replace=False))
            xb, yb = x_train[idx], y_train[idx]
        else:
            xb, yb = x_train, y_train

        model.zero_grad(set_to_none=True)
        pred = model(xb)
        loss = compute_loss(criterion, pred, yb)
        loss.backward()

        grads = [
            (p.grad if p.grad is not None else torch.zeros_like(p)).detach()
            for p in model.parameters()
        ]
        with trace_range("hogwild_push"):
            put_deltas_to_server(master_url, grads)
        metrics.stop(float(loss.detach()))

        loss_for_es = loss
        if x_val is not None:
            with torch.no_grad():
                model.eval()
                loss_for_es = compute_loss(criterion, model(x_val), y_val)
                model.train()

        if early_stop_patience > 0:
            resp = put_early_stop(master_url, float(loss_for_es))
            if resp.get("stop"):
                break

        if verbose:
            print("hogwild iter %d loss %.6f" % (i, float(loss)), flush=True)

    if verbose:
        print("hogwild metrics %r" % metrics, flush=True)

    return iter([])


def train(
    rdd,
    torch_obj: str,
    server,
    iters: int = 10,
    partition_shuffles: int = 1,
    verbose: int = 0,
    mini_batch: int = -1,
    validation_pct: float = 0.0,
    device: str = "cpu",
    early_stop_patience: int = -1,
) -> dict:
    """Driver side of hogwild training (reference hogwild.py:145-186)."""
    master_url = server.master_url
    try:
        for shuffle_round in range(max(1, partition_shuffles)):
            def worker(iterator, _obj=torch_obj, _url=master_url):
                return handle_model(
                    iterator,
                    _obj,
                    _url,
                    iters=iters,
                    verbose=verbose,
                    mini_batch=mini_batch,
                    validation_pct=validation_pct,
                    device=device,
                    early_stop_patience=early_stop_patience,
                )

            rdd.mapPartitions(worker).foreach(lambda _x: None)
            if shuffle_round + 1 < partition_shuffles:
                # a barrier-wrapped rdd (pyspark RDDBarrier) exposes only
                # mapPartitions*; shuffle the underlying RDD and re-wrap
                base = getattr(rdd, "rdd", rdd)
                shuffled = base.repartition(base.getNumPartitions())
                rdd = shuffled.barrier() if base is not rdd else shuffled

        return get_state_dict(master_url)
    finally:
        server.stop_server()
