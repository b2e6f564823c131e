"""gRPC star-topology transport for multi-node / cross-silo parity
(SURVEY §5.8 / K17 "optional gRPC path": the reference's entire transport is
flwr's localhost-gRPC client-server star — `fl.server.start_server` +
`fl.client.start_client`).

Single-node multi-GPU deployments should use the RCCL runtime
(`parallel/distributed.py`); this transport exists for deployments the
collectives cannot reach: clients on other machines, behind NAT, or joining
late. Protocol (client-dials-server, like flwr):

- ``Join``:  client registers, receives its cid
- ``Pull``:  client long-polls for its next instruction (op, payload)
- ``Push``:  client returns the result for an instruction

Payloads are pickled op-dicts with CPU tensors (grpcio is in the image;
grpcio-tools/protoc is not, so the service is built from generic bytes
handlers — no .proto compilation step).
"""
from __future__ import annotations

import logging
import pickle
import queue
import threading
import time
import uuid
from typing import Any

import grpc
import torch

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import (
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    GetParametersIns,
    GetParametersRes,
    GetPropertiesIns,
    GetPropertiesRes,
    Parameters,
)

log = logging.getLogger(__name__)

SERVICE = "fl4health.Transport"
_GRPC_OPTS = [
    ("grpc.max_send_message_length", -1),
    ("grpc.max_receive_message_length", -1),
]


def _to_wire(obj: Any) -> bytes:
    return pickle.dumps(_tensors_to_cpu(obj))


def _from_wire(data: bytes) -> Any:
    return pickle.loads(data)


def _tensors_to_cpu(obj: Any) -> Any:
    if isinstance(obj, torch.Tensor):
        return obj.detach().cpu()
    if isinstance(obj, Parameters):
        return Parameters([t.detach().cpu() for t in obj.tensors], dict(obj.meta))
    if isinstance(obj, dict):
        return {k: _tensors_to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        out = [_tensors_to_cpu(v) for v in obj]
        return type(obj)(out) if isinstance(obj, list) else tuple(out)
    return obj


# ---------------------------------------------------------------------------
# server side
# ---------------------------------------------------------------------------


class GrpcServerTransport:
    """Transport implementation backed by per-client instruction/result
    queues, served over gRPC. Register this as ``server.transport`` and its
    proxies into the client manager via ``wait_for_clients``."""

    def __init__(self, address: str = "0.0.0.0:8080", accept_failures: bool = True,
                 max_workers: int = 32) -> None:
        self.accept_failures = accept_failures
        self._clients: dict[str, dict[str, Any]] = {}
        self._lock = threading.Lock()
        self._next_cid = 0
        from concurrent import futures

        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers), options=_GRPC_OPTS)
        handlers = {
            "Join": grpc.unary_unary_rpc_method_handler(self._join),
            "Pull": grpc.unary_unary_rpc_method_handler(self._pull),
            "Push": grpc.unary_unary_rpc_method_handler(self._push),
        }
        self._server.add_generic_rpc_handlers((grpc.method_handlers_generic_handler(SERVICE, handlers),))
        self._port = self._server.add_insecure_port(address)
        self._server.start()

    @property
    def port(self) -> int:
        return self._port

    # ---- rpc handlers (bytes in, bytes out) ---------------------------
    def _join(self, request: bytes, context) -> bytes:
        with self._lock:
            cid = str(self._next_cid)
            self._next_cid += 1
            self._clients[cid] = {"instructions": queue.Queue(), "results": {}, "event": threading.Event()}
        log.info("grpc client joined as cid %s", cid)
        return _to_wire({"cid": cid})

    def _pull(self, request: bytes, context) -> bytes:
        cid = _from_wire(request)["cid"]
        q = self._clients[cid]["instructions"]
        try:
            item = q.get(timeout=300.0)
        except queue.Empty:
            item = {"op": "noop", "op_id": ""}
        return _to_wire(item)

    def _push(self, request: bytes, context) -> bytes:
        msg = _from_wire(request)
        entry = self._clients[msg["cid"]]
        entry["results"][msg["op_id"]] = msg
        entry["event"].set()
        return _to_wire({"ok": True})

    # ---- round-trip ---------------------------------------------------
    def _call(self, cid: str, op: str, payload: dict, timeout: float | None) -> dict:
        op_id = uuid.uuid4().hex
        entry = self._clients[cid]
        entry["event"].clear()
        entry["instructions"].put({"op": op, "op_id": op_id, **payload})
        deadline = time.monotonic() + (timeout or 600.0)
        while op_id not in entry["results"]:
            if not entry["event"].wait(timeout=max(0.0, deadline - time.monotonic())):
                raise TimeoutError(f"client {cid} did not answer {op} in time")
            entry["event"].clear()
        res = entry["results"].pop(op_id)
        if "error" in res:
            raise RuntimeError(res["error"])
        return res

    def _fan_out(self, calls: list[tuple[ClientProxy, str, dict]], timeout: float | None):
        """Concurrent RPC fan-out (the reference's ThreadPoolExecutor pattern,
        servers/polling.py:63-98)."""
        from concurrent import futures

        results, failures = [], []
        with futures.ThreadPoolExecutor(max_workers=max(len(calls), 1)) as pool:
            futs = {pool.submit(self._call, p.cid, op, payload, timeout): p for p, op, payload in calls}
            for fut, proxy in futs.items():
                try:
                    results.append((proxy, fut.result()))
                except Exception as e:  # noqa: BLE001 — per-client failure policy
                    log.exception("grpc client %s call failed (accept_failures=%s)",
                                  proxy.cid, self.accept_failures)
                    if not self.accept_failures:
                        raise
                    failures.append(e)
        return results, failures

    # ---- Transport interface ------------------------------------------
    def did_collective_aggregate(self) -> bool:
        return False

    def collective_result(self):
        raise RuntimeError("grpc transport performs no collective aggregation")

    def fit_clients(self, instructions, strategy, timeout: float | None = None):
        calls = [(p, "fit", {"parameters": ins.parameters, "config": ins.config}) for p, ins in instructions]
        raw, failures = self._fan_out(calls, timeout)
        results = [
            (p, FitRes(parameters=r["parameters"], num_examples=r["num_examples"], metrics=r["metrics"]))
            for p, r in raw
        ]
        return results, failures

    def evaluate_clients(self, instructions, timeout: float | None = None):
        calls = [(p, "evaluate", {"parameters": ins.parameters, "config": ins.config}) for p, ins in instructions]
        raw, failures = self._fan_out(calls, timeout)
        results = [
            (p, EvaluateRes(loss=r["loss"], num_examples=r["num_examples"], metrics=r["metrics"]))
            for p, r in raw
        ]
        return results, failures

    def poll_clients(self, instructions, timeout: float | None = None):
        calls = [(p, "get_properties", {"config": ins.config}) for p, ins in instructions]
        raw, _failures = self._fan_out(calls, timeout)
        return [(p, GetPropertiesRes(properties=r["properties"])) for p, r in raw]

    def get_parameters(self, cid: str, config: dict, timeout: float | None = None) -> Parameters:
        return self._call(cid, "get_parameters", {"config": config}, timeout)["parameters"]

    def wait_for_clients(self, n: int, timeout: float = 120.0) -> list["GrpcClientProxy"]:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            with self._lock:
                if len(self._clients) >= n:
                    return [GrpcClientProxy(cid, self) for cid in sorted(self._clients, key=int)]
            time.sleep(0.05)
        raise TimeoutError(f"only {len(self._clients)} of {n} clients joined")

    def shutdown_clients(self) -> None:
        with self._lock:
            for cid, entry in self._clients.items():
                entry["instructions"].put({"op": "shutdown", "op_id": uuid.uuid4().hex})

    def stop(self) -> None:
        self._server.stop(grace=1.0)


class GrpcClientProxy(ClientProxy):
    def __init__(self, cid: str, transport: GrpcServerTransport) -> None:
        super().__init__(cid)
        self.transport = transport

    def get_properties(self, ins: GetPropertiesIns, timeout: float | None = None) -> GetPropertiesRes:
        r = self.transport._call(self.cid, "get_properties", {"config": ins.config}, timeout)
        return GetPropertiesRes(properties=r["properties"])

    def get_parameters(self, ins: GetParametersIns, timeout: float | None = None) -> GetParametersRes:
        return GetParametersRes(parameters=self.transport.get_parameters(self.cid, ins.config, timeout))

    def fit(self, ins: FitIns, timeout: float | None = None) -> FitRes:
        r = self.transport._call(self.cid, "fit", {"parameters": ins.parameters, "config": ins.config}, timeout)
        return FitRes(parameters=r["parameters"], num_examples=r["num_examples"], metrics=r["metrics"])

    def evaluate(self, ins: EvaluateIns, timeout: float | None = None) -> EvaluateRes:
        r = self.transport._call(self.cid, "evaluate", {"parameters": ins.parameters, "config": ins.config}, timeout)
        return EvaluateRes(loss=r["loss"], num_examples=r["num_examples"], metrics=r["metrics"])


# ---------------------------------------------------------------------------
# client side
# ---------------------------------------------------------------------------


def start_grpc_client(client: Any, server_address: str) -> None:
    """Client main loop (the reference's ``fl.client.start_client``): join,
    long-poll for instructions, execute on the local client, push results,
    exit on shutdown."""
    channel = grpc.insecure_channel(server_address, options=_GRPC_OPTS)
    call = channel.unary_unary  # generic bytes-in/bytes-out stubs

    def rpc(method: str, payload: Any) -> Any:
        fn = call(f"/{SERVICE}/{method}", request_serializer=None, response_deserializer=None)
        return _from_wire(fn(_to_wire(payload)))

    cid = rpc("Join", {})["cid"]
    log.info("joined as cid %s", cid)
    while True:
        ins = rpc("Pull", {"cid": cid})
        op = ins.get("op")
        if op == "noop":
            continue
        if op == "shutdown":
            client.shutdown()
            channel.close()
            return
        out: dict[str, Any] = {"cid": cid, "op_id": ins["op_id"]}
        try:
            if op == "fit":
                params, n, metrics = client.fit(ins["parameters"], ins["config"])
                out.update({"parameters": params, "num_examples": n, "metrics": metrics})
            elif op == "evaluate":
                loss, n, metrics = client.evaluate(ins["parameters"], ins["config"])
                out.update({"loss": loss, "num_examples": n, "metrics": metrics})
            elif op == "get_properties":
                out["properties"] = client.get_properties(ins["config"])
            elif op == "get_parameters":
                out["parameters"] = client.get_parameters(ins["config"])
            else:
                out["error"] = f"unknown op {op}"
        except Exception as e:  # noqa: BLE001 — report to the server, keep serving
            log.exception("client op %s failed", op)
            out["error"] = repr(e)
        rpc("Push", out)


def start_grpc_server(server, address: str, n_clients: int, num_rounds: int,
                      join_timeout: float = 120.0):
    """Server main (the reference's ``fl.server.start_server``): bind, wait
    for the cohort, run the round loop, shut clients down."""
    transport = GrpcServerTransport(address, accept_failures=server.accept_failures)
    server.transport = transport
    for proxy in transport.wait_for_clients(n_clients, timeout=join_timeout):
        server.client_manager.register(proxy)
    try:
        history, _elapsed = server.fit(num_rounds)
    finally:
        transport.shutdown_clients()
        time.sleep(0.2)  # let shutdown instructions drain
        transport.stop()
        server.shutdown()
    return history
