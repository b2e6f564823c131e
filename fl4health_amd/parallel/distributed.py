"""Distributed runtime: one FL client process per MI355X GPU over RCCL/xGMI.

This replaces the reference's Flower gRPC star topology (SURVEY §5.8,
examples/basic_example/server.py:76 / client.py:48) for single-node multi-GPU
deployment: rank 0 hosts the server logic (strategy, client manager,
reporting) alongside its own client; every round is issued as ONE batched
command to all ranks, parameters move as a single flat fp32 tensor broadcast,
and aggregation uses a pre-scaled all-reduce over xGMI when the strategy
supports it (sum of alpha_i * w_i == weighted FedAvg, K1 in SURVEY §2.13) —
no serialization, no gather, no host round-trip.

Backend: "nccl" (RCCL on ROCm) when CUDA devices are present, else "gloo"
(CPU tests, world_size>1 works in CI containers). Collective payloads are
concatenated into one comm buffer per round so RCCL's ring/bucket machinery
sees few, large messages (7 xGMI links per GPU are per-link bound; big
buffers let RCCL stripe them).
"""
from __future__ import annotations

import datetime
import logging
import os
from typing import Any, Callable

import torch
import torch.distributed as dist

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
from fl4health_amd.metrics.metric_aggregation import metric_aggregation
from fl4health_amd.utils.tracing import trace_range

log = logging.getLogger(__name__)


class RankClientProxy(ClientProxy):
    """Server-side handle for the client living on a given rank."""

    def __init__(self, cid: str, runtime: "DistributedRuntime") -> None:
        super().__init__(cid)
        self.runtime = runtime

    def get_properties(self, ins: GetPropertiesIns, timeout: float | None = None) -> GetPropertiesRes:
        res = self.runtime.poll_clients([(self, ins)], timeout)
        return res[0][1]

    def get_parameters(self, ins: GetParametersIns, timeout: float | None = None) -> GetParametersRes:
        return GetParametersRes(parameters=self.runtime.targeted_get_parameters(int(self.cid), ins.config))

    def fit(self, ins: FitIns, timeout: float | None = None) -> FitRes:
        raise RuntimeError("rank-backed clients are driven in batched rounds via the transport")

    def evaluate(self, ins: EvaluateIns, timeout: float | None = None) -> EvaluateRes:
        raise RuntimeError("rank-backed clients are driven in batched rounds via the transport")


class DistributedRuntime:
    """Both the rank-0 Transport implementation and the worker serve() loop."""

    def __init__(self, backend: str | None = None, timeout_s: float = 1800.0) -> None:
        if not dist.is_initialized():
            rank = int(os.environ.get("RANK", "0"))
            world = int(os.environ.get("WORLD_SIZE", "1"))
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29511")
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world, timeout=datetime.timedelta(seconds=timeout_s)
            )
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.backend = dist.get_backend()
        if str(self.backend) == "nccl":
            local_rank = int(os.environ.get("LOCAL_RANK", self.rank))
            torch.cuda.set_device(local_rank)
            self.comm_device: torch.device = torch.device("cuda", local_rank)
        else:
            self.comm_device = torch.device("cpu")
        self.local_client: Any = None
        self._collective_state: tuple[Parameters, dict] | None = None
        self._shutdown = False

    # ------------------------------------------------------------------
    # low-level helpers
    # ------------------------------------------------------------------
    def _bcast_obj(self, obj: Any = None, src: int = 0) -> Any:
        lst = [obj]
        dist.broadcast_object_list(lst, src=src)
        return lst[0]

    def _all_gather_obj(self, obj: Any) -> list[Any]:
        out: list[Any] = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def _bcast_parameters(self, params: Parameters | None, src: int = 0) -> Parameters:
        """Broadcast Parameters: one meta object + ONE concatenated tensor.

        The wire buffer is fp32, which is exact for every floating dtype up to
        fp32 but silently corrupts int64 indices/masks above 2^24 — so integer
        payloads ride in the header object instead of the fused buffer."""
        if self.rank == src:
            assert params is not None
            float_idx = [i for i, t in enumerate(params.tensors) if t.is_floating_point()]
            int_payload = {
                i: t.detach().cpu() for i, t in enumerate(params.tensors) if not t.is_floating_point()
            }
            numels = [int(params.tensors[i].numel()) for i in float_idx]
            shapes = [list(params.tensors[i].shape) for i in float_idx]
            dtypes = [str(params.tensors[i].dtype) for i in float_idx]
            header = {
                "n_tensors": len(params.tensors),
                "float_idx": float_idx,
                "numels": numels,
                "shapes": shapes,
                "dtypes": dtypes,
                "int_payload": int_payload,
                "meta": params.meta,
            }
            self._bcast_obj(header, src=src)
            buf = (
                torch.cat([params.tensors[i].reshape(-1).to(self.comm_device, torch.float32) for i in float_idx])
                if numels
                else torch.zeros(0, device=self.comm_device)
            )
            dist.broadcast(buf, src=src)
            return params
        header = self._bcast_obj(None, src=src)
        total = sum(header["numels"])
        buf = torch.empty(total, dtype=torch.float32, device=self.comm_device)
        dist.broadcast(buf, src=src)
        tensors: list[torch.Tensor | None] = [None] * header["n_tensors"]
        off = 0
        for i, n, shp, dt in zip(header["float_idx"], header["numels"], header["shapes"], header["dtypes"]):
            t = buf[off : off + n].view(shp).clone()
            want = getattr(torch, dt.replace("torch.", ""))
            tensors[i] = t.to(want) if want != torch.float32 else t
            off += n
        for i, t in header["int_payload"].items():
            tensors[i] = t.to(self.comm_device) if self.comm_device.type != "cpu" else t
        assert all(t is not None for t in tensors)
        return Parameters(tensors, header["meta"])

    # ------------------------------------------------------------------
    # transport interface (rank 0)
    # ------------------------------------------------------------------
    def did_collective_aggregate(self) -> bool:
        return self._collective_state is not None

    def collective_result(self) -> tuple[Parameters, dict]:
        assert self._collective_state is not None
        res = self._collective_state
        self._collective_state = None
        return res

    def fit_clients(self, instructions: list[tuple[ClientProxy, FitIns]], strategy, timeout: float | None = None):
        assert self.rank == 0
        cohort = sorted(int(p.cid) for p, _ in instructions)
        configs = {int(p.cid): ins.config for p, ins in instructions}
        collective = strategy.supports_collective_aggregation()
        self._bcast_obj({"op": "fit", "cohort": cohort, "configs": configs, "collective": collective})
        params = instructions[0][1].parameters
        self._last_broadcast_params = self._bcast_parameters(params, src=0)
        return self._fit_body(cohort, configs, collective, strategy)

    def evaluate_clients(self, instructions: list[tuple[ClientProxy, EvaluateIns]], timeout: float | None = None):
        assert self.rank == 0
        cohort = sorted(int(p.cid) for p, _ in instructions)
        configs = {int(p.cid): ins.config for p, ins in instructions}
        self._bcast_obj({"op": "evaluate", "cohort": cohort, "configs": configs})
        self._last_broadcast_params = self._bcast_parameters(instructions[0][1].parameters, src=0)
        gathered = self._evaluate_body(cohort, configs)
        results, failures = [], []
        proxies = {int(p.cid): p for p, _ in instructions}
        for cid, payload in enumerate(gathered):
            if payload is None or cid not in proxies:
                continue
            if isinstance(payload, dict) and "error" in payload:
                failures.append(RuntimeError(payload["error"]))
            else:
                loss, n, metrics = payload
                results.append((proxies[cid], EvaluateRes(loss=loss, num_examples=n, metrics=metrics)))
        return results, failures

    def poll_clients(self, instructions: list[tuple[ClientProxy, GetPropertiesIns]], timeout: float | None = None):
        assert self.rank == 0
        cids = [int(p.cid) for p, _ in instructions]
        config = instructions[0][1].config if instructions else {}
        self._bcast_obj({"op": "get_properties", "cids": cids, "config": config})
        gathered = self._properties_body(cids, config)
        out = []
        proxies = {int(p.cid): p for p, _ in instructions}
        for cid, props in enumerate(gathered):
            if props is not None and cid in proxies:
                out.append((proxies[cid], GetPropertiesRes(properties=props)))
        return out

    def targeted_get_parameters(self, cid: int, config: dict) -> Parameters:
        assert self.rank == 0
        self._bcast_obj({"op": "get_parameters", "cid": cid, "config": config})
        return self._get_parameters_body(cid, config)

    def shutdown_clients(self) -> None:
        if self.rank == 0:
            self._bcast_obj({"op": "shutdown"})

    # ------------------------------------------------------------------
    # symmetric round bodies (run on EVERY rank)
    # ------------------------------------------------------------------
    def _fit_body(self, cohort: list[int], configs: dict[int, dict], collective: bool, strategy=None):
        in_cohort = self.rank in cohort
        fit_res: FitRes | None = None
        error: str | None = None
        if in_cohort:
            try:
                params_local = self._last_broadcast_params
                p, n, metrics = self.local_client.fit(params_local, configs[self.rank])
                fit_res = FitRes(parameters=p, num_examples=n, metrics=metrics)
            except Exception as e:  # noqa: BLE001
                log.exception("client fit failed on rank %d", self.rank)
                error = repr(e)

        if collective:
            # 1) tiny metadata all-gather: counts, tensor layout, metrics
            info = None
            if fit_res is not None:
                info = {
                    "n": fit_res.num_examples,
                    "numels": [int(t.numel()) for t in fit_res.parameters.tensors],
                    "shapes": [list(t.shape) for t in fit_res.parameters.tensors],
                    "meta": fit_res.parameters.meta,
                    "metrics": fit_res.metrics,
                }
            elif error is not None:
                info = {"error": error}
            gathered = self._all_gather_obj(info)
            ok = {cid: g for cid, g in enumerate(gathered) if g is not None and "error" not in g}
            if not ok:
                if self.rank == 0:
                    self._collective_state = (None, {})
                return [], [RuntimeError(g["error"]) for g in gathered if g and "error" in g]
            total_examples = sum(g["n"] for g in ok.values())
            cohort_size = len(ok)
            layout = next(iter(ok.values()))
            # every participating rank evaluates this on the SAME gathered
            # metadata: a heterogeneous payload (client bug, or a strategy that
            # should have forced the gather path) fails loudly and consistently
            # on all ranks instead of hanging the size-mismatched all-reduce
            if any(g["numels"] != layout["numels"] for g in ok.values()):
                raise RuntimeError(
                    "collective aggregation requires homogeneous client payloads; "
                    f"got numels {[g['numels'] for g in ok.values()]} — use a strategy "
                    "with supports_collective_aggregation() == False for per-client layouts"
                )
            total_numel = sum(layout["numels"])
            # 2) pre-scaled all-reduce over ONE concatenated comm buffer
            if self.rank in ok and fit_res is not None and strategy is not None:
                scales = strategy.collective_scales(
                    fit_res.num_examples, total_examples, cohort_size, len(fit_res.parameters.tensors)
                )
                # pre-scale + concat: one comm buffer per round (K1 pre-scaling)
                buf = torch.cat(
                    [
                        (t.reshape(-1).to(self.comm_device, torch.float32) * s)
                        for t, s in zip(fit_res.parameters.tensors, scales)
                    ]
                )
            else:
                # non-participating (or failed) rank contributes zeros; strategy
                # object exists on every rank (replicated, state-consistent)
                if self.rank in ok and strategy is None:
                    raise RuntimeError("collective aggregation requires a replicated strategy object on every rank")
                buf = torch.zeros(total_numel, dtype=torch.float32, device=self.comm_device)
            with trace_range("fl_allreduce_aggregate"):
                dist.all_reduce(buf, op=dist.ReduceOp.SUM)
            if self.rank == 0:
                tensors = []
                off = 0
                for n_el, shp in zip(layout["numels"], layout["shapes"]):
                    tensors.append(buf[off : off + n_el].view(shp).clone())
                    off += n_el
                summed = Parameters(tensors, dict(layout["meta"]))
                totals = {
                    "total_examples": float(total_examples),
                    "cohort_size": float(cohort_size),
                    "world_size": float(self.world_size),
                }
                new_params = strategy.finalize_collective(summed, -1, totals)
                agg_metrics = strategy.fit_metrics_aggregation_fn(
                    [(g["n"], g["metrics"]) for g in ok.values()]
                ) if getattr(strategy, "fit_metrics_aggregation_fn", None) else {}
                self._collective_state = (new_params, agg_metrics)
                failures = [RuntimeError(g["error"]) for g in gathered if g and "error" in g]
                # results list retains (proxy-less) metadata for server bookkeeping
                return [(RankClientProxy(str(cid), self), FitRes(Parameters([]), g["n"], g["metrics"])) for cid, g in ok.items()], failures
            return [], []

        # generic gather path: full FitRes (tensors to CPU) to every rank via
        # all_gather_object; rank 0 hands results to strategy.aggregate_fit
        payload = None
        if fit_res is not None:
            payload = (
                [t.detach().cpu() for t in fit_res.parameters.tensors],
                fit_res.parameters.meta,
                fit_res.num_examples,
                fit_res.metrics,
            )
        elif error is not None:
            payload = {"error": error}
        gathered = self._all_gather_obj(payload)
        if self.rank != 0:
            return [], []
        results, failures = [], []
        for cid, g in enumerate(gathered):
            if g is None:
                continue
            if isinstance(g, dict):
                failures.append(RuntimeError(g["error"]))
            else:
                tensors, meta, n, metrics = g
                results.append(
                    (RankClientProxy(str(cid), self), FitRes(Parameters(tensors, meta), n, metrics))
                )
        return results, failures

    def _evaluate_body(self, cohort: list[int], configs: dict[int, dict]):
        payload = None
        if self.rank in cohort:
            try:
                loss, n, metrics = self.local_client.evaluate(self._last_broadcast_params, configs[self.rank])
                payload = (loss, n, metrics)
            except Exception as e:  # noqa: BLE001
                log.exception("client evaluate failed on rank %d", self.rank)
                payload = {"error": repr(e)}
        return self._all_gather_obj(payload)

    def _properties_body(self, cids: list[int], config: dict):
        props = None
        if self.rank in cids:
            props = self.local_client.get_properties(config)
        return self._all_gather_obj(props)

    def _get_parameters_body(self, cid: int, config: dict) -> Parameters:
        if self.rank == cid:
            params = self.local_client.get_parameters(config)
            return self._bcast_parameters(params, src=cid)
        return self._bcast_parameters(None, src=cid)

    # ------------------------------------------------------------------
    # worker loop (ranks != 0)
    # ------------------------------------------------------------------
    _last_broadcast_params: Parameters | None = None

    def serve(self, strategy=None) -> None:
        """Worker ranks: process batched commands until shutdown. ``strategy``
        is the replicated strategy object used for collective pre-scaling."""
        assert self.rank != 0
        while True:
            cmd = self._bcast_obj(None, src=0)
            op = cmd["op"]
            if op == "shutdown":
                return
            if op == "sync":
                self.sync()
            elif op == "elapsed_gather":
                self._all_gather_obj(self._elapsed)
            elif op == "mark":
                import time

                self._mark = time.perf_counter()
            elif op == "elapse":
                import time

                self._elapsed = time.perf_counter() - self._mark
            elif op == "fit":
                self._last_broadcast_params = self._bcast_parameters(None, src=0)
                self._fit_body(cmd["cohort"], cmd["configs"], cmd["collective"], strategy)
            elif op == "evaluate":
                self._last_broadcast_params = self._bcast_parameters(None, src=0)
                self._evaluate_body(cmd["cohort"], cmd["configs"])
            elif op == "get_properties":
                self._properties_body(cmd["cids"], cmd["config"])
            elif op == "get_parameters":
                self._get_parameters_body(cmd["cid"], cmd["config"])
            else:
                raise RuntimeError(f"unknown command {op}")

    # rank-0 wrapper so fit_clients sees the same broadcast params as workers
    def _record_broadcast(self, params: Parameters) -> None:
        self._last_broadcast_params = params

    # ------------------------------------------------------------------
    # benchmarking support: barrier+device-sync brackets and max-elapsed
    # ------------------------------------------------------------------
    _mark: float = 0.0
    _elapsed: float = 0.0

    def sync(self) -> None:
        if self.comm_device.type == "cuda":
            torch.cuda.synchronize(self.comm_device)
        dist.barrier()
        if self.comm_device.type == "cuda":
            torch.cuda.synchronize(self.comm_device)

    def bench_sync(self) -> None:
        """Rank 0: bring every rank to a barrier + device sync."""
        assert self.rank == 0
        self._bcast_obj({"op": "sync"})
        self.sync()

    def bench_mark(self) -> None:
        import time

        assert self.rank == 0
        self._bcast_obj({"op": "mark"})
        self._mark = time.perf_counter()

    def bench_elapsed_max(self) -> float:
        """Rank 0: max over ranks of time since bench_mark."""
        import time

        assert self.rank == 0
        self._bcast_obj({"op": "elapse"})
        self._elapsed = time.perf_counter() - self._mark
        self._bcast_obj({"op": "elapsed_gather"})
        gathered = self._all_gather_obj(self._elapsed)
        return max(float(g) for g in gathered)
