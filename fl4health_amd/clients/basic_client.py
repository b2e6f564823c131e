"""BasicClient — the client-side train/eval engine (one per MI355X GPU).

Capability map to reference fl4health/clients/basic_client.py:43-1321:
- config parsing (epochs XOR steps)                 <- process_config :253
- lazy setup_client with user hooks                 <- :929-975
- train_by_epochs / train_by_steps hot loops        <- :627/:699
- per-step hook points (update_before/after_step,
  update_before/after_train, update_before_epoch,
  transform_gradients)                              <- :1233-1302
- loss meters / metric managers / early stopping    <- :458-521, 867
- pre/post-aggregation checkpointing                <- :141, :348-352, :415
- per-round state save/load for preemption resume   <- :1304-1321
- val + optional test loader evaluation             <- :821-928

MI355X-native differences: parameters ride as one flat fp32 device tensor
(FullParameterExchanger / FlatParameterView), training may run under bf16
autocast with fp32 master weights, and heavy per-step math (proximal terms,
variate corrections...) happens in fused HIP kernels in the subclasses.
"""
from __future__ import annotations

import datetime
import logging
from pathlib import Path
from typing import Any, Iterator

import torch
import torch.nn as nn
from torch.optim import Optimizer
from torch.utils.data import DataLoader

from fl4health_amd.checkpointing.client_module import CheckpointMode, ClientCheckpointAndStateModule
from fl4health_amd.common import Config, Metrics, Scalar
from fl4health_amd.metrics.base_metrics import Metric
from fl4health_amd.metrics.metric_managers import MetricManager
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchanger, ParameterExchanger
from fl4health_amd.common import Parameters
from fl4health_amd.reporting.reports_manager import ReportsManager
from fl4health_amd.utils.config import narrow_dict_type
from fl4health_amd.utils.losses import EvaluationLosses, LossMeter, LossMeterType, TrainingLosses
from fl4health_amd.utils.random import generate_hash
from fl4health_amd.utils.tracing import trace_range

log = logging.getLogger(__name__)

TorchInputType = torch.Tensor | dict[str, torch.Tensor]
TorchTargetType = torch.Tensor | dict[str, torch.Tensor]
TorchPredType = dict[str, torch.Tensor]


class BasicClient:
    def __init__(
        self,
        data_path: str | Path = ".",
        metrics: list[Metric] | None = None,
        device: torch.device | str | None = None,
        loss_meter_type: LossMeterType = LossMeterType.AVERAGE,
        checkpoint_and_state_module: ClientCheckpointAndStateModule | None = None,
        reporters: list | None = None,
        progress_bar: bool = False,
        client_name: str | None = None,
    ) -> None:
        self.data_path = Path(data_path)
        self.metrics = metrics or []
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.client_name = client_name if client_name is not None else generate_hash()
        self.checkpoint_and_state_module = checkpoint_and_state_module or ClientCheckpointAndStateModule()
        self.progress_bar = progress_bar

        self.initialized = False
        self.model: nn.Module
        self.optimizers: dict[str, Optimizer]
        self.train_loader: DataLoader
        self.val_loader: DataLoader | None = None
        self.test_loader: DataLoader | None = None
        self.lr_schedulers: dict[str, Any] = {}
        self.criterion: nn.Module
        self.parameter_exchanger: ParameterExchanger

        self.train_loss_meter = LossMeter.for_type(loss_meter_type)
        self.val_loss_meter = LossMeter.for_type(loss_meter_type)
        self.test_loss_meter = LossMeter.for_type(loss_meter_type)
        self.train_metric_manager = MetricManager(self.metrics, "train")
        self.val_metric_manager = MetricManager(self.metrics, "val")
        self.test_metric_manager = MetricManager(self.metrics, "test")

        self.reports_manager = ReportsManager(reporters)
        self.reports_manager.initialize(id=self.client_name, name="client")

        # bf16 autocast for forward/backward (fp32 master weights + fp32
        # flat aggregation); set autocast_dtype=torch.bfloat16 to enable
        self.autocast_dtype: torch.dtype | None = None
        # hipGraph capture of the train step: the CIFAR-scale hot loop is
        # launch-bound (rocprof: ~10us avg kernel, thousands of dispatches per
        # round), so the whole zero_grad->forward->loss->backward->fused-step
        # sequence is captured once and replayed per batch. Requires static
        # batch shapes (drop_last) and flat-bound params (stable pointers).
        self.use_cuda_graph = False
        self._graph: torch.cuda.CUDAGraph | None = None
        self._graph_static: dict[str, Any] | None = None
        # persistent bf16 compute mirror for multi-dim params (autocast weight
        # casts disappear; fused optimizer consumes bf16 grads + refreshes the
        # mirror in one pass). Requires autocast training + flat optimizers.
        self.use_bf16_mirror = False

        self.total_steps = 0
        self.total_epochs = 0
        self.current_server_round = 0
        self.num_train_samples = 0
        self.num_val_samples = 0
        self.early_stopper = None

        self._train_iterator: Iterator | None = None

    # ------------------------------------------------------------------
    # user hooks (reference basic_client.py:1111-1201)
    # ------------------------------------------------------------------
    def get_model(self, config: Config) -> nn.Module:
        raise NotImplementedError("subclass or user must implement get_model")

    def get_data_loaders(self, config: Config) -> tuple[DataLoader, DataLoader | None]:
        raise NotImplementedError("subclass or user must implement get_data_loaders")

    def get_test_data_loader(self, config: Config) -> DataLoader | None:
        return None

    def get_optimizer(self, config: Config) -> Optimizer | dict[str, Optimizer]:
        raise NotImplementedError("subclass or user must implement get_optimizer")

    def get_criterion(self, config: Config) -> nn.Module:
        raise NotImplementedError("subclass or user must implement get_criterion")

    def get_lr_scheduler(self, optimizer_key: str, config: Config):
        return None

    def get_parameter_exchanger(self, config: Config) -> ParameterExchanger:
        return FullParameterExchanger()

    # ------------------------------------------------------------------
    # setup / config
    # ------------------------------------------------------------------
    def setup_client(self, config: Config) -> None:
        """Lazy one-time initialization (reference :929-975)."""
        model = self.get_model(config)
        self.model = model.to(self.device)
        # flat substrate: params-first fp32 buffer, module params bound as views
        # (zero-copy exchange + fused flat optimizer kernels)
        from fl4health_amd.parameter_exchange.flat import FlatParameterView

        self.flat_view = FlatParameterView(self.model, bind=True)
        if self.use_bf16_mirror and self.device.type == "cuda":
            self.flat_view.enable_bf16_mirror()
        train_loader, val_loader = self.get_data_loaders(config)
        self.train_loader = train_loader
        self.val_loader = val_loader
        self.test_loader = self.get_test_data_loader(config)
        self.set_optimizer(config)
        for key in self.optimizers:
            sched = self.get_lr_scheduler(key, config)
            if sched is not None:
                self.lr_schedulers[key] = sched
        self.criterion = self.get_criterion(config)
        self.parameter_exchanger = self.get_parameter_exchanger(config)
        if isinstance(self.parameter_exchanger, FullParameterExchanger):
            self.parameter_exchanger._view = self.flat_view
        self.num_train_samples = len(getattr(self.train_loader, "dataset", [])) or 0
        if self.val_loader is not None:
            self.num_val_samples = len(getattr(self.val_loader, "dataset", [])) or 0
        self.initialized = True

    def set_optimizer(self, config: Config) -> None:
        opt = self.get_optimizer(config)
        self.optimizers = opt if isinstance(opt, dict) else {"global": opt}

    def maybe_setup_client(self, config: Config) -> None:
        if not self.initialized:
            self.setup_client(config)

    def process_config(self, config: Config) -> tuple[int | None, int | None, int, bool, bool]:
        """epochs XOR steps (reference :253-292)."""
        current_server_round = narrow_dict_type(config, "current_server_round", int)
        if ("local_epochs" in config) and ("local_steps" in config):
            raise ValueError("config has both local_epochs and local_steps: exactly one is allowed")
        local_epochs = config.get("local_epochs")
        local_steps = config.get("local_steps")
        if local_epochs is None and local_steps is None:
            raise ValueError("config needs one of local_epochs or local_steps")
        evaluate_after_fit = bool(config.get("evaluate_after_fit", False))
        pack_losses_with_val_metrics = bool(config.get("pack_losses_with_val_metrics", False))
        return (
            int(local_epochs) if local_epochs is not None else None,
            int(local_steps) if local_steps is not None else None,
            current_server_round,
            evaluate_after_fit,
            pack_losses_with_val_metrics,
        )

    # ------------------------------------------------------------------
    # parameter exchange
    # ------------------------------------------------------------------
    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            return self.setup_client_and_return_all_model_parameters(config)
        return self.parameter_exchanger.push_parameters(self.model, config=config)

    def setup_client_and_return_all_model_parameters(self, config: Config) -> Parameters:
        """Round-0 initialization handshake (reference :216-245)."""
        self.maybe_setup_client(config)
        return FullParameterExchanger().push_parameters(self.model, config=config)

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert self.initialized
        # round-0 initialization handshake ships FULL weights regardless of the
        # configured (possibly partial) exchanger (reference basic_client.py:179):
        # a payload without layer metadata whose size matches the full flat
        # buffer is loaded wholesale
        if (
            "layer_names" not in parameters.meta
            and len(parameters.tensors) >= 1
            and parameters.tensors[0].numel() == self.flat_view.spec.total
        ):
            self.flat_view.load_flat(parameters.tensors[0])
            return
        self.parameter_exchanger.pull_parameters(parameters, self.model, config)

    # ------------------------------------------------------------------
    # fit
    # ------------------------------------------------------------------
    def fit(self, parameters: Parameters, config: Config) -> tuple[Parameters, int, Metrics]:
        round_start = datetime.datetime.now()
        local_epochs, local_steps, current_server_round, evaluate_after_fit, pack_losses = self.process_config(config)
        self.current_server_round = current_server_round
        self.maybe_setup_client(config)
        if current_server_round == 1:
            self._maybe_load_client_state()
        self.set_parameters(parameters, config, fitting_round=True)
        self.update_before_train(current_server_round)

        if local_epochs is not None:
            loss_dict, metrics = self.train_by_epochs(local_epochs, current_server_round)
            local_steps = len(self.train_loader) * local_epochs
        else:
            assert local_steps is not None
            loss_dict, metrics = self.train_by_steps(local_steps, current_server_round)
        self.update_after_train(local_steps, loss_dict, config)

        if evaluate_after_fit:
            val_loss, val_metrics = self.validate()
            metrics.update(val_metrics)
            self._maybe_checkpoint(val_loss, val_metrics, CheckpointMode.PRE_AGGREGATION)
            if pack_losses:
                metrics["val - checkpoint"] = val_loss

        round_end = datetime.datetime.now()
        self.reports_manager.report(
            {
                "fit_losses": loss_dict,
                "round": current_server_round,
                "round_start": str(round_start),
                "round_end": str(round_end),
                "fit_metrics": metrics,
                "fit_round_time_elapsed": round((round_end - round_start).total_seconds()),
            },
            current_server_round,
        )
        self._save_client_state()
        return self.get_parameters(config), self.num_train_samples, metrics

    # ------------------------------------------------------------------
    # evaluate
    # ------------------------------------------------------------------
    def evaluate(self, parameters: Parameters, config: Config) -> tuple[float, int, Metrics]:
        start = datetime.datetime.now()
        current_server_round = narrow_dict_type(config, "current_server_round", int)
        self.current_server_round = current_server_round
        pack_losses = bool(config.get("pack_losses_with_val_metrics", False))
        self.maybe_setup_client(config)
        self.set_parameters(parameters, config, fitting_round=False)
        loss, metrics = self.validate(include_losses_in_metrics=pack_losses)
        self._maybe_checkpoint(loss, metrics, CheckpointMode.POST_AGGREGATION)
        end = datetime.datetime.now()
        self.reports_manager.report(
            {
                "eval_metrics": metrics,
                "eval_loss": loss,
                "eval_start": str(start),
                "eval_time_elapsed": round((end - start).total_seconds()),
                "round": current_server_round,
            },
            current_server_round,
        )
        return loss, self.num_val_samples, metrics

    # ------------------------------------------------------------------
    # properties (server polling)
    # ------------------------------------------------------------------
    def get_properties(self, config: Config) -> Config:
        """Return properties: sample counts after minimal setup (reference :200-214)."""
        self.maybe_setup_client(config)
        return {
            "num_train_samples": self.num_train_samples,
            "num_val_samples": self.num_val_samples,
        }

    # ------------------------------------------------------------------
    # training loops (the hot path)
    # ------------------------------------------------------------------
    def train_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        """forward -> loss -> backward -> transform_gradients -> step (reference :578-603)."""
        self.set_optimizer_zero_grad()
        losses, preds = self._forward_backward(input, target)
        self.transform_gradients(losses)
        self.step_optimizers()
        return losses, preds

    def _forward_backward(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        if self.autocast_dtype is not None and self.device.type == "cuda":
            with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
                preds, features = self.predict(input)
                target = self.transform_target(target)
                losses = self.compute_training_loss(preds, features, target)
        else:
            preds, features = self.predict(input)
            target = self.transform_target(target)
            losses = self.compute_training_loss(preds, features, target)
        losses.backward["backward"].backward()
        return losses, preds

    def set_optimizer_zero_grad(self) -> None:
        for opt in self.optimizers.values():
            opt.zero_grad(set_to_none=False)

    def step_optimizers(self) -> None:
        for opt in self.optimizers.values():
            opt.step()

    def _dispatch_train_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        """train_step, via hipGraph replay when enabled and shapes are static."""
        if not (
            self.use_cuda_graph
            and self.device.type == "cuda"
            and isinstance(input, torch.Tensor)
            and isinstance(target, torch.Tensor)
        ):
            return self.train_step(input, target)
        if self._graph is None:
            self._capture_train_graph(input, target)
        st = self._graph_static
        assert st is not None and self._graph is not None
        if st["input"].shape != input.shape:
            return self.train_step(input, target)  # ragged tail batch: eager
        st["input"].copy_(input, non_blocking=True)
        st["target"].copy_(target, non_blocking=True)
        self._graph.replay()
        return st["losses"], st["preds"]

    @staticmethod
    def _grad_as_1d(t: torch.Tensor) -> torch.Tensor:
        """Contiguous 1-D alias of a grad tensor. Flat-bound NHWC conv views
        (NCHW-logical permutes of a flat slice) and channels-last backward
        outputs share the same memory order, so both sides reduce to plain
        contiguous vectors and the batched copy stays on the foreach fast
        path."""
        if t.dim() == 4 and not t.is_contiguous() and t.is_contiguous(memory_format=torch.channels_last):
            return t.permute(0, 2, 3, 1).reshape(-1)
        return t.reshape(-1)

    def _graph_train_step(self, input: torch.Tensor, target: torch.Tensor):
        """train_step body used ONLY under hipGraph capture: steal-then-pack
        gradient flow. With flat-bound grads, eager steps pay zero_grad fill +
        AccumulateGrad add per parameter (2 elementwise kernels each, ~73
        small launches/step on ResNet-18). Inside a graph the backward's
        output buffers live at stable pool addresses, so we let AccumulateGrad
        steal them (p.grad=None first: no fill, no add) and pack all of them
        into the flat grad buffers with one batched foreach copy before the
        fused optimizer step."""
        views = [(p, p.grad) for p in self.model.parameters() if p.requires_grad and p.grad is not None]
        for p, _ in views:
            p.grad = None
        losses, preds = self._forward_backward(input, target)
        srcs, dsts = [], []
        for p, v in views:
            g = p.grad if p.grad is not None else torch.zeros_like(v)
            srcs.append(self._grad_as_1d(g))
            dsts.append(self._grad_as_1d(v))
            p.grad = v  # restore flat aliasing for transform/step/exchange
        # one foreach per dtype: a mixed-dtype list knocks _foreach_copy_ off
        # its fused multi-tensor kernel into per-pair hipMemcpy
        by_dtype: dict[torch.dtype, tuple[list, list]] = {}
        for d, s in zip(dsts, srcs):
            by_dtype.setdefault(d.dtype, ([], []))[0].append(d)
            by_dtype.setdefault(d.dtype, ([], []))[1].append(s.to(d.dtype) if s.dtype != d.dtype else s)
        for dl, sl in by_dtype.values():
            torch._foreach_copy_(dl, sl)
        self.transform_gradients(losses)
        self.step_optimizers()
        return losses, preds, srcs

    def _capture_train_graph(self, input: torch.Tensor, target: torch.Tensor) -> None:
        st: dict[str, Any] = {"input": input.clone(), "target": target.clone()}
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):  # warmup: MIOpen algo find + allocator steady-state
                self.train_step(st["input"], st["target"])
        torch.cuda.current_stream().wait_stream(side)
        # Subclasses with their own train_step keep their logic under capture;
        # only the stock body opts into the steal-then-pack grad flow.
        stock = type(self).train_step is BasicClient.train_step
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            if stock:
                losses, preds, stolen = self._graph_train_step(st["input"], st["target"])
                st["stolen"] = stolen  # keep pool buffers alive for replays
            else:
                losses, preds = self.train_step(st["input"], st["target"])
        st["losses"], st["preds"] = losses, preds
        self._graph, self._graph_static = graph, st
        log.info("Client %s: captured hipGraph train step (batch %s)", self.client_name, tuple(input.shape))

    def predict(self, input: TorchInputType) -> tuple[TorchPredType, dict[str, torch.Tensor]]:
        output = self.model(input) if not isinstance(input, dict) else self.model(**input)
        if isinstance(output, dict):
            return output, {}
        if isinstance(output, tuple) and len(output) == 2:
            preds, features = output
            return (preds if isinstance(preds, dict) else {"prediction": preds}), features
        return {"prediction": output}, {}

    def transform_target(self, target: TorchTargetType) -> TorchTargetType:
        return target

    def transform_gradients(self, losses: TrainingLosses) -> None:
        """Hook: gradient surgery before optimizer step (SCAFFOLD correction, clipping...)."""

    def compute_loss_and_additional_losses(
        self, preds: TorchPredType, features: dict[str, torch.Tensor], target: TorchTargetType
    ) -> tuple[torch.Tensor, dict[str, torch.Tensor] | None]:
        pred = preds["prediction"] if "prediction" in preds else next(iter(preds.values()))
        return self.criterion(pred, target), None

    def compute_training_loss(
        self, preds: TorchPredType, features: dict[str, torch.Tensor], target: TorchTargetType
    ) -> TrainingLosses:
        loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        return TrainingLosses(backward=loss, additional_losses=additional)

    def compute_evaluation_loss(
        self, preds: TorchPredType, features: dict[str, torch.Tensor], target: TorchTargetType
    ) -> EvaluationLosses:
        with torch.no_grad():
            loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        return EvaluationLosses(checkpoint=loss, additional_losses=additional)

    def _move_to_device(self, x):
        if isinstance(x, dict):
            return {k: v.to(self.device, non_blocking=True) for k, v in x.items()}
        return x.to(self.device, non_blocking=True)

    def train_by_epochs(self, epochs: int, current_round: int | None = None) -> tuple[dict[str, float], Metrics]:
        self.model.train()
        steps_this_round = 0
        report_data: dict[str, Any] = {"round": current_round}
        for local_epoch in range(epochs):
            self.train_metric_manager.clear()
            self.train_loss_meter.clear()
            self.update_before_epoch(local_epoch)
            report_data.update({"fit_epoch": local_epoch})
            for input, target in self.train_loader:
                self.update_before_step(steps_this_round, current_round)
                input, target = self._move_to_device(input), self._move_to_device(target)
                with trace_range("train_step"):
                    losses, preds = self._dispatch_train_step(input, target)
                self.train_loss_meter.update(losses)
                self.train_metric_manager.update(preds, target)
                self.update_after_step(steps_this_round, current_round)
                self.update_lr_schedulers(epoch=local_epoch)
                report_data.update({"fit_step": self.total_steps})
                self.reports_manager.report(report_data, current_round, local_epoch, self.total_steps)
                self.total_steps += 1
                steps_this_round += 1
                if self.early_stopper is not None and self.early_stopper.should_stop(self.total_steps):
                    self.early_stopper.load_snapshot()
                    metrics = self.train_metric_manager.compute()
                    loss_dict = self.train_loss_meter.compute()
                    return loss_dict, metrics
            metrics = self.train_metric_manager.compute()
            loss_dict = self.train_loss_meter.compute()
            self._log_results(loss_dict, metrics, current_round, local_epoch)
            self.total_epochs += 1
        return loss_dict, metrics

    def train_by_steps(self, steps: int, current_round: int | None = None) -> tuple[dict[str, float], Metrics]:
        self.model.train()
        if self._train_iterator is None:
            self._train_iterator = iter(self.train_loader)
        self.train_loss_meter.clear()
        self.train_metric_manager.clear()
        report_data: dict[str, Any] = {"round": current_round}
        for step in range(steps):
            self.update_before_step(step, current_round)
            try:
                input, target = next(self._train_iterator)
            except StopIteration:
                self._train_iterator = iter(self.train_loader)
                input, target = next(self._train_iterator)
            input, target = self._move_to_device(input), self._move_to_device(target)
            losses, preds = self._dispatch_train_step(input, target)
            self.train_loss_meter.update(losses)
            self.train_metric_manager.update(preds, target)
            self.update_after_step(step, current_round)
            self.update_lr_schedulers(step=step)
            report_data.update({"fit_step": self.total_steps})
            self.reports_manager.report(report_data, current_round, None, self.total_steps)
            self.total_steps += 1
            if self.early_stopper is not None and self.early_stopper.should_stop(self.total_steps):
                self.early_stopper.load_snapshot()
                break
        loss_dict = self.train_loss_meter.compute()
        metrics = self.train_metric_manager.compute()
        self._log_results(loss_dict, metrics, current_round)
        return loss_dict, metrics

    # ------------------------------------------------------------------
    # validation / test
    # ------------------------------------------------------------------
    def val_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[EvaluationLosses, TorchPredType]:
        with torch.no_grad():
            if self.autocast_dtype is not None and self.device.type == "cuda":
                # bf16-mirrored weights require autocast in eval as well
                with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
                    preds, features = self.predict(input)
                    target = self.transform_target(target)
                    losses = self.compute_evaluation_loss(preds, features, target)
            else:
                preds, features = self.predict(input)
                target = self.transform_target(target)
                losses = self.compute_evaluation_loss(preds, features, target)
        return losses, preds

    def _validate_on_loader(
        self, loader: DataLoader, loss_meter: LossMeter, metric_manager: MetricManager, include_losses_in_metrics: bool
    ) -> tuple[float, Metrics]:
        self.model.eval()
        metric_manager.clear()
        loss_meter.clear()
        with torch.no_grad():
            for input, target in loader:
                input, target = self._move_to_device(input), self._move_to_device(target)
                losses, preds = self.val_step(input, target)
                loss_meter.update(losses)
                metric_manager.update(preds, target)
        loss_dict = loss_meter.compute()
        metrics = metric_manager.compute()
        if include_losses_in_metrics:
            for key, val in loss_dict.items():
                metrics[f"{metric_manager.metric_manager_name} - {key}"] = val
        return loss_dict.get("checkpoint", 0.0), metrics

    def validate(self, include_losses_in_metrics: bool = False) -> tuple[float, Metrics]:
        """Validate (+ optional test loader, 'test -' prefixed keys; reference :821-928)."""
        if self.val_loader is None:
            return 0.0, {}
        val_loss, val_metrics = self._validate_on_loader(
            self.val_loader, self.val_loss_meter, self.val_metric_manager, include_losses_in_metrics
        )
        if self.test_loader is not None:
            test_loss, test_metrics = self._validate_on_loader(
                self.test_loader, self.test_loss_meter, self.test_metric_manager, include_losses_in_metrics
            )
            val_metrics.update({f"test - {k}" if not str(k).startswith("test") else k: v for k, v in test_metrics.items()})
            val_metrics["test - num_examples"] = len(getattr(self.test_loader, "dataset", []))
            val_metrics["test - checkpoint"] = test_loss
        return val_loss, val_metrics

    # ------------------------------------------------------------------
    # hooks (reference :1233-1302)
    # ------------------------------------------------------------------
    def update_before_train(self, current_server_round: int) -> None: ...

    def update_after_train(self, local_steps: int, loss_dict: dict[str, float], config: Config) -> None: ...

    def update_before_epoch(self, epoch: int) -> None: ...

    def update_before_step(self, step: int, current_round: int | None = None) -> None: ...

    def update_after_step(self, step: int, current_round: int | None = None) -> None: ...

    def update_lr_schedulers(self, step: int | None = None, epoch: int | None = None) -> None:
        for sched in self.lr_schedulers.values():
            sched.step()

    # ------------------------------------------------------------------
    # state / checkpointing / logging
    # ------------------------------------------------------------------
    def _maybe_checkpoint(self, loss: float, metrics: Metrics, mode: CheckpointMode) -> None:
        self.checkpoint_and_state_module.maybe_checkpoint(self.model, loss, metrics, mode)

    def _save_client_state(self) -> None:
        self.checkpoint_and_state_module.save_state(self, f"client_{self.client_name}_state.pt")

    def _maybe_load_client_state(self) -> bool:
        return self.checkpoint_and_state_module.maybe_load_state(self, f"client_{self.client_name}_state.pt")

    def _log_results(
        self, loss_dict: dict[str, float], metrics: Metrics, current_round: int | None = None, current_epoch: int | None = None
    ) -> None:
        round_str = f"Round: {current_round}" if current_round is not None else ""
        epoch_str = f"Epoch: {current_epoch}" if current_epoch is not None else ""
        losses = " ".join(f"{k}: {v:.4f}" for k, v in loss_dict.items())
        mets = " ".join(f"{k}: {v}" for k, v in metrics.items())
        log.info("Client %s %s %s | losses %s | metrics %s", self.client_name, round_str, epoch_str, losses, mets)

    def shutdown(self) -> None:
        self.reports_manager.report({"shutdown": str(datetime.datetime.now())})
        self.reports_manager.shutdown()
