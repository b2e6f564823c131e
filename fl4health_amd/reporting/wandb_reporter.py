"""WandBReporter (capability of reference reporting/wandb_reporter.py:21-247).

wandb is not installed in this offline image: the reporter degrades to a
warning + no-op unless `wandb` is importable.
"""
from __future__ import annotations

import logging
from typing import Any

log = logging.getLogger(__name__)


class WandBReporter:
    def __init__(self, wandb_step_type: str = "round", **wandb_init_kwargs: Any) -> None:
        self.wandb_step_type = wandb_step_type
        self.init_kwargs = wandb_init_kwargs
        self._run = None
        try:
            import wandb  # noqa: F401

            self._wandb = wandb
        except ImportError:
            self._wandb = None
            log.warning("wandb not installed; WandBReporter is a no-op")

    def initialize(self, **kwargs: Any) -> None:
        if self._wandb is not None and self._run is None:
            self._run = self._wandb.init(**self.init_kwargs)

    def report(self, data: dict[str, Any], round: int | None = None, epoch: int | None = None, step: int | None = None) -> None:
        if self._run is None:
            return
        payload = dict(data)
        if round is not None:
            payload["fl_round"] = round
        if epoch is not None:
            payload["epoch"] = epoch
        if step is not None:
            payload["step"] = step
        self._run.log({k: v for k, v in payload.items() if isinstance(v, (int, float, str))})

    def shutdown(self) -> None:
        if self._run is not None:
            self._run.finish()
