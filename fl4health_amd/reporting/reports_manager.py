"""Broadcasts report events to all reporters (reference reporting/reports_manager.py:7-27)."""
from __future__ import annotations

from typing import Any

from fl4health_amd.reporting.base_reporter import BaseReporter


class ReportsManager:
    def __init__(self, reporters: list[BaseReporter] | None = None) -> None:
        self.reporters = list(reporters) if reporters else []

    def initialize(self, **kwargs: Any) -> None:
        for r in self.reporters:
            r.initialize(**kwargs)

    def report(
        self,
        data: dict[str, Any],
        round: int | None = None,
        epoch: int | None = None,
        step: int | None = None,
    ) -> None:
        for r in self.reporters:
            r.report(data, round, epoch, step)

    def shutdown(self) -> None:
        for r in self.reporters:
            r.shutdown()
