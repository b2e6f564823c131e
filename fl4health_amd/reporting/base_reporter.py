"""Push-based reporter observers (reference fl4health/reporting/base_reporter.py:10-51)."""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any


class BaseReporter(ABC):
    def initialize(self, **kwargs: Any) -> None:
        """Called once with identifying info (id, name...)."""

    @abstractmethod
    def report(
        self,
        data: dict[str, Any],
        round: int | None = None,
        epoch: int | None = None,
        step: int | None = None,
    ) -> None: ...

    def shutdown(self) -> None:
        """Flush/close."""
