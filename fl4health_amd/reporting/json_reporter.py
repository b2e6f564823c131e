"""JsonReporter: accumulates round-level dicts, dumps {output_folder}/{run_id}.json
at shutdown (reference fl4health/reporting/json_reporter.py:45-97). Its output is
what the golden-metric smoke tests compare against.
"""
from __future__ import annotations

import json
import logging
import os
from pathlib import Path
from typing import Any

from fl4health_amd.utils.random import generate_hash

log = logging.getLogger(__name__)


class DateTimeEncoder(json.JSONEncoder):
    def default(self, o: Any):
        import datetime

        if isinstance(o, (datetime.datetime, datetime.date)):
            return str(o)
        try:
            return float(o)
        except (TypeError, ValueError):
            return str(o)


class JsonReporter:
    def __init__(self, run_id: str | None = None, output_folder: str | Path = "metrics") -> None:
        self.run_id = run_id if run_id is not None else generate_hash()
        self.output_folder = Path(output_folder)
        self.metrics: dict[str, Any] = {}
        self.initialized = False

    def initialize(self, **kwargs: Any) -> None:
        if "id" in kwargs and not self.initialized:
            self.run_id = kwargs.get("id", self.run_id)
        self.initialized = True
        self.metrics.setdefault("host_type", kwargs.get("name", ""))

    def report(self, data: dict[str, Any], round: int | None = None, epoch: int | None = None, step: int | None = None) -> None:
        if round is None:
            self.metrics.update(data)
        else:
            rounds = self.metrics.setdefault("rounds", {})
            rd = rounds.setdefault(round, {})
            if epoch is None and step is None:
                rd.update(data)
            elif epoch is not None:
                rd.setdefault("epochs", {}).setdefault(epoch, {}).update(data)
            else:
                rd.setdefault("steps", {}).setdefault(step, {}).update(data)

    def dump(self) -> None:
        os.makedirs(self.output_folder, exist_ok=True)
        path = self.output_folder / f"{self.run_id}.json"
        with open(path, "w") as f:
            json.dump(self.metrics, f, indent=4, cls=DateTimeEncoder)
        log.info("Dumped metrics to %s", path)

    def shutdown(self) -> None:
        self.dump()
