"""EarlyStopper (reference fl4health/utils/early_stopper.py:14-98):
snapshot/restore the best in-round client state with patience + interval."""
from __future__ import annotations

import tempfile
from pathlib import Path
from typing import Any

from fl4health_amd.checkpointing.state_checkpointer import ClientStateCheckpointer


class EarlyStopper:
    def __init__(
        self,
        client: Any,
        patience: int | None = 1,
        interval_steps: int = 5,
        snapshot_dir: str | Path | None = None,
    ) -> None:
        self.client = client
        self.patience = patience
        self.count_down = patience
        self.interval_steps = interval_steps
        self.best_score: float | None = None
        dir_ = Path(snapshot_dir) if snapshot_dir is not None else Path(tempfile.mkdtemp())
        self.state_checkpointer = ClientStateCheckpointer(dir_, "early_stop_best.pt")

    def load_snapshot(self) -> None:
        if self.state_checkpointer.state_exists():
            self.state_checkpointer.load_state(self.client)

    def should_stop(self, steps: int) -> bool:
        """Validate; snapshot on improvement; count down patience otherwise."""
        if steps % self.interval_steps != 0:
            return False
        val_loss, _ = self.client.validate(include_losses_in_metrics=False)
        if self.best_score is None or val_loss < self.best_score:
            self.best_score = val_loss
            self.count_down = self.patience
            self.state_checkpointer.save_state(self.client)
            return False
        if self.count_down is not None:
            self.count_down -= 1
            if self.count_down <= 0:
                return True
        return False
