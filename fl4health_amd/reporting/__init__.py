from fl4health_amd.reporting.base_reporter import BaseReporter
from fl4health_amd.reporting.reports_manager import ReportsManager
from fl4health_amd.reporting.json_reporter import JsonReporter
from fl4health_amd.reporting.wandb_reporter import WandBReporter

__all__ = ["BaseReporter", "ReportsManager", "JsonReporter", "WandBReporter"]
