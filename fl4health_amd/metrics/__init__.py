from fl4health_amd.metrics.base_metrics import Metric, SimpleMetric, TorchMetric
from fl4health_amd.metrics.metrics import (
    F1,
    Accuracy,
    BalancedAccuracy,
    BinarySoftDiceCoefficient,
    RocAuc,
)
from fl4health_amd.metrics.efficient_metrics import BinaryDice, MultiClassDice
from fl4health_amd.metrics.compound_metrics import EmaMetric, TransformsMetric
from fl4health_amd.metrics.metric_managers import MetricManager

__all__ = [
    "Metric",
    "SimpleMetric",
    "TorchMetric",
    "Accuracy",
    "BalancedAccuracy",
    "RocAuc",
    "F1",
    "BinarySoftDiceCoefficient",
    "BinaryDice",
    "MultiClassDice",
    "EmaMetric",
    "TransformsMetric",
    "MetricManager",
]
