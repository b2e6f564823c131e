from fl4health_amd.privacy.grad_sample import GradSampleModule
from fl4health_amd.privacy.dp_sgd import DpSgdEngine
from fl4health_amd.privacy.moments_accountant import MomentsAccountant
from fl4health_amd.privacy.fl_accountants import (
    FlClientLevelAccountantFixedSamplingNoReplacement,
    FlClientLevelAccountantPoissonSampling,
    FlInstanceLevelAccountant,
)

__all__ = [
    "GradSampleModule",
    "DpSgdEngine",
    "MomentsAccountant",
    "FlInstanceLevelAccountant",
    "FlClientLevelAccountantPoissonSampling",
    "FlClientLevelAccountantFixedSamplingNoReplacement",
]
