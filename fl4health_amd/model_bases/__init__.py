from fl4health_amd.model_bases.sequential_split_models import (
    SequentiallySplitExchangeBaseModel,
    SequentiallySplitModel,
)
from fl4health_amd.model_bases.parallel_split_models import (
    ParallelFeatureJoinMode,
    ParallelSplitHeadModule,
    ParallelSplitModel,
)
from fl4health_amd.model_bases.fenda_base import FendaModel, FendaModelWithFeatureState
from fl4health_amd.model_bases.moon_base import MoonModel
from fl4health_amd.model_bases.perfcl_base import PerFclModel
from fl4health_amd.model_bases.apfl_base import ApflModule
from fl4health_amd.model_bases.fedrep_base import FedRepModel
from fl4health_amd.model_bases.ensemble_base import EnsembleAggregationMode, EnsembleModel
from fl4health_amd.model_bases.fedsimclr_base import FedSimClrModel
from fl4health_amd.model_bases.feature_extractor_buffer import FeatureExtractorBuffer

__all__ = [
    "SequentiallySplitModel",
    "SequentiallySplitExchangeBaseModel",
    "ParallelSplitModel",
    "ParallelSplitHeadModule",
    "ParallelFeatureJoinMode",
    "FendaModel",
    "FendaModelWithFeatureState",
    "MoonModel",
    "PerFclModel",
    "ApflModule",
    "FedRepModel",
    "EnsembleModel",
    "EnsembleAggregationMode",
    "FedSimClrModel",
    "FeatureExtractorBuffer",
]
