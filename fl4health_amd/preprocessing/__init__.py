from fl4health_amd.preprocessing.warmed_up_module import WarmedUpModule
from fl4health_amd.preprocessing.pca_preprocessor import PcaPreprocessor

__all__ = ["WarmedUpModule", "PcaPreprocessor"]
