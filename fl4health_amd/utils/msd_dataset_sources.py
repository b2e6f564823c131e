"""Medical Segmentation Decathlon task registry (capability of reference
fl4health/utils/msd_dataset_sources.py). This image has no network access, so
the registry carries task names, modality counts and label counts for plans
bootstrapping only — point `msd_root` at an already-downloaded copy."""
from __future__ import annotations

from enum import Enum


class MsdDataset(Enum):
    TASK01_BRAINTUMOUR = "Task01_BrainTumour"
    TASK02_HEART = "Task02_Heart"
    TASK03_LIVER = "Task03_Liver"
    TASK04_HIPPOCAMPUS = "Task04_Hippocampus"
    TASK05_PROSTATE = "Task05_Prostate"
    TASK06_LUNG = "Task06_Lung"
    TASK07_PANCREAS = "Task07_Pancreas"
    TASK08_HEPATICVESSEL = "Task08_HepaticVessel"
    TASK09_SPLEEN = "Task09_Spleen"
    TASK10_COLON = "Task10_Colon"


# (input modalities, segmentation labels incl. background)
MSD_TASK_DIMS: dict[MsdDataset, tuple[int, int]] = {
    MsdDataset.TASK01_BRAINTUMOUR: (4, 4),
    MsdDataset.TASK02_HEART: (1, 2),
    MsdDataset.TASK03_LIVER: (1, 3),
    MsdDataset.TASK04_HIPPOCAMPUS: (1, 3),
    MsdDataset.TASK05_PROSTATE: (2, 3),
    MsdDataset.TASK06_LUNG: (1, 2),
    MsdDataset.TASK07_PANCREAS: (1, 3),
    MsdDataset.TASK08_HEPATICVESSEL: (1, 3),
    MsdDataset.TASK09_SPLEEN: (1, 2),
    MsdDataset.TASK10_COLON: (1, 2),
}


def get_msd_dataset_enum(dataset_name: str) -> MsdDataset:
    for task in MsdDataset:
        if task.value == dataset_name:
            return task
    raise ValueError(f"unknown MSD dataset {dataset_name!r}")
