from fl4health_amd.datasets.synthetic import synthetic_classification_dataset, synthetic_cifar_loaders
from fl4health_amd.datasets.partitioners import DirichletLabelPartitioner

__all__ = ["synthetic_classification_dataset", "synthetic_cifar_loaders", "DirichletLabelPartitioner"]
