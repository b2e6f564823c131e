"""Privacy module utilities (reference fl4health/utils/privacy_utilities.py:11-71)."""
from fl4health_amd.privacy.grad_sample import (  # noqa: F401
    GradSampleModule,
    convert_batchnorm_modules,
    validate_module,
)


def privacy_validate_and_fix_modules(model):
    """Convert BatchNorm to GroupNorm and validate DP compatibility."""
    model = convert_batchnorm_modules(model)
    validate_module(model)
    return model, []


def map_model_to_opacus_model(model):
    """Wrap with the per-sample gradient engine (Opacus GradSampleModule analog)."""
    return GradSampleModule(model)
