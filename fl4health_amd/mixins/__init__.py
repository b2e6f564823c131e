from fl4health_amd.mixins.adaptive_drift_constrained import (
    AdaptiveDriftConstrainedMixin,
    apply_adaptive_drift_to_client,
)

__all__ = ["AdaptiveDriftConstrainedMixin", "apply_adaptive_drift_to_client"]
