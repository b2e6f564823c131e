from fl4health_amd.mixins.adaptive_drift_constrained import (
    AdaptiveDriftConstrainedMixin,
    BaseFlexibleMixin,
    BasicClientProtocol,
    apply_adaptive_drift_to_client,
)
from fl4health_amd.mixins.personalized import (
    DittoPersonalizedMixin,
    MrMtlPersonalizedMixin,
    ensure_protocol_compliance,
    make_it_personal,
)

__all__ = [
    "AdaptiveDriftConstrainedMixin",
    "BaseFlexibleMixin",
    "BasicClientProtocol",
    "DittoPersonalizedMixin",
    "MrMtlPersonalizedMixin",
    "apply_adaptive_drift_to_client",
    "ensure_protocol_compliance",
    "make_it_personal",
]
