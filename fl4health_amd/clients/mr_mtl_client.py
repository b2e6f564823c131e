"""MR-MTL client (re-export; implementation lives with the adaptive
drift-constraint family — reference fl4health/clients/mr_mtl_client.py:18)."""
from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient

__all__ = ["MrMtlClient"]
