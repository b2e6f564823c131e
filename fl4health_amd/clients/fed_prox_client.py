"""FedProx client (re-export; reference fl4health/clients/fed_prox_client.py:4)."""
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient

__all__ = ["FedProxClient"]
