from fl4health_amd.clients.basic_client import BasicClient

__all__ = ["BasicClient"]
