from fl4health_amd.servers.base_server import FlServer

__all__ = ["FlServer"]
