from .context import Comm, get_comm, init_comm, shutdown_comm, PartitionDescriptor

__all__ = ["Comm", "get_comm", "init_comm", "shutdown_comm", "PartitionDescriptor"]
