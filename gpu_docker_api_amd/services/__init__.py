from .saga import Saga
from .replicaset import ReplicaSetService
from .volume import VolumeService

__all__ = ["Saga", "ReplicaSetService", "VolumeService"]
