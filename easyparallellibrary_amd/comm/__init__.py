from easyparallellibrary_amd.comm.backend import (
    Communicator, create_communicator, destroy_all_communicators,
    get_communicator)
from easyparallellibrary_amd.comm.facade import CollectiveCommunicator
from easyparallellibrary_amd.comm.pool import CommunicationPool

__all__ = [
    "Communicator", "create_communicator", "get_communicator",
    "destroy_all_communicators", "CollectiveCommunicator",
    "CommunicationPool",
]
