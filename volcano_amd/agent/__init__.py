from .eventsmgr import EventsManager
from .oversubscription import OversubscriptionHandler
from .eviction import EvictionHandler
from .qos import CpuQosHandler, MemoryQosHandler, NetworkQosHandler
