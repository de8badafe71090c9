from .cache import Binder, FakeBinder, SchedulerCache
from .config import SchedulerConfiguration, default_config
from .engine import Scheduler
from .session import Session
