from .sharding import ShardingPolicy
from .dist import DistributedScheduler, init_distributed
