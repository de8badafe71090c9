"""cProfile one bench step at the headline shape (CPU oracle path).

Usage: python tools/profile_step.py [nodes] [jobs] [pods_per_job] [--mix]
"""
import cProfile
import gc
import os
import pstats
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
mix = "--mix" in sys.argv
args = [a for a in sys.argv[1:] if not a.startswith("--")]
nodes = int(args[0]) if len(args) > 0 else 10000
jobs = int(args[1]) if len(args) > 1 else 10000
ppj = int(args[2]) if len(args) > 2 else 10
sys.argv = ["bench"]
from bench import build_cluster, reset_cluster  # noqa: E402

from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)

config = default_config()
config.use_hip = False
cache = SchedulerCache(store=None, binder=FakeBinder(), device="cpu")
sched = Scheduler(cache, config)
jl = build_cluster(cache, nodes, jobs, ppj, 0, 1, mix=mix)

# warmup
reset_cluster(cache, jl)
sched.run_once()
gc.collect()
gc.freeze()

pr = cProfile.Profile()
pr.enable()
reset_cluster(cache, jl)
sched.run_once()
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("cumulative").print_stats(45)
