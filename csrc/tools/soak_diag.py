import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
os.chdir(os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
sys.argv = ["bench.py", "--gpus", "1", "--steps", "300", "--warmup", "8"]
import time
import bench
from production_stack_amd.engine.engine import LLMEngine
orig_step = LLMEngine.step
stats = {"n": 0}
def step(self):
    stats["n"] += 1
    if stats["n"] % 100 == 0:
        m = self.engine_metrics()
        bm = self.block_manager
        import torch
        print(f"step {stats['n']}: running={m['num_requests_running']:.0f} "
              f"waiting={m['num_requests_waiting']:.0f} "
              f"cache_usage={m['gpu_cache_usage_perc']:.3f} "
              f"free_blocks={bm.num_free} "
              f"prefix_hits={m['gpu_prefix_cache_hits_total']:.0f}/"
              f"{m['gpu_prefix_cache_queries_total']:.0f} "
              f"preempted={getattr(self.scheduler, 'num_preempted', '?')}",
              file=sys.stderr, flush=True)
    return orig_step(self)
LLMEngine.step = step
# count preemptions
from production_stack_amd.engine import scheduler as sched
orig_p = sched.Scheduler._preempt_last
def pre(self, out, keep):
    r = orig_p(self, out, keep)
    if r:
        self.num_preempted = getattr(self, "num_preempted", 0) + 1
    return r
sched.Scheduler._preempt_last = pre
bench.main()
