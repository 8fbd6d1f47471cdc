import sys, os, tempfile
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""P2E-DV3 exploration graphed gradient-step rate at default dims: run the
real CLI loop for ~160 post-prefill steps and read the cumulative
Time/train_time from the timer registry (p2e accumulates it but never
resets)."""
from sheeprl_amd.cli import run
from sheeprl_amd.utils.timer import timer

# the loop's final log drains+resets the registry; intercept the reset to
# accumulate the totals across log intervals
INTERVALS = []
_orig_reset = timer.reset
def _patched_reset():
    t = timer.compute().get("Time/train_time")
    if t:
        INTERVALS.append(t)
    _orig_reset()
timer.reset = _patched_reset

STEPS = 160
tmp = tempfile.mkdtemp()
run(["exp=p2e_dv3_exploration", "env=synthetic_atari", "runtime.accelerator=cuda",
     "runtime.precision=bf16", "algo.replay_ratio=1", f"algo.total_steps={1024 + STEPS}",
     "algo.learning_starts=1024", "metric.log_level=1", "metric.log_every=25",
     "metric.disable_timer=False", "algo.run_test=False", "checkpoint.every=0",
     "checkpoint.save_last=False", f"root_dir={tmp}"])
t = timer.compute().get("Time/train_time")
if t:
    INTERVALS.append(t)
# 25 gradient steps per interval (replay_ratio 1, log_every 25); the first
# intervals carry eager warmup + graph capture — report the steady median
import statistics
steady = INTERVALS[2:] or INTERVALS
ms = statistics.median(steady) / 25 * 1000
print(f"intervals (s per 25 steps): {[round(x,2) for x in INTERVALS]}")
print(f"p2e_dv3_exploration steady: {ms:.1f} ms per graphed gradient step")
