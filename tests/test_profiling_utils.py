"""StepTimer / profiling helpers (aux subsystem)."""
import time

from perceiver_amd.utils.profiling import StepTimer


def test_step_timer_collects_samples():
    t = StepTimer()
    for _ in range(3):
        with t:
            time.sleep(0.01)
    s = t.summary()
    assert len(t.samples_ms) == 3
    assert s["mean_ms"] >= 5.0
    assert s["min_ms"] <= s["mean_ms"]
