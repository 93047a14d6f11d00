"""Profiling / tracing harness.

The reference has no observability at all (stdout prints
only).  This module provides:

* `profile_trace(path)` — torch.profiler context capturing CPU + HIP
  kernel activity, exported as a chrome trace.
* `StepTimer` — rolling wall/device timing of training steps with
  percentile summaries (the per-kernel view comes from
  `rocprofv3 --kernel-trace --stats --output-format csv`, see README).
* `kernel_stats_summary(csv_path)` — parse a rocprofv3 kernel-stats CSV
  into (name, calls, total_ms, pct) rows, the format used for the
  committed evidence under profiles/.
"""
import contextlib
import csv
import time

import torch


@contextlib.contextmanager
def profile_trace(path='trace.json', warmup=1, active=3):
    """Capture a chrome trace of CPU + device activity around the body."""
    from torch.profiler import ProfilerActivity, profile
    activities = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(ProfilerActivity.CUDA)
    with profile(activities=activities, record_shapes=False) as prof:
        yield prof
    prof.export_chrome_trace(path)


class StepTimer:
    """Rolling step timer: call .step() around each iteration."""

    def __init__(self, sync_cuda=True):
        self.sync_cuda = sync_cuda and torch.cuda.is_available()
        self.times = []
        self._t0 = None

    def __enter__(self):
        if self.sync_cuda:
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        if self.sync_cuda:
            torch.cuda.synchronize()
        self.times.append(time.perf_counter() - self._t0)
        return False

    def summary(self):
        if not self.times:
            return {}
        ts = sorted(self.times)
        n = len(ts)
        return {
            'steps': n,
            'mean_ms': sum(ts) / n * 1e3,
            'p50_ms': ts[n // 2] * 1e3,
            'p90_ms': ts[min(n - 1, int(n * 0.9))] * 1e3,
            'max_ms': ts[-1] * 1e3,
        }


def kernel_stats_summary(csv_path, top=20):
    """Parse a rocprofv3 kernel_stats CSV -> list of dicts sorted by
    total time, with percentage share."""
    with open(csv_path) as f:
        rows = list(csv.DictReader(f))
    rows.sort(key=lambda r: float(r['TotalDurationNs']), reverse=True)
    total = sum(float(r['TotalDurationNs']) for r in rows) or 1.0
    out = []
    for r in rows[:top]:
        t = float(r['TotalDurationNs'])
        out.append({
            'name': r['Name'],
            'calls': int(r['Calls']),
            'total_ms': t / 1e6,
            'avg_us': float(r['AverageNs']) / 1e3,
            'pct': 100.0 * t / total,
        })
    return out
