"""Resource management: usable CPU count, memory release, model lifecycle.

References: /root/reference/cpu_budget.py (usable_cpu_count :107 —
cgroup/affinity-aware), tasks/memory_utils.py (release_memory_to_os
:185 malloc_trim, cleanup pools, SessionRecycler :340). On MI355X the
288 GB HBM makes the reference's aggressive unload cycle mostly moot —
models stay resident — but the hooks exist for small-GPU deployments.
"""

from __future__ import annotations

import ctypes
import gc
import os

import torch


def usable_cpu_count() -> int:
    """Affinity- and cgroup-aware CPU budget (cpu_budget.py:107)."""
    try:
        affinity = len(os.sched_getaffinity(0))
    except AttributeError:
        affinity = os.cpu_count() or 1
    quota = None
    try:
        with open("/sys/fs/cgroup/cpu.max") as fh:
            parts = fh.read().split()
            if parts[0] != "max":
                quota = max(1, int(int(parts[0]) / int(parts[1])))
    except OSError:
        try:
            with open("/sys/fs/cgroup/cpu/cpu.cfs_quota_us") as fh:
                q = int(fh.read())
            with open("/sys/fs/cgroup/cpu/cpu.cfs_period_us") as fh:
                p = int(fh.read())
            if q > 0:
                quota = max(1, q // p)
        except OSError:
            pass
    return min(affinity, quota) if quota else affinity


def release_memory_to_os() -> None:
    """gc + malloc_trim + HIP cache release (memory_utils.py:185)."""
    gc.collect()
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
    try:
        ctypes.CDLL("libc.so.6").malloc_trim(0)
    except Exception:
        pass


class ModelLifecycle:
    """Warm/idle model lifecycle (reference: CLAP text lazy-load +
    warm-up countdown, clap_text_search.warmup_text_search_model :99).
    Keeps a factory-built model resident for `idle_seconds` after last
    use, then unloads."""

    def __init__(self, factory, idle_seconds: float = 600.0):
        import time

        self._factory = factory
        self._idle = idle_seconds
        self._model = None
        self._last = 0.0
        self._time = time.monotonic

    def get(self):
        self._last = self._time()
        if self._model is None:
            self._model = self._factory()
        return self._model

    def remaining(self) -> float:
        """Seconds until idle unload; 0.0 when not loaded."""
        if self._model is None:
            return 0.0
        return max(0.0, self._idle - (self._time() - self._last))

    def maybe_unload(self) -> bool:
        if self._model is not None and self._time() - self._last > self._idle:
            self._model = None
            release_memory_to_os()
            return True
        return False

    @property
    def loaded(self) -> bool:
        return self._model is not None


def oom_retry(fn, x: torch.Tensor, min_chunk: int = 1):
    """Run `fn(batch)` over `x`, halving the batch on HIP out-of-memory
    until it fits (reference: memory_utils.handle_onnx_memory_error :282
    — its ONNX OOM -> CPU-retry ladder; here the ladder is batch halving
    on the same GPU, since 288 GB of HBM makes a CPU fallback strictly
    worse). Results are concatenated along dim 0."""
    n = x.shape[0]
    chunk = n
    while True:
        try:
            if chunk >= n:
                return fn(x)
            outs = [fn(x[i : i + chunk]) for i in range(0, n, chunk)]
            return torch.cat(outs, dim=0)
        except torch.cuda.OutOfMemoryError:
            release_memory_to_os()
            if chunk <= min_chunk:
                raise
            chunk = max(min_chunk, chunk // 2)
