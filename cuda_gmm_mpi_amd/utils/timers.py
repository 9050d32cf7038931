"""Accumulating profile timers with the reference's bucket names.

Mirrors ``profile_t`` (gaussian.cu:76-106): buckets e_step / m_step /
constants / reduce / memcpy / cpu / comm (the reference's "mpi" bucket),
each an accumulating timer, reported with per-iteration averages like
gaussian.cu:967. On GPU the timers use hipEvents (torch.cuda.Event);
on CPU, perf_counter.
"""
from __future__ import annotations

import time
from contextlib import contextmanager

import torch

BUCKETS = ("e_step", "m_step", "constants", "reduce", "memcpy", "cpu", "comm")


class _Timer:
    """One accumulating timer (cudaTimer_t equivalent, gaussian.cu:33-72)."""

    def __init__(self, use_events: bool):
        self.total_ms = 0.0
        self.use_events = use_events
        self._pairs: list[tuple[torch.cuda.Event, torch.cuda.Event]] = []
        self._t0 = 0.0

    def start(self):
        if self.use_events:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._pairs.append((e, None))
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if self.use_events:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            s, _ = self._pairs[-1]
            self._pairs[-1] = (s, e)
        else:
            self.total_ms += (time.perf_counter() - self._t0) * 1e3

    def value_ms(self) -> float:
        if self.use_events and self._pairs:
            torch.cuda.synchronize()
            for s, e in self._pairs:
                if e is not None:
                    self.total_ms += s.elapsed_time(e)
            self._pairs.clear()
        return self.total_ms


class Profile:
    """profile_t equivalent: named accumulating timers + iteration counts."""

    def __init__(self, device: torch.device | str = "cpu"):
        dev = torch.device(device)
        use_events = dev.type == "cuda" and torch.cuda.is_available()
        self.timers = {b: _Timer(use_events) for b in BUCKETS}
        self.iterations = {"regroup": 0, "params": 0, "constants": 0, "reduce": 0}
        self.paused = False  # hipEvents cannot be recorded inside graph capture

    @contextmanager
    def time(self, bucket: str):
        if self.paused:
            yield
            return
        t = self.timers[bucket]
        t.start()
        try:
            yield
        finally:
            t.stop()

    def count(self, which: str, n: int = 1):
        if not self.paused:
            self.iterations[which] += n

    def report(self, rank: int = 0, gpu: int = 0) -> str:
        """Per-GPU report in the reference's shape (gaussian.cu:967)."""
        v = {b: self.timers[b].value_ms() / 1000.0 for b in BUCKETS}

        def avg(total, iters):
            return total / iters if iters else 0.0

        it = self.iterations
        return (
            f"Node {rank:02d} GPU {gpu}:\n"
            f"\tE-step Kernel:\t{v['e_step']:7.4f}\t{it['regroup']}\t"
            f"{avg(v['e_step'], it['regroup']):7.4f}\n"
            f"\tM-step Kernel:\t{v['m_step']:7.4f}\t{it['params']}\t"
            f"{avg(v['m_step'], it['params']):7.4f}\n"
            f"\tConsts Kernel:\t{v['constants']:7.4f}\t{it['constants']}\t"
            f"{avg(v['constants'], it['constants']):7.4f}\n"
            f"\tOrder Reduce:\t{v['reduce']:7.4f}\t{it['reduce']}\t"
            f"{avg(v['reduce'], it['reduce']):7.4f}\n"
            f"\tGPU Memcpy:\t{v['memcpy']:7.4f}\n"
            f"\tCPU:\t\t{v['cpu']:7.4f}\n"
            f"\tComm:\t\t{v['comm']:7.4f}\n"
        )

    def totals_ms(self) -> dict[str, float]:
        return {b: self.timers[b].value_ms() for b in BUCKETS}
