"""Pure-compute vs sync-wait timing — the DBS sensor.

The reference derives "pure compute time" as epoch wall time minus the
summed blocking all-reduce waits (dbs.py:241-250).  With communication
overlapped with backward that subtraction no longer exists, so we measure
directly on the GPU timeline with hipEvents (torch.cuda.Event on ROCm):

per iteration, four events on the compute stream:
    e0 — iteration start (before forward)
    e1 — end of backward's compute kernels
    e2 — recorded after the bucket-reduce stream dependencies are joined
         (completes only once the last all-reduce has finished)
    e3 — end of the optimizer step

compute += (e1-e0) + (e3-e2)        # fwd+bwd plus optimizer
sync    += (e2-e1)                  # comm tail not hidden by backward

This keeps the reference's semantic (sync time = time the step had to wait
on communication beyond compute) exact under overlap; SURVEY.md §5
"Timing fidelity" documents the deviation.  Injected fault sleeps are
added to compute (a slow worker's extra latency is what DBS must see).

CPU/gloo debug path uses perf_counter with the same accounting.
"""

from __future__ import annotations

import time

import torch

__all__ = ["StepTimer"]


class StepTimer:
    def __init__(self, device: torch.device):
        self.is_cuda = device.type == "cuda"
        self.reset()
        self._pool: list[torch.cuda.Event] = []
        self._used = 0

    # -------------------------------------------------- event plumbing
    def _event(self) -> "torch.cuda.Event":
        if self._used == len(self._pool):
            self._pool.append(torch.cuda.Event(enable_timing=True))
        e = self._pool[self._used]
        self._used += 1
        return e

    def reset(self) -> None:
        self.compute_s = 0.0
        self.sync_s = 0.0
        self._marks: list = []  # (e0, e1, e2, e3) per iteration
        self._cpu_t = None
        self._used = 0

    # -------------------------------------------------- per-iteration
    def iter_start(self):
        self._cur_inject = 0.0
        if self.is_cuda:
            e = self._event(); e.record()
            self._cur = [e]
        else:
            self._cur = [time.perf_counter()]

    def backward_done(self):
        self._mark()

    def comm_done(self):
        self._mark()

    def step_done(self):
        self._mark()
        self._cur.append(self._cur_inject)
        self._marks.append(self._cur)

    def _mark(self):
        if self.is_cuda:
            e = self._event(); e.record()
            self._cur.append(e)
        else:
            self._cur.append(time.perf_counter())

    def add_compute(self, seconds: float) -> None:
        """Fold host-side injected delay (fault injector) into compute.

        The sleep happens between backward_done and comm_done, so the same
        gap also lands in this iteration's e1→e2 interval; it is remembered
        here and subtracted from that interval in epoch_totals so the
        injected delay is attributed exactly once (to compute)."""
        self.compute_s += seconds
        if hasattr(self, "_cur_inject"):
            self._cur_inject += seconds

    # -------------------------------------------------- epoch close
    def drain(self) -> tuple[float, float]:
        """Consume completed iteration marks into the running totals;
        returns the (compute_s, sync_s) DELTA for the drained span.

        On GPU this synchronizes once.  Used mid-epoch by the
        iteration-granularity DBS mode (the per-interval sensor) and by
        epoch_totals at epoch close.
        """
        c0, s0 = self.compute_s, self.sync_s
        if self.is_cuda:
            torch.cuda.synchronize()
            for e0, e1, e2, e3, inj in self._marks:
                self.compute_s += (e0.elapsed_time(e1) + e2.elapsed_time(e3)) / 1e3
                self.sync_s += max(0.0, e1.elapsed_time(e2) / 1e3 - inj)
        else:
            for t0, t1, t2, t3, inj in self._marks:
                self.compute_s += (t1 - t0) + (t3 - t2)
                self.sync_s += max(0.0, (t2 - t1) - inj)
        self._marks.clear()
        self._used = 0
        return self.compute_s - c0, self.sync_s - s0

    def epoch_totals(self) -> tuple[float, float]:
        """(compute seconds, sync seconds) for all iterations since reset."""
        self.drain()
        return self.compute_s, self.sync_s
