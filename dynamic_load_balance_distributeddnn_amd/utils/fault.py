"""Straggler (fault) injector — robustness testing for the DBS loop.

Reference behavior (dbs.py:94-129): each epoch a worker rolls the dice
once; with probability ``chance`` it enters a slow phase that adds 5-10
seconds per epoch (spread over that epoch's iterations) lasting until a
random future epoch 4-20 epochs away.  DBS should respond by shrinking the
slow rank's batch share.

Fixes vs reference: the reference's ``saved_epoch`` is read before ever
being assigned (dbs.py:95,109 — NameError on first use when `-ft true`);
here all state lives in this object and is initialized.  Randomness is
seeded per-rank so tests can be deterministic.
"""

from __future__ import annotations

import random
import time

__all__ = ["FaultInjector"]


class FaultInjector:
    def __init__(self, enabled: bool, chance: float, rank: int,
                 seed: int | None = None, logger=None):
        self.enabled = enabled
        self.chance = chance
        self.rank = rank
        self.rng = random.Random(seed if seed is not None else (rank * 7919 + 1))
        self.logger = logger
        self.slow_until_epoch = -1   # epoch (inclusive) to stay slow
        self.extra_per_epoch = 0.0   # seconds of injected delay per epoch
        self._last_rolled_epoch = -1

    def maybe_wait(self, epoch: int, steps_per_epoch: int) -> float:
        """Call once per iteration; sleeps if in a slow phase.

        Returns the injected seconds (the engine adds it to measured
        compute time so the DBS sensor sees the straggle).
        """
        if not self.enabled:
            return 0.0

        if epoch <= self.slow_until_epoch:
            delay = self.extra_per_epoch / max(1, steps_per_epoch)
            time.sleep(delay)
            return delay

        if self._last_rolled_epoch == epoch:
            return 0.0
        self._last_rolled_epoch = epoch

        if self.rng.random() < self.chance:
            self.extra_per_epoch = self.rng.randint(5, 10)
            self.slow_until_epoch = epoch + self.rng.randint(4, 20)
            if self.logger:
                self.logger.info(
                    f"Rank {self.rank}: injected straggle of "
                    f"{self.extra_per_epoch}s/epoch until epoch "
                    f"{self.slow_until_epoch}")
        return 0.0
