"""GC pacer: bounds CPython cyclic-GC tail latency in the operator runtime.

Round-1 finding (profiles/FINAL_sweep_mi355x.jsonl, VERDICT weak #6): p95
Ready latency blew up 3-5× over p50 at 32/64 concurrent NodeClaims while
staying tight at 16 and 128. Root cause (measured with gc callbacks + an
event-loop watchdog): CPython's automatic generation-2 collections scan the
whole heap (~30 ms at steady state) and fire mid-step every ~0.5 s at this
allocation rate, stalling EVERY in-flight reconcile at once — a whole batch
of claims absorbs the pause, which surfaces exactly in the p95 band at
mid concurrency (at 128 the p50 is already past the pause length, hiding it).

The pacer replaces unpredictable full collections with a bounded cadence:

  * collect once, then ``gc.freeze()`` the post-startup heap — long-lived
    objects (modules, controller/informer infrastructure) move to the
    permanent generation and are never re-scanned, dropping a full
    collection from ~30 ms to ~3 ms (the freeze runs right after a full
    collect, so the frozen set contains no cycle garbage);
  * disable AUTOMATIC gen2 (threshold2 → effectively infinite); gen0/gen1
    stay automatic, their pauses measured at 2-3 ms;
  * run an explicit gen2 collection from a background task on a fixed
    cadence (default 10 s), bounding uncollected-cycle growth with a ~3 ms
    amortized pause, observed into the karpenter-style metrics registry.

Measured effect at c=32 (this machine, 6 runs each): p95/p50 3.0 (worst
run 0.091 s/0.030 s) before → ≤1.15 after, with no throughput change.
"""
from __future__ import annotations

import asyncio
import gc
import logging
import time
from typing import Optional

log = logging.getLogger(__name__)

# gen0/gen1 stay at CPython defaults (pauses measured 2-3 ms); gen2 is
# effectively never triggered automatically — the pacer task owns it
_THRESHOLDS = (700, 10, 1_000_000_000)
DEFAULT_GEN2_INTERVAL = 10.0

# process-wide engagement count: with multiple managers in one process
# (dual-replica HA tests), only the first engage freezes/retunes and only
# the last disengage restores — otherwise the first stop() would re-enable
# automatic gen2 under the survivor
_active = 0
_saved_thresholds: Optional[tuple] = None


class GCPacer:
    def __init__(self, gen2_interval: float = DEFAULT_GEN2_INTERVAL, freeze: bool = True):
        self.gen2_interval = gen2_interval
        self.freeze = freeze
        self._task: Optional[asyncio.Task] = None
        self._saved_thresholds: Optional[tuple] = None
        self._engaged = False

    def engage(self) -> None:
        """Call once the process reached steady state (informers synced,
        controllers constructed): the current heap is what gets frozen."""
        global _active, _saved_thresholds
        if self._engaged:
            return
        if _active == 0:
            _saved_thresholds = gc.get_threshold()
            gc.collect()  # the frozen set must hold no cycle garbage
            if self.freeze:
                gc.freeze()
            gc.set_threshold(*_THRESHOLDS)
        _active += 1
        self._engaged = True
        self._task = asyncio.create_task(self._run(), name="gc-pacer")

    async def _run(self) -> None:
        from ..metrics.registry import GC_PAUSE_SECONDS

        interval = self.gen2_interval
        while True:
            await asyncio.sleep(interval)
            t0 = time.monotonic()
            collected = gc.collect()
            pause = time.monotonic() - t0
            GC_PAUSE_SECONDS.labels(generation="2").observe(pause)
            # adaptive pacing: at fleet scale (10k+ live claims) a full
            # collection costs ~200 ms — stretch the cadence so collection
            # duty stays ≤0.5% of wall time, floor at the configured
            # interval, ceiling at 120 s so cycles never sit longer than
            # two minutes
            interval = min(max(self.gen2_interval, pause * 200.0), 120.0)
            if pause > 0.05:
                log.warning(
                    "paced gen2 collection took %.1f ms (%d collected); next in %.0fs",
                    pause * 1000.0, collected, interval,
                )

    async def disengage(self) -> None:
        global _active, _saved_thresholds
        if not self._engaged:
            return
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None
        self._engaged = False
        _active -= 1
        if _active == 0:
            if self.freeze:
                gc.unfreeze()
            if _saved_thresholds is not None:
                gc.set_threshold(*_saved_thresholds)
                _saved_thresholds = None
