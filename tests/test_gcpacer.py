"""GC pacer: threshold/freeze lifecycle and paced collection."""
import asyncio
import gc

from gpu_provisioner_amd.operator.gcpacer import GCPacer
from tests.conftest import run


def test_pacer_engage_disengage_restores_state():
    async def main():
        before = gc.get_threshold()
        p = GCPacer(gen2_interval=0.05)
        p.engage()
        try:
            assert gc.get_threshold()[2] >= 1_000_000  # automatic gen2 off
            assert gc.isenabled()  # gen0/gen1 stay automatic
            # paced collection actually runs and observes the metric
            from gpu_provisioner_amd.metrics.registry import GC_PAUSE_SECONDS

            h = GC_PAUSE_SECONDS.labels(generation="2")
            count0 = sum(b.get() for b in h._buckets)
            await asyncio.sleep(0.15)
            assert sum(b.get() for b in h._buckets) > count0
        finally:
            await p.disengage()
        assert gc.get_threshold() == before

    run(main())


def test_pacer_engage_idempotent():
    async def main():
        p = GCPacer(gen2_interval=60.0)
        p.engage()
        p.engage()  # no double-freeze / task leak
        await p.disengage()
        await p.disengage()

    run(main())


def test_nested_pacers_restore_only_at_last_disengage():
    """Two managers in one process (HA tests): the first disengage must
    NOT re-enable automatic gen2 under the survivor."""

    async def main():
        before = gc.get_threshold()
        p1, p2 = GCPacer(gen2_interval=60.0), GCPacer(gen2_interval=60.0)
        p1.engage()
        p2.engage()
        await p1.disengage()
        assert gc.get_threshold()[2] >= 1_000_000  # p2 still paced
        await p2.disengage()
        assert gc.get_threshold() == before

    run(main())
