"""Wakeup semantics of the rewritten RateLimitingQueue: single-wakeup adds
(no thundering herd), no lost wakeups under contention, cancellation
hand-off, and shutdown waking every parked worker."""
import asyncio

from gpu_provisioner_amd.kube.workqueue import RateLimiter, RateLimitingQueue
from tests.conftest import run


def fast_limiter() -> RateLimiter:
    return RateLimiter(base=0.001, cap=0.01, qps=1e6, burst=1000000)


def test_add_wakes_exactly_one_parked_getter():
    async def main():
        q = RateLimitingQueue(fast_limiter())
        results = []

        async def getter(i):
            item = await q.get()
            results.append((i, item))

        tasks = [asyncio.create_task(getter(i)) for i in range(4)]
        await asyncio.sleep(0.01)  # all four park
        assert len(q._getters) == 4
        await q.add("a")
        await asyncio.sleep(0.01)
        # one getter got the item; the other three are still parked
        assert len(results) == 1
        assert len(q._getters) == 3
        for item in ("b", "c", "d"):
            await q.add(item)
        await asyncio.sleep(0.01)
        assert len(results) == 4
        assert {r[1] for r in results} == {"a", "b", "c", "d"}
        for t in tasks:
            t.cancel()

    run(main())


def test_no_lost_wakeups_under_contention():
    """100 items through 8 workers with processing delays: every item is
    delivered exactly once (dedup) and nothing hangs."""

    async def main():
        q = RateLimitingQueue(fast_limiter())
        seen = []

        async def worker():
            while True:
                item = await q.get()
                if item is None:
                    return
                await asyncio.sleep(0)  # yield mid-processing
                seen.append(item)
                await q.done(item)

        workers = [asyncio.create_task(worker()) for _ in range(8)]
        for i in range(100):
            await q.add(i)
            if i % 7 == 0:
                await asyncio.sleep(0)
        # wait until drained
        for _ in range(1000):
            if len(seen) == 100:
                break
            await asyncio.sleep(0.005)
        assert sorted(seen) == list(range(100))
        await q.shutdown()
        await asyncio.gather(*workers)

    run(main())


def test_cancelled_getter_passes_wakeup_on():
    async def main():
        q = RateLimitingQueue(fast_limiter())
        got = []

        async def getter():
            got.append(await q.get())

        t1 = asyncio.create_task(getter())
        t2 = asyncio.create_task(getter())
        await asyncio.sleep(0.01)  # both park; t1 parked first
        await q.add("x")
        # cancel the getter that was woken before it runs: the wakeup must
        # pass to the other parked getter instead of vanishing
        t1.cancel()
        await asyncio.sleep(0.05)
        assert got == ["x"]
        t2.cancel()

    run(main())


def test_shutdown_wakes_all_parked_getters():
    async def main():
        q = RateLimitingQueue(fast_limiter())

        async def getter():
            return await q.get()

        tasks = [asyncio.create_task(getter()) for _ in range(5)]
        await asyncio.sleep(0.01)
        await q.shutdown()
        results = await asyncio.wait_for(asyncio.gather(*tasks), timeout=2)
        assert results == [None] * 5

    run(main())


def test_delayed_add_timer_wakes_parked_getter():
    async def main():
        q = RateLimitingQueue(fast_limiter())

        async def getter():
            return await q.get()

        t = asyncio.create_task(getter())
        await asyncio.sleep(0.01)  # park with no timeout
        await q.add_after("later", 0.03)
        item = await asyncio.wait_for(t, timeout=2)
        assert item == "later"

    run(main())


def test_shared_timer_rearms_to_earlier_deadline():
    """A delayed add with a SOONER deadline than the armed shared timer
    must re-arm it — otherwise the earlier item waits for the later
    deadline (lost-wakeup class introduced by the one-timer design)."""

    async def main():
        import time

        q = RateLimitingQueue()
        worker_got = []

        async def worker():
            item = await q.get()
            worker_got.append((item, time.monotonic()))

        t = asyncio.create_task(worker())
        await asyncio.sleep(0.01)  # park the getter (no items)
        await q.add_after("late", 5.0)   # timer armed at +5 s
        await q.add_after("soon", 0.05)  # must re-arm to +0.05 s
        t0 = time.monotonic()
        await asyncio.wait_for(t, 2.0)
        item, when = worker_got[0]
        assert item == "soon"
        assert when - t0 < 1.0, f"woke after {when - t0:.2f}s — timer not re-armed"
        await q.shutdown()

    run(main())


def test_shared_timer_chained_wakeups_drain_batch():
    """When one timer tick readies MANY delayed items, chained wakeups must
    hand one to each parked worker (no single-consumer bottleneck)."""

    async def main():
        q = RateLimitingQueue()
        got = []

        async def worker():
            while True:
                item = await q.get()
                if item is None:
                    return
                got.append(item)
                await q.done(item)

        workers = [asyncio.create_task(worker()) for _ in range(8)]
        await asyncio.sleep(0.01)
        for i in range(50):
            await q.add_after(f"i{i}", 0.05)  # all due at the same instant
        await asyncio.sleep(0.5)
        assert sorted(got) == sorted(f"i{i}" for i in range(50)), got
        await q.shutdown()
        await asyncio.gather(*workers)

    run(main())
