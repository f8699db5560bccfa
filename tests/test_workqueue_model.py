"""Model-checking stress for RateLimitingQueue: random concurrent
add/add_after/fail/requeue traffic against an oracle. THE invariant (the
class of bug the chaos suite caught in the reconcile aggregator): any key
whose last interaction promised more work — an add, a failure retry, a
requeue — must eventually be processed again. At quiesce, no key may have
outstanding work, and dedup must never have dropped a distinct generation."""
import asyncio
import random

from gpu_provisioner_amd.kube.workqueue import RateLimiter, RateLimitingQueue
from tests.conftest import run

KEYS = [f"k{i}" for i in range(8)]


def test_no_lost_work_randomized():
    async def one_seed(seed: int) -> None:
        rng = random.Random(seed)
        q = RateLimitingQueue(RateLimiter(base=0.001, cap=0.01, qps=1e6, burst=10**6))
        # oracle: generation counter per key; processed_gen records the
        # latest generation observed by a completed processing
        gen = {k: 0 for k in KEYS}
        processed_gen = {k: 0 for k in KEYS}
        stop = False

        async def worker():
            while True:
                item = await q.get()
                if item is None:
                    return
                my_gen = gen[item]
                # random processing behavior
                r = rng.random()
                if r < 0.3:
                    await asyncio.sleep(0)  # yield mid-processing
                processed_gen[item] = max(processed_gen[item], my_gen)
                if r < 0.15 and not stop:
                    # "failure": controller pattern — done, then retry later
                    await q.done(item)
                    gen[item] += 1  # retry promises more work
                    await q.add_rate_limited(item)
                elif r < 0.25 and not stop:
                    # "requeue_after": promised future work
                    await q.done(item)
                    gen[item] += 1
                    await q.add_after(item, rng.random() * 0.004)
                else:
                    await q.done(item)

        workers = [asyncio.create_task(worker()) for _ in range(6)]
        # external traffic: events and delayed adds
        for _ in range(150):
            k = rng.choice(KEYS)
            gen[k] += 1
            if rng.random() < 0.3:
                await q.add_after(k, rng.random() * 0.005)
            else:
                q.add_nowait(k)
            if rng.random() < 0.3:
                await asyncio.sleep(0)
        stop = True
        # quiesce: every promised generation must get processed
        for _ in range(2000):
            if all(processed_gen[k] >= gen[k] for k in KEYS):
                break
            await asyncio.sleep(0.005)
        lost = {k: (gen[k], processed_gen[k]) for k in KEYS if processed_gen[k] < gen[k]}
        assert not lost, f"seed={seed}: lost work {lost} (q depth={q.depth}, delayed={len(q._delayed)}, dirty={q._dirty}, processing={q._processing})"
        await q.shutdown()
        await asyncio.gather(*workers)

    async def main():
        for seed in range(40):
            await one_seed(seed)

    run(main(), timeout=300)
