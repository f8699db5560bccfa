"""Informer cache coherence under concurrent writes: N writers mutate the
apiserver (create/update/patch/delete, random but seeded) while an informer
watches — including forced watch-stream drops and history-overflow relists.
At quiesce the cache, its indexes, and every handler-notified state must
match the store exactly. This is the machinery every controller's view of
the world stands on."""
import asyncio
import random

from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.kube import objects as ko
from gpu_provisioner_amd.kube.client import ConflictError, NotFoundError
from gpu_provisioner_amd.kube.informer import Informer, object_key
from tests.conftest import run


def make_obj(name: str, env: str) -> dict:
    return {
        "apiVersion": "v1",
        "kind": "Node",
        "metadata": {"name": name, "labels": {"env": env}},
        "spec": {"providerID": f"azure:///x/{name}"},
        "status": {},
    }


async def writer(kube, rng: random.Random, names: list, ops: int) -> None:
    for _ in range(ops):
        name = rng.choice(names)
        verb = rng.random()
        try:
            if verb < 0.35:
                await kube.create(make_obj(name, rng.choice("abc")))
            elif verb < 0.6:
                obj = await kube.get("v1", "Node", name)
                obj["metadata"]["labels"]["env"] = rng.choice("abc")
                await kube.update(obj)
            elif verb < 0.8:
                await kube.patch(
                    "v1", "Node", name,
                    {"metadata": {"labels": {"env": rng.choice("abc")}}},
                )
            else:
                await kube.delete("v1", "Node", name)
        except (NotFoundError, ConflictError, Exception):
            pass
        if rng.random() < 0.2:
            await asyncio.sleep(0)


def test_informer_matches_store_after_concurrent_chaos():
    async def main():
        rng = random.Random(424242)
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        inf = Informer(kube, "v1", "Node")
        inf.add_index("env", lambda o: ko.labels_of(o).get("env"))
        inf.start()
        await inf.wait_for_sync()

        names = [f"n{i:02d}" for i in range(20)]
        writers = [
            asyncio.create_task(writer(kube, random.Random(rng.random()), names, 120))
            for _ in range(6)
        ]

        async def dropper():
            for _ in range(8):
                await asyncio.sleep(0.01)
                server.break_watches()

        drop = asyncio.create_task(dropper())
        await asyncio.gather(*writers, drop)
        # quiesce: let the informer catch up (relist after drops)
        for _ in range(200):
            await asyncio.sleep(0.01)
            store = await kube.list("v1", "Node")
            if {object_key(o) for o in store} == set(inf._cache.keys()):
                cache_ok = all(
                    inf.get(ko.name_of(o))["metadata"]["resourceVersion"]
                    == o["metadata"]["resourceVersion"]
                    for o in store
                )
                if cache_ok:
                    break
        store = await kube.list("v1", "Node")
        assert {object_key(o) for o in store} == set(inf._cache.keys())
        for o in store:
            cached = inf.get(ko.name_of(o))
            assert cached["metadata"]["resourceVersion"] == o["metadata"]["resourceVersion"]
            assert ko.labels_of(cached) == ko.labels_of(o)
        # indexes agree with the cache
        for env in "abc":
            indexed = {ko.name_of(o) for o in inf.by_index("env", env)}
            expect = {
                ko.name_of(o) for o in store if ko.labels_of(o).get("env") == env
            }
            assert indexed == expect, f"env={env}"
        await inf.stop()

    run(main(), timeout=120)
