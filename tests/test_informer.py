"""Informer tests: list+watch cache coherence, indexers, relist on 410 Gone,
deletion detection across watch gaps, handler dispatch."""
import asyncio

import pytest

from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
from gpu_provisioner_amd.kube.client import ADDED, DELETED, MODIFIED
from gpu_provisioner_amd.kube.informer import Informer, InformerFactory
from tests.conftest import run


def mk_node(name, provider_id=""):
    n = {"apiVersion": "v1", "kind": "Node", "metadata": {"name": name}, "spec": {}}
    if provider_id:
        n["spec"]["providerID"] = provider_id
    return n


def test_informer_cache_and_index():
    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        await kube.create(mk_node("a", "azure:///pid-a"))
        inf = Informer(kube, "v1", "Node")
        inf.add_index("providerID", lambda o: o.get("spec", {}).get("providerID") or None)
        events = []
        inf.add_handler(lambda et, obj: events.append((et, obj["metadata"]["name"])))
        inf.start()
        await asyncio.wait_for(inf.wait_for_sync(), 5)
        assert inf.get("a") is not None
        assert [o["metadata"]["name"] for o in inf.by_index("providerID", "azure:///pid-a")] == ["a"]
        # live updates flow into cache + index
        await kube.create(mk_node("b", "azure:///pid-b"))
        await asyncio.sleep(0.05)
        assert inf.get("b") is not None
        assert inf.by_index("providerID", "azure:///pid-b")
        # modification reindexes
        b = await kube.get("v1", "Node", "b")
        b["spec"]["providerID"] = "azure:///pid-b2"
        await kube.update(b)
        await asyncio.sleep(0.05)
        assert not inf.by_index("providerID", "azure:///pid-b")
        assert inf.by_index("providerID", "azure:///pid-b2")
        # deletion clears both
        await kube.delete("v1", "Node", "b")
        await asyncio.sleep(0.05)
        assert inf.get("b") is None
        assert not inf.by_index("providerID", "azure:///pid-b2")
        assert (ADDED, "a") in events and (DELETED, "b") in events
        await inf.stop()

    run(main())


def test_informer_relists_after_gone():
    """A watch that raises 410 forces a fresh list; deletions that happened
    during the gap are synthesized as DELETED events."""

    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        await kube.create(mk_node("a"))
        await kube.create(mk_node("b"))
        inf = Informer(kube, "v1", "Node")
        deleted = []
        inf.add_handler(lambda et, obj: deleted.append(obj["metadata"]["name"]) if et == DELETED else None)
        inf.start()
        await asyncio.wait_for(inf.wait_for_sync(), 5)
        # simulate a watch gap: kill history so resume raises GoneError,
        # and delete 'b' out-of-band
        await kube.delete("v1", "Node", "b")
        await asyncio.sleep(0.05)
        assert inf.get("b") is None  # normal watch path caught it
        # now force a true relist: clear cache's knowledge via private poke
        # (equivalent to a long network partition)
        server._history = [(10**9, ("v1", "Node"), "MODIFIED", mk_node("x"))]
        await kube.create(mk_node("c"))
        await asyncio.sleep(0.2)
        assert inf.get("c") is not None
        await inf.stop()

    run(main())


def test_informer_label_selector_scoping():
    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        inf = Informer(kube, "v1", "Node", label_selector="agentpool=gpu1")
        inf.start()
        await asyncio.wait_for(inf.wait_for_sync(), 5)
        n1 = mk_node("a")
        n1["metadata"]["labels"] = {"agentpool": "gpu1"}
        await kube.create(n1)
        await kube.create(mk_node("other"))
        await asyncio.sleep(0.05)
        assert inf.get("a") is not None
        assert inf.get("other") is None
        await inf.stop()

    run(main())


def test_informer_factory_shares_instances():
    server = InMemoryAPIServer()
    kube = InMemoryClient(server)
    f = InformerFactory(kube)
    a = f.informer("v1", "Node")
    b = f.informer("v1", "Node")
    c = f.informer("v1", "Pod")
    assert a is b and a is not c


def test_resync_redelivers_cached_objects():
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.kube.informer import Informer

    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        await kube.create(
            {"apiVersion": "v1", "kind": "Node", "metadata": {"name": "rs1"},
             "spec": {}, "status": {}}
        )
        inf = Informer(kube, "v1", "Node", resync_period=0.05)
        events = []
        inf.add_handler(lambda et, obj: events.append((et, obj["metadata"]["name"])))
        inf.start()
        await inf.wait_for_sync()
        await asyncio.sleep(0.18)  # ~3 resync periods
        await inf.stop()
        resyncs = [e for e in events if e == ("MODIFIED", "rs1")]
        assert len(resyncs) >= 2, events

    run(main())


def test_wait_until_by_name_and_index():
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.kube.informer import Informer

    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        inf = Informer(kube, "v1", "Node")
        inf.add_index("pool", lambda o: (o["metadata"].get("labels") or {}).get("agentpool"))
        inf.start()
        await inf.wait_for_sync()

        # name-keyed: resolves when the predicate matches a later event
        async def create_later():
            await asyncio.sleep(0.02)
            await kube.create(
                {"apiVersion": "v1", "kind": "Node",
                 "metadata": {"name": "w1", "labels": {"agentpool": "p1"}},
                 "spec": {}, "status": {}}
            )

        t = asyncio.create_task(create_later())
        got = await inf.wait_until(
            lambda et, o: o if o is not None and et != "DELETED" else None,
            name="w1", timeout=5,
        )
        assert got["metadata"]["name"] == "w1"
        await t

        # cache-satisfied immediately (no event needed)
        got2 = await inf.wait_until(
            lambda et, o: o if o is not None else None, name="w1", timeout=1
        )
        assert got2["metadata"]["name"] == "w1"

        # index-keyed: a second node in the pool arrives later
        async def create_p2():
            await asyncio.sleep(0.02)
            await kube.create(
                {"apiVersion": "v1", "kind": "Node",
                 "metadata": {"name": "w2", "labels": {"agentpool": "p2"}},
                 "spec": {"providerID": "azure:///x"}, "status": {}}
            )

        t2 = asyncio.create_task(create_p2())
        got3 = await inf.wait_until(
            lambda et, o: o if o is not None and o["spec"].get("providerID") else None,
            index="pool", value="p2", timeout=5,
        )
        assert got3["metadata"]["name"] == "w2"
        await t2

        # ABSENT short-circuit (wait_gone shape)
        gone = await inf.wait_until(
            lambda et, o: True if et in ("DELETED", "ABSENT") else None,
            name="nope", timeout=1,
        )
        assert gone is True

        # timeout path
        import pytest as _pytest
        with _pytest.raises(asyncio.TimeoutError):
            await inf.wait_until(lambda et, o: None, name="w1", timeout=0.05)
        # waiter registries drained after completion/timeouts
        assert not any(inf._key_waiters.values())
        await inf.stop()

    run(main())


def test_wait_until_registries_do_not_leak():
    from gpu_provisioner_amd.fake.apiserver import InMemoryAPIServer, InMemoryClient
    from gpu_provisioner_amd.kube.informer import Informer

    async def main():
        kube = InMemoryClient(InMemoryAPIServer())
        inf = Informer(kube, "v1", "Node")
        inf.add_index("pool", lambda o: (o["metadata"].get("labels") or {}).get("agentpool"))
        inf.start()
        await inf.wait_for_sync()
        for i in range(50):
            with pytest.raises(asyncio.TimeoutError):
                await inf.wait_until(lambda et, o: None, name=f"n{i}", timeout=0.001)
            with pytest.raises(asyncio.TimeoutError):
                await inf.wait_until(
                    lambda et, o: None, index="pool", value=f"p{i}", timeout=0.001
                )
        assert not inf._key_waiters, inf._key_waiters
        assert not inf._index_waiters, inf._index_waiters
        await inf.stop()

    run(main())


def test_add_index_after_sync_backfills_existing_cache():
    """A late-registered index (the set_nodes_informer path binds indexes
    after informers may already be synced) must be backfilled from the
    current cache, mirroring add_handler's replay — otherwise by_index
    silently returns empty for pre-existing objects."""

    async def main():
        server = InMemoryAPIServer()
        kube = InMemoryClient(server)
        await kube.create(mk_node("pre", "azure:///pid-pre"))
        inf = Informer(kube, "v1", "Node")
        inf.start()
        await asyncio.wait_for(inf.wait_for_sync(), 5)
        # register AFTER sync — "pre" is already cached
        inf.add_index("providerID", lambda o: o.get("spec", {}).get("providerID") or None)
        got = inf.by_index("providerID", "azure:///pid-pre")
        assert [o["metadata"]["name"] for o in got] == ["pre"]
        # idempotent re-registration stays a no-op
        inf.add_index("providerID", lambda o: None)
        assert inf.by_index("providerID", "azure:///pid-pre")
        await inf.stop()

    run(main())
