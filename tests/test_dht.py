import asyncio

from bee2bee_amd.mesh.dht import (
    DHTNode,
    announce_piece,
    announce_rank,
    find_providers,
    find_ranks,
)


def test_announce_find():
    async def run():
        dht = DHTNode()
        await dht.start()
        await announce_piece(dht, "hash1", "ws://a:1")
        await announce_piece(dht, "hash1", "ws://b:2")
        await announce_piece(dht, "hash1", "ws://a:1")  # dedup
        provs = await find_providers(dht, "hash1")
        assert provs == ["ws://a:1", "ws://b:2"]
        assert await find_providers(dht, "missing") == []

    asyncio.run(run())


def test_rccl_rendezvous_records():
    async def run():
        dht = DHTNode()
        await dht.start()
        await announce_rank(dht, "pp-llama70b", "peer-a", {"host": "10.0.0.1", "gpu": 0})
        await announce_rank(dht, "pp-llama70b", "peer-b", {"host": "10.0.0.1", "gpu": 1})
        ranks = await find_ranks(dht, "pp-llama70b")
        assert set(ranks) == {"peer-a", "peer-b"}
        assert ranks["peer-b"]["gpu"] == 1

    asyncio.run(run())
