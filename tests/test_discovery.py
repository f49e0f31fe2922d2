"""Native UDP multicast discovery: two instances on loopback find each
other (reference tier: lib/dnet-p2p discovery tests)."""
import asyncio
import socket

import pytest

from dnet_amd.discovery import UdpDiscovery


def _free_udp_port():
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_udp_discovery_two_instances():
    port = _free_udp_port()

    async def run():
        api = UdpDiscovery("api", 8080, 50051, is_manager=True, port=port,
                           interval_s=0.1)
        shard = UdpDiscovery("shard0", 8081, 50052, gpu_index=0, port=port,
                             interval_s=0.1)
        await api.async_start()
        await shard.async_start()
        try:
            assert await api.wait_for_peers(1, timeout_s=10.0)
            assert await shard.wait_for_peers(1, timeout_s=10.0)
            props = await api.async_get_properties()
            assert set(props) == {"api", "shard0"}
            assert props["api"].is_manager and not props["shard0"].is_manager
            assert props["shard0"].shard_port == 50052
            assert props["shard0"].gpu_index == 0
            # busy flag propagates in later announces
            await shard.async_set_is_busy(True)
            for _ in range(50):
                props = await api.async_get_properties()
                if props["shard0"].is_busy:
                    break
                await asyncio.sleep(0.1)
            assert props["shard0"].is_busy
        finally:
            await api.async_stop()
            await shard.async_stop()

    asyncio.run(run())


def test_udp_discovery_expiry():
    port = _free_udp_port()

    async def run():
        a = UdpDiscovery("a", 1, 2, port=port, interval_s=0.1, expire_s=0.6)
        b = UdpDiscovery("b", 3, 4, port=port, interval_s=0.1, expire_s=0.6)
        await a.async_start()
        await b.async_start()
        try:
            assert await a.wait_for_peers(1, timeout_s=10.0)
        finally:
            await b.async_stop()
        # b stops announcing -> expires from a's map
        for _ in range(40):
            props = await a.async_get_properties()
            if "b" not in props:
                break
            await asyncio.sleep(0.1)
        await a.async_stop()
        assert "b" not in props

    asyncio.run(run())


def test_cluster_manager_over_udp_discovery():
    """ClusterManager consumes UdpDiscovery exactly like StaticDiscovery."""
    from dnet_amd.api.cluster import ClusterManager
    port = _free_udp_port()

    async def run():
        api = UdpDiscovery("api", 8080, 50051, is_manager=True, port=port,
                           interval_s=0.1)
        shard = UdpDiscovery("shard0", 8081, 50052, port=port, interval_s=0.1)
        await api.async_start()
        await shard.async_start()
        try:
            assert await api.wait_for_peers(1, timeout_s=10.0)
            cm = ClusterManager(api)
            devs = await cm.scan_devices()
            assert "shard0" in devs
            assert [d.instance for d in cm.shard_devices()] == ["shard0"]
        finally:
            await api.async_stop()
            await shard.async_stop()

    asyncio.run(run())
