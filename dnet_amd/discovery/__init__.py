"""UDP multicast peer discovery (native C++ daemon + asyncio wrapper).

MI355X counterpart of the reference's Rust dnet-p2p library (reference:
lib/dnet-p2p, consumed via AsyncDnetP2P in src/cli/api.py:58-60 and
src/dnet/api/cluster.py:32-36): shards and the API announce themselves
over UDP multicast and discover each other with no hostfile. The wrapper
below exposes the same async surface as ``StaticDiscovery``
(``async_start`` / ``async_get_properties`` / ``async_stop`` /
``async_set_is_busy``) so ``ClusterManager`` takes either interchangeably.
"""
from __future__ import annotations

import asyncio
from typing import Optional

from ..utils.hostfile import DeviceProperties


def _load_ext():
    try:
        from . import _p2p  # type: ignore
        return _p2p
    except ImportError:
        from .build import build
        build()
        from . import _p2p  # type: ignore
        return _p2p


class UdpDiscovery:
    """One announcing instance on the discovery multicast group."""

    def __init__(self, instance: str, http_port: int, shard_port: int,
                 is_manager: bool = False, gpu_index: int = -1,
                 group: str = "239.192.31.41", port: int = 52525,
                 interval_s: float = 0.5, expire_s: float = 5.0):
        ext = _load_ext()
        self._inst = ext.P2PInstance(
            instance=instance, http_port=http_port, shard_port=shard_port,
            is_manager=is_manager, gpu_index=gpu_index, group=group,
            port=port, interval_s=interval_s, expire_s=expire_s)
        self.instance = instance

    @property
    def local_ip(self) -> str:
        return self._inst.local_ip

    async def async_start(self):
        await asyncio.get_event_loop().run_in_executor(None, self._inst.start)

    async def async_stop(self):
        await asyncio.get_event_loop().run_in_executor(None, self._inst.stop)

    async def async_set_is_busy(self, busy: bool):
        self._inst.set_is_busy(busy)

    async def async_get_properties(self) -> dict[str, DeviceProperties]:
        raw = self._inst.get_properties()
        return {name: DeviceProperties(
                    instance=p["instance"], local_ip=p["local_ip"],
                    server_port=p["server_port"], shard_port=p["shard_port"],
                    is_manager=p["is_manager"], is_busy=p["is_busy"],
                    gpu_index=p["gpu_index"])
                for name, p in raw.items()}

    async def wait_for_peers(self, n: int, timeout_s: float = 10.0) -> bool:
        """Poll until at least ``n`` peers (beyond self) are visible."""
        deadline = asyncio.get_event_loop().time() + timeout_s
        while asyncio.get_event_loop().time() < deadline:
            props = await self.async_get_properties()
            if len(props) - 1 >= n:
                return True
            await asyncio.sleep(0.1)
        return False
