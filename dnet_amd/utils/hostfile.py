"""Static discovery: hostfile parsing + a discovery object with the same
surface as the reference's StaticDiscovery / AsyncDnetP2P properties map
(reference: lib/dnet-p2p StaticDiscovery + DnetDeviceProperties; hostfile
formats per tests/test_static_discovery.py there).

Formats:
  SSH-style lines:  "<name> <ip> <http_port> <grpc_port>"  (# comments)
  JSON:             [{"name": ..., "ip": ..., "http_port": ..., "grpc_port": ...}]
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional


@dataclass
class DeviceProperties:
    instance: str
    local_ip: str
    server_port: int          # HTTP control port
    shard_port: int           # data-plane port
    is_manager: bool = False
    is_busy: bool = False
    # MI355X extras (replaces the reference's thunderbolt field)
    gpu_index: int = -1
    xgmi_links: dict = field(default_factory=dict)   # peer instance -> GB/s


def load_hostfile(path: str) -> list[DeviceProperties]:
    text = Path(path).expanduser().read_text()
    stripped = text.lstrip()
    devices: list[DeviceProperties] = []
    if stripped.startswith("[") or stripped.startswith("{"):
        data = json.loads(text)
        if isinstance(data, dict):
            data = data.get("devices", [])
        for i, e in enumerate(data):
            devices.append(DeviceProperties(
                instance=e.get("name", e.get("instance", f"shard{i}")),
                local_ip=e.get("ip", e.get("local_ip", "127.0.0.1")),
                server_port=int(e.get("http_port", e.get("server_port", 8081))),
                shard_port=int(e.get("grpc_port", e.get("shard_port", 50052))),
                gpu_index=int(e.get("gpu", i))))
        return devices
    for line in text.splitlines():
        line = line.split("#", 1)[0].strip()
        if not line:
            continue
        parts = line.split()
        if len(parts) < 4:
            raise ValueError(f"bad hostfile line: {line!r}")
        devices.append(DeviceProperties(
            instance=parts[0], local_ip=parts[1], server_port=int(parts[2]),
            shard_port=int(parts[3]),
            gpu_index=int(parts[4]) if len(parts) > 4 else len(devices)))
    return devices


class StaticDiscovery:
    """Hostfile-backed discovery with the p2p-discovery object surface."""

    def __init__(self, hostfile: str, own_instance: str = "",
                 own_http_port: int = 0, own_grpc_port: int = 0,
                 own_ip: str = "127.0.0.1", own_is_manager: bool = False):
        self.devices = load_hostfile(hostfile)
        self.own_instance = own_instance
        if own_instance and not any(d.instance == own_instance
                                    for d in self.devices):
            self.devices.append(DeviceProperties(
                instance=own_instance, local_ip=own_ip,
                server_port=own_http_port, shard_port=own_grpc_port,
                is_manager=own_is_manager))
        else:
            for d in self.devices:
                if d.instance == own_instance:
                    d.is_manager = own_is_manager
        self._running = False

    async def async_start(self):
        self._running = True

    async def async_stop(self):
        self._running = False

    def is_running(self) -> bool:
        return self._running

    async def async_get_properties(self) -> dict[str, DeviceProperties]:
        return {d.instance: d for d in self.devices}

    async def async_get_own_properties(self) -> Optional[DeviceProperties]:
        for d in self.devices:
            if d.instance == self.own_instance:
                return d
        return None

    @property
    def instance_name(self) -> str:
        return self.own_instance
