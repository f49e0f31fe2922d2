"""ClusterManager: discovery scan, health/profile/latency fan-out, topology
solving (reference: src/dnet/api/cluster.py).
"""
from __future__ import annotations

import asyncio
from typing import Optional

import httpx

from ..core.types import TopologyInfo
from ..models import ModelConfig, PRESETS, QuantConfig
from ..parallel.profiler import DeviceProfile
from ..utils.hostfile import DeviceProperties
from ..utils.logger import get_logger

log = get_logger("api")


def estimate_layer_bytes(cfg: ModelConfig) -> int:
    """Weight bytes per transformer layer for the configured quantization."""
    c = cfg
    per_elem = 1.0 if c.quant else 2.0
    n_mlp = 2 * c.intermediate_size * c.hidden_size * 1.5  # gate+up+down
    if c.num_experts:
        inter = c.moe_intermediate_size or c.intermediate_size
        n_mlp = c.num_experts * 3 * inter * c.hidden_size
    n_attn = (c.qkv_out + c.num_q_heads * c.head_dim) * c.hidden_size
    n = n_attn + n_mlp
    scale_overhead = 1.02 if c.quant else 1.0
    return int(n * per_elem * scale_overhead)


class ClusterManager:
    def __init__(self, discovery, solver_settings=None, strategy=None):
        from .strategies import ring_strategy
        self.strategy = strategy or ring_strategy()
        self.discovery = discovery
        self.devices: dict[str, DeviceProperties] = {}
        self.profiles: dict[str, DeviceProfile] = {}
        self.link_ms: dict[tuple, float] = {}   # (src, dst) -> median ms
        self.excluded: set = set()               # failed shards (recovery)
        self.topology: Optional[TopologyInfo] = None

    async def scan_devices(self) -> dict[str, DeviceProperties]:
        self.devices = await self.discovery.async_get_properties()
        return self.devices

    def shard_devices(self) -> list[DeviceProperties]:
        return [d for d in self.devices.values()
                if not d.is_manager and d.instance not in self.excluded]

    async def profile_cluster(self, parallel: bool = True) -> dict:
        """Health-check then /profile each shard; merge latency medians."""
        await self.scan_devices()
        shards = self.shard_devices()
        async with httpx.AsyncClient(timeout=120.0) as client:
            healthy = []
            ranks: dict[int, str] = {}
            xgmi: dict = {}
            for d in shards:
                try:
                    r = await client.get(
                        f"http://{d.local_ip}:{d.server_port}/health")
                    if r.status_code == 200:
                        healthy.append(d)
                        h = r.json()
                        if h.get("rank", -1) >= 0:
                            ranks[int(h["rank"])] = d.instance
                        if h.get("xgmi"):
                            xgmi = h["xgmi"]
                except httpx.HTTPError:
                    log.warning("shard %s unreachable", d.instance)
            results = await asyncio.gather(*[
                client.post(f"http://{d.local_ip}:{d.server_port}/profile")
                for d in healthy], return_exceptions=True)
            for d, r in zip(healthy, results):
                if isinstance(r, Exception):
                    continue
                self.profiles[d.instance] = DeviceProfile.from_dict(r.json())
            # latency sweep: every shard probes every other shard so the
            # solver sees the full link matrix (ring ordering follows it)
            if len(healthy) > 1:
                peers = [{"instance": d.instance, "host": d.local_ip,
                          "port": d.shard_port} for d in healthy]
                sweeps = await asyncio.gather(*[
                    client.post(
                        f"http://{d.local_ip}:{d.server_port}"
                        "/measure_latency",
                        json={"peers": [p for p in peers
                                        if p["instance"] != d.instance],
                              "payload_sizes": [65536], "reps": 5})
                    for d in healthy], return_exceptions=True)
                for src, r in zip(healthy, sweeps):
                    if isinstance(r, Exception):
                        continue
                    lat = r.json().get("latencies", {})
                    for inst, sizes in lat.items():
                        med = next(iter(sizes.values()), {}).get("median_ms")
                        if med is None:
                            continue
                        self.link_ms[(src.instance, inst)] = med
                        if inst in self.profiles:
                            self.profiles[inst].t_comm_ms = med
            # measured xGMI fabric map (collected in-group at the previous
            # load, surfaced via /health) overrides the TCP RTT numbers:
            # ring ordering then follows the actual per-link fabric
            # 
            for key, r in xgmi.items():
                try:
                    i, j = (int(x) for x in key.split("-"))
                except ValueError:
                    continue
                a, b = ranks.get(i), ranks.get(j)
                if a and b and "latency_ms" in r:
                    self.link_ms[(a, b)] = r["latency_ms"]
        return {k: v.to_dict() for k, v in self.profiles.items()}

    async def healthy_shards(self) -> list[DeviceProperties]:
        """Quick /health sweep (reference: cluster.py health filtering) —
        used by failure recovery to re-solve over the survivors."""
        await self.scan_devices()
        out = []
        async with httpx.AsyncClient(timeout=5.0) as client:
            for d in [d for d in self.devices.values() if not d.is_manager]:
                try:
                    r = await client.get(
                        f"http://{d.local_ip}:{d.server_port}/health")
                    if r.status_code == 200:
                        out.append(d)
                except httpx.HTTPError:
                    pass
        return out

    def get_head_node(self) -> Optional[DeviceProperties]:
        """Owner of layer 0 (reference: cluster.py get_head_node)."""
        if not self.topology or not self.topology.assignments:
            return None
        for a in self.topology.assignments:
            if a.layers and a.layers[0] and a.layers[0][0] == 0:
                return self.devices.get(a.instance)
        return self.devices.get(self.topology.assignments[0].instance)

    def solve_topology(self, model_id: str, cfg: ModelConfig,
                       master_port: int = 29500, kv_bits: int = 16,
                       batch: int = 1, seq_len: int = 4096) -> TopologyInfo:
        shards = self.shard_devices()
        assert shards, "no shard devices discovered"
        profiles = {}
        for d in shards:
            p = self.profiles.get(d.instance)
            if p is None:
                p = DeviceProfile(instance=d.instance, hbm_gbps=5000.0,
                                  h2d_gbps=50.0, hbm_free_gb=280.0)
            profiles[d.instance] = p
        lb = estimate_layer_bytes(cfg)
        kv_per_layer = (2 * batch * cfg.num_kv_heads * seq_len * cfg.head_dim
                        * (kv_bits / 8))
        topo = self.strategy.solver.solve(
            model_id, cfg, shards, profiles, self.link_ms, lb, kv_per_layer,
            master_port, kv_bits)
        self.topology = topo
        return topo
