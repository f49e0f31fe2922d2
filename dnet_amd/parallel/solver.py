"""Heterogeneity-aware layer-assignment solver (HALDA-style w/n/k).

MI355X re-derivation of the reference's MILP layer distribution
(reference: lib/distilp halda_solve -> HALDAResult{w, n, k, obj_value, sets};
consumed in src/dnet/api/strategies/ring.py:59-65): per-device layer counts
``w``, GPU-resident counts ``n`` and ring-round count ``k``, fed by the
MI355X profiler's numbers (HBM bandwidth/capacity, host-DRAM H2D bandwidth,
xGMI hop latency) instead of UMA/disk/Thunderbolt.

Decode is weight-read bound, so per-layer cost on device i is
    c_i = layer_bytes/hbm_bw_i              (resident layer)
    c_i' = layer_bytes/h2d_bw_i             (host-swapped layer; copy-stream
                                             overlap hides compute)
The makespan objective min max_i t_i (bottleneck stage = pipelined decode
throughput) is solved EXACTLY by a convex-cost DP over (device, count),
with total time (single-stream latency) as the lexicographic tiebreak;
residency n_i follows from HBM capacity with swapped layers priced at
c_i'. k>1 (multiple ring rounds) is chosen so each round's window fits
residency.
"""
from __future__ import annotations


from dataclasses import dataclass, field

from .profiler import DeviceProfile


@dataclass
class SolveResult:
    w: list            # layers per device
    n: list            # GPU-resident layers per device
    k: int             # ring rounds
    obj_value: float   # estimated per-token latency (ms)
    sets: dict = field(default_factory=dict)   # device classes M1/M2/M3


def halda_solve(devices: list[DeviceProfile], num_layers: int,
                layer_bytes: float, kv_bytes_per_layer: float = 0.0,
                overhead_gb: float = 4.0, hop_ms: float = 0.02,
                kv_bits: int = 16) -> SolveResult:
    """Assign ``num_layers`` identical layers across ``devices``.

    layer_bytes: weight bytes per layer (after quantization).
    kv_bytes_per_layer: KV cache bytes per layer for the planned batch/seq.
    """
    nd = len(devices)
    assert nd > 0
    c_res = []      # ms per resident layer
    c_swap = []     # ms per host-swapped layer
    cap_layers = []
    for d in devices:
        bw = max(d.hbm_gbps, 1.0) * 1e9
        c_res.append(layer_bytes / bw * 1e3)
        c_swap.append(layer_bytes / (max(d.h2d_gbps, 0.5) * 1e9) * 1e3)
        usable = max(d.hbm_free_gb - overhead_gb, 0.5) * 1e9
        cap_layers.append(max(int(usable // max(layer_bytes + kv_bytes_per_layer, 1)), 1))

    def dev_time(i: int, count: int) -> float:
        """Per-token time if device i holds `count` layers (resident up
        to capacity, host-swapped beyond)."""
        resident = min(count, cap_layers[i])
        return resident * c_res[i] + (count - resident) * c_swap[i]

    # Exact min-makespan DP over (device, layer-count) — O(nd * L^2),
    # trivially cheap for <=16 devices x <=128 layers. Minimizes the
    # bottleneck stage time (pipelined decode throughput), with total
    # time (single-stream latency) as the lexicographic tiebreak.
    # (the round-1 greedy list scheduler was exact
    # only for identical layers on a homogeneous node; this is optimal
    # for any profile mix, asserted vs brute force in tests.)
    L = num_layers
    INF = float("inf")
    f = [(INF, INF)] * (L + 1)
    f[0] = (0.0, 0.0)
    choice = [[0] * (L + 1) for _ in range(nd)]
    for i in range(nd):
        g = [(INF, INF)] * (L + 1)
        for l in range(L + 1):
            best = (INF, INF)
            bx = 0
            for x in range(l + 1):
                prev = f[l - x]
                if prev[0] == INF:
                    continue
                t = dev_time(i, x)
                cand = (max(prev[0], t), prev[1] + t)
                if cand < best:
                    best = cand
                    bx = x
            g[l] = best
            choice[i][l] = bx
        f = g
    w = [0] * nd
    rem = L
    for i in range(nd - 1, -1, -1):
        w[i] = choice[i][rem]
        rem -= w[i]

    n = [min(w[i], cap_layers[i]) for i in range(nd)]
    # k rounds: if any device swaps, use enough rounds that one round's
    # window fits residency (reference: k-round interleaving per prima.cpp).
    k = 1
    over = [w[i] / max(n[i], 1) for i in range(nd)]
    if max(over) > 1.0:
        k = min(int(max(over).__ceil__()), 8)

    t_devices = [n[i] * c_res[i] + (w[i] - n[i]) * c_swap[i] for i in range(nd)]
    obj = sum(t_devices) + hop_ms * nd * k
    sets = {
        "M1": [devices[i].instance for i in range(nd) if w[i] == n[i]],
        "M2": [devices[i].instance for i in range(nd)
               if n[i] < w[i] <= 2 * n[i]],
        "M3": [devices[i].instance for i in range(nd) if w[i] > 2 * n[i]],
    }
    return SolveResult(w=w, n=n, k=k, obj_value=obj, sets=sets)


def postprocess_single_round(w: list, devices: list[DeviceProfile]) -> list:
    """Fold single-layer devices into their lighter neighbor when k == 1
    (reference: src/dnet/api/utils.py postprocess_single_round)."""
    if len(w) <= 1 or sum(1 for x in w if x > 0) <= 1:
        return w
    w = list(w)
    while True:
        try:
            i = next(i for i, x in enumerate(w) if 0 < x <= 1 and len([y for y in w if y > 0]) > 1)
        except StopIteration:
            return w
        others = [(w[j], j) for j in range(len(w)) if j != i and w[j] > 0]
        if not others:
            return w
        _, j = min(others)
        w[j] += w[i]
        w[i] = 0


def optimize_device_ordering(instances: list, link_ms: dict) -> list:
    """Greedy ring ordering: chain devices so consecutive hops ride the
    fastest measured links (reference: src/dnet/api/utils.py
    optimize_device_ordering — greedy Thunderbolt adjacency; here the edge
    weights are median link latencies from the /measure_latency sweep, so
    on an MI355X node the ordering follows the xGMI/TCP fabric instead of
    Thunderbolt cables).

    ``link_ms`` maps (src, dst) -> median ms; missing pairs are treated as
    equally slow so an empty map preserves the input order. The head stays
    first (layer-0 affinity)."""
    if len(instances) <= 2 or not link_ms:
        return list(instances)

    def cost(a: str, b: str) -> float:
        v = link_ms.get((a, b), link_ms.get((b, a)))
        return float(v) if v is not None else 1e9

    order = [instances[0]]
    rest = list(instances[1:])
    while rest:
        cur = order[-1]
        # stable: ties keep input order
        nxt = min(range(len(rest)), key=lambda j: (cost(cur, rest[j]), j))
        order.append(rest.pop(nxt))
    return order


def compute_layer_assignments(w: list, k: int, num_layers: int) -> list:
    """Round-robin k rounds of w[i] layers per device -> per-device list of
    per-round layer lists (reference: src/dnet/api/utils.py
    compute_layer_assignments)."""
    nd = len(w)
    per_round = [[max(w[i] // k, 0) for i in range(nd)] for _ in range(k)]
    # distribute remainders into the earliest rounds
    for i in range(nd):
        rem = w[i] - sum(pr[i] for pr in per_round)
        for r in range(rem):
            per_round[r % k][i] += 1
    out = [[] for _ in range(nd)]
    nxt = 0
    for r in range(k):
        for i in range(nd):
            cnt = per_round[r][i]
            out[i].append(list(range(nxt, min(nxt + cnt, num_layers))))
            nxt += cnt
    assert nxt == num_layers, f"assigned {nxt} != {num_layers}"
    return out
