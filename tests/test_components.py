"""Unit tests: config tree, hostfile discovery, solver, wire protocol,
tokenizer/detokenizer, repack, metadata parsing."""
import asyncio
import json
import os

import pytest
import torch

from dnet_amd.config import get_settings, reset_settings
from dnet_amd.parallel.profiler import DeviceProfile
from dnet_amd.parallel.solver import (compute_layer_assignments, halda_solve,
                                      postprocess_single_round)
from dnet_amd.utils.hostfile import StaticDiscovery, load_hostfile


def test_settings_env_override(monkeypatch):
    monkeypatch.setenv("DNET_API_PORT", "9999")
    monkeypatch.setenv("DNET_TRANSPORT_COMPRESS", "true")
    reset_settings()
    s = get_settings()
    assert s.api.port == 9999
    assert s.transport.compress is True
    monkeypatch.delenv("DNET_API_PORT")
    monkeypatch.delenv("DNET_TRANSPORT_COMPRESS")
    reset_settings()


def test_hostfile_ssh_style(tmp_path):
    hf = tmp_path / "hosts"
    hf.write_text("# comment\nshard0 127.0.0.1 8081 50052 0\n"
                  "shard1 127.0.0.1 8181 50152 1\n")
    devs = load_hostfile(str(hf))
    assert [d.instance for d in devs] == ["shard0", "shard1"]
    assert devs[1].gpu_index == 1


def test_hostfile_json(tmp_path):
    hf = tmp_path / "hosts.json"
    hf.write_text(json.dumps([
        {"name": "a", "ip": "10.0.0.1", "http_port": 1, "grpc_port": 2}]))
    devs = load_hostfile(str(hf))
    assert devs[0].local_ip == "10.0.0.1"


def test_static_discovery(tmp_path):
    hf = tmp_path / "hosts"
    hf.write_text("shard0 127.0.0.1 8081 50052\n")
    d = StaticDiscovery(str(hf), own_instance="api", own_http_port=8080,
                        own_grpc_port=50051)
    props = asyncio.run(d.async_get_properties())
    assert set(props) == {"shard0", "api"}


def _profiles(n, bw=None):
    bw = bw or [5000.0] * n
    return [DeviceProfile(instance=f"s{i}", hbm_gbps=bw[i], h2d_gbps=50.0,
                          hbm_free_gb=280.0) for i in range(n)]


def test_solver_homogeneous():
    res = halda_solve(_profiles(4), 64, 500e6)
    assert sorted(res.w) == [16, 16, 16, 16]
    assert res.k == 1 and res.n == res.w


def test_solver_heterogeneous():
    res = halda_solve(_profiles(2, bw=[6000.0, 2000.0]), 64, 500e6)
    assert res.w[0] > res.w[1]
    assert sum(res.w) == 64


def test_solver_capacity_offload():
    profs = _profiles(1)
    profs[0].hbm_free_gb = 10.0   # fits ~12 layers of 0.5 GB after overhead
    res = halda_solve(profs, 64, 500e6)
    assert res.n[0] < res.w[0] == 64
    assert res.k > 1
    assert profs[0].instance in res.sets["M3"]


def test_postprocess_single_round():
    profs = _profiles(3)
    assert postprocess_single_round([10, 1, 9], profs) == [10, 0, 10]


def test_compute_layer_assignments():
    out = compute_layer_assignments([3, 2], 1, 5)
    assert out == [[[0, 1, 2]], [[3, 4]]]
    out2 = compute_layer_assignments([2, 2], 2, 4)
    flat = [l for dev in out2 for r in dev for l in r]
    assert sorted(flat) == [0, 1, 2, 3]


def test_wire_roundtrip():
    from dnet_amd.protos.wire import WireClient, WireServer

    async def run():
        seen = []

        async def handler(frame, writer):
            seen.append(frame)
            return {"t": "pong", "echo": frame.get("x")}

        srv = WireServer("127.0.0.1", 29877, handler)
        await srv.start()
        cli = WireClient("127.0.0.1", 29877)
        resp = await cli.request({"t": "ping", "x": 42,
                                  "blob": b"\x00\x01" * 100})
        assert resp["echo"] == 42
        assert seen[0]["blob"] == b"\x00\x01" * 100
        await cli.close()
        await srv.stop()

    asyncio.run(run())


def test_byte_tokenizer_roundtrip():
    from dnet_amd.api.tokenizer import ByteTokenizer, Detokenizer
    t = ByteTokenizer(512)
    ids = t.encode("hello world")
    assert ids[0] == t.BOS
    assert t.decode(ids) == "hello world"
    d = Detokenizer(t)
    out = "".join(d.add_token(i) for i in ids)
    assert out == "hello world"


def test_metadata_and_repack(tmp_path, monkeypatch):
    from safetensors.torch import save_file

    from dnet_amd.utils.model_meta import get_model_metadata
    from dnet_amd.utils.repack import (delete_repacked_layers,
                                       ensure_repacked_for_layers)
    mdir = tmp_path / "model"
    mdir.mkdir()
    (mdir / "config.json").write_text(json.dumps({"model_type": "llama"}))
    sd = {
        "model.embed_tokens.weight": torch.randn(8, 4),
        "model.layers.0.self_attn.q_proj.weight": torch.randn(4, 4),
        "model.layers.1.self_attn.q_proj.weight": torch.randn(4, 4),
        "model.norm.weight": torch.randn(4),
    }
    save_file(sd, str(mdir / "model.safetensors"))
    meta = get_model_metadata(str(mdir))
    assert meta.num_layers == 2
    assert meta.layer_bytes(0) == 4 * 4 * 4
    assert meta.embed and meta.final_norm
    monkeypatch.setenv("DNET_STORAGE_REPACK_DIR", str(tmp_path / "repack"))
    reset_settings()
    out = ensure_repacked_for_layers(str(mdir), "m", [0])
    assert (out / "layer_0000.safetensors").exists()
    assert (out / "repack-manifest.json").exists()
    # idempotent
    out2 = ensure_repacked_for_layers(str(mdir), "m", [0])
    assert out2 == out
    assert delete_repacked_layers("m") == 1
    reset_settings()


def test_estimate_layer_bytes():
    from dnet_amd.api.cluster import estimate_layer_bytes
    from dnet_amd.models import ModelConfig, PRESETS, QuantConfig
    cfg = ModelConfig.from_hf(dict(PRESETS["qwen-2.5-32b"]),
                              quant=QuantConfig(8, 128))
    b = estimate_layer_bytes(cfg)
    assert 400e6 < b < 600e6  # ~487 MB per layer int8


def test_offload_policy_matches_fit():
    """Offload (windowed weight cache) must produce the same tokens as the
    fully-resident path for identical weights."""
    from dnet_amd.models import ModelConfig, PRESETS
    from dnet_amd.parallel.ring import RingExecutor
    from dnet_amd.shard.policies import plan_policy

    assert plan_policy(4, 2, 4) == "fit"
    assert plan_policy(8, 2, 4) == "offload"
    assert plan_policy(8, 4, 2) == "sliding_fit"

    cfg = ModelConfig.from_hf(dict(PRESETS["tiny"]))
    toks = torch.randint(0, cfg.vocab_size, (1, 2, 6),
                         generator=torch.Generator().manual_seed(3))

    def run(residency):
        ex = RingExecutor(cfg, 0, 1, "cpu", mb_count=1, mb_size=2, smax=32,
                          seed=11, use_graphs=False, residency=residency)
        first = ex.prefill(toks.clone())
        gen = ex.decode_rounds(4)
        return torch.cat([first.unsqueeze(-1), gen], dim=-1)

    fit = run(0)
    off = run(2)
    assert torch.equal(fit, off)


def test_compression_roundtrip():
    from dnet_amd import compression as cz
    torch.manual_seed(0)
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    x[:, :8] *= 10  # dominant columns must survive
    idx, packed = cz.column_sparsify(x, 0.5)
    assert idx.numel() == 32 and packed.shape == (4, 32)
    assert set(range(8)) <= set(idx.tolist())
    y = cz.column_unsparsify(packed, idx, 64)
    assert torch.equal(y.index_select(-1, idx.long()), packed)
    # wire blob roundtrip
    blob = cz.compress_tensor_to_bytes(x, 0.5)
    y2 = cz.decompress_tensor_from_bytes(blob)
    assert torch.equal(y, y2)
    assert len(blob) < x.numel() * 2  # actually smaller
    assert cz.is_compressed_dtype(cz.dtype_string("bfloat16", 0.5))


def test_memory_pool():
    from dnet_amd.core.memory_pool import LayerAwareMemoryPool
    pool = LayerAwareMemoryPool(max_bytes=1024)
    a = pool.acquire((8, 8), torch.bfloat16)
    pool.release(a)
    b = pool.acquire_for_layer(3, (8, 8), torch.bfloat16)
    assert b.data_ptr() == a.data_ptr()      # reused
    assert pool.hits == 1 and pool.misses == 1
    assert pool.layer_stats[3]["count"] == 1
    # budget eviction: release more than max_bytes
    for _ in range(12):
        pool.release(torch.empty(8, 8, dtype=torch.bfloat16))
    assert pool.free_bytes <= 1024


def test_serialization_roundtrip():
    from dnet_amd.utils.serialization import bytes_to_tensor, tensor_to_bytes
    for dtype in (torch.bfloat16, torch.float32, torch.int32):
        t = (torch.randn(3, 5) * 10).to(dtype)
        data, name, shape = tensor_to_bytes(t)
        t2 = bytes_to_tensor(data, name, shape)
        assert torch.equal(t, t2), dtype


def test_weight_cache_prefetch_no_thrash():
    """Sequential ring access with prefetch depth 3 must make every bind a
    hit after warmup (the LRU counts prefetches as uses — otherwise each
    prefetch evicts the previous one and every bind becomes a sync miss)."""
    from dnet_amd.core.weight_cache import PinnedLayerStore, WeightCache
    store = PinnedLayerStore(pin=False)
    for lid in range(24):
        store.put_layer(lid, {"w": torch.full((4, 4), float(lid))})
    order = list(range(24))
    cache = WeightCache(store, residency=8, device=torch.device("cpu"),
                        order=order)
    for lid in order[:8]:
        cache.prefetch(lid)
    for epoch in range(2):
        for i, lid in enumerate(order):
            t = cache.bind(lid)["w"]
            assert float(t[0, 0]) == lid
            for d in range(1, 4):
                cache.prefetch(order[(i + d) % 24])
    assert cache.misses == 0, f"bind misses: {cache.misses}"


def test_optimize_device_ordering():
    from dnet_amd.parallel.solver import optimize_device_ordering
    inst = ["a", "b", "c", "d"]
    # no link info: order preserved
    assert optimize_device_ordering(inst, {}) == inst
    # a-c and c-b are the fast links; d is far from everyone
    links = {("a", "c"): 0.1, ("c", "b"): 0.1, ("a", "b"): 5.0,
             ("a", "d"): 9.0, ("b", "d"): 9.0, ("c", "d"): 9.0}
    assert optimize_device_ordering(inst, links) == ["a", "c", "b", "d"]
    # symmetric lookup: only (dst, src) present still found
    assert optimize_device_ordering(["x", "y", "z"],
                                    {("z", "x"): 0.1, ("y", "x"): 5.0,
                                     ("y", "z"): 1.0}) == ["x", "z", "y"]


def test_compression_qsparse8_roundtrip():
    from dnet_amd import compression as cz
    torch.manual_seed(1)
    x = torch.randn(4, 128, dtype=torch.bfloat16)
    x[:, :16] *= 10
    blob8 = cz.compress_tensor_to_bytes(x, 0.5, quantize=True)
    blob16 = cz.compress_tensor_to_bytes(x, 0.5)
    # payload shrinks ~4x (int8 codes + g64 scales); headers are fixed-cost
    assert len(blob8) < len(blob16)
    y = cz.decompress_tensor_from_bytes(blob8)
    # kept columns reconstruct within int8 quant error
    idx, packed = cz.column_sparsify(x, 0.5)
    kept = y.index_select(-1, idx.long()).float()
    err = (kept - packed.float()).abs().max()
    scale = packed.float().abs().max() / 127
    assert err <= 2 * scale
    # dropped columns are zero
    mask = torch.ones(128, dtype=torch.bool)
    mask[idx.long()] = False
    assert y[:, mask].abs().sum() == 0
    assert cz.is_compressed_dtype(
        cz.dtype_string("bfloat16", 0.5, cz.FMT_QSPARSE8_V1))


def test_row_sampler():
    from dnet_amd.core.sampler import DecodingConfig, RowSampler
    torch.manual_seed(0)
    rs = RowSampler(3)
    rs.set_row(0, DecodingConfig(temperature=0.0))
    rs.set_row(1, DecodingConfig(temperature=1.0, top_k=1))
    rs.set_row(2, DecodingConfig(temperature=0.8, top_p=0.01))
    logits = torch.tensor([[0.1, 5.0, 0.2, 0.3]] * 3)
    for _ in range(8):
        assert rs.sample(logits).tolist() == [1, 1, 1]
    # min-p path: mass below min_p * max prob is filtered
    rs2 = RowSampler(1)
    rs2.set_row(0, DecodingConfig(temperature=1.0, min_p=0.9))
    for _ in range(8):
        assert int(rs2.sample(logits[:1])[0]) == 1


@pytest.mark.parametrize("model_type,qbits", [
    ("mixtral", 0), ("mixtral", 8), ("gpt_oss", 0), ("gpt_oss", "mxfp4"),
    ("qwen2_moe", 0)])
def test_moe_offload_matches_fit(model_type, qbits):
    """MoE layers stream through the weight cache (stacked expert banks in
    the slot template) and produce the fit path's exact tokens."""
    from dnet_amd.models import ModelConfig
    from dnet_amd.parallel.ring import RingExecutor

    hf = dict(model_type=model_type, hidden_size=64, num_hidden_layers=4,
              num_attention_heads=4, num_key_value_heads=2, head_dim=16,
              vocab_size=128, intermediate_size=64, num_local_experts=4,
              num_experts_per_tok=2, rope_theta=10000.0, rms_norm_eps=1e-5)
    if model_type == "gpt_oss":
        hf["sliding_window"] = 16
        hf["attention_bias"] = True
    if model_type == "qwen2_moe":
        hf["num_experts"] = hf.pop("num_local_experts")
        hf["moe_intermediate_size"] = 32
        hf["shared_expert_intermediate_size"] = 48
        hf["norm_topk_prob"] = True
    from dnet_amd.models import QuantConfig
    quant = (QuantConfig(8, 16) if qbits == 8 else
             QuantConfig(4, 32, fmt="mxfp4") if qbits == "mxfp4" else None)
    cfg = ModelConfig.from_hf(hf, quant=quant)
    toks = torch.randint(0, cfg.vocab_size, (1, 2, 6),
                         generator=torch.Generator().manual_seed(5))

    def run(residency):
        ex = RingExecutor(cfg, 0, 1, "cpu", mb_count=1, mb_size=2, smax=32,
                          seed=13, use_graphs=False, residency=residency)
        first = ex.prefill(toks.clone())
        gen = ex.decode_rounds(4)
        return torch.cat([first.unsqueeze(-1), gen], dim=-1)

    fit = run(0)
    off = run(2)
    assert torch.equal(fit, off)


@pytest.mark.parametrize("scoring", ["softmax", "sigmoid"])
def test_deepseek_partial_offload_matches_fit(scoring):
    """Mixed dense/MoE stacks (deepseek first_k_dense_replace): the
    uniform MoE suffix streams, the dense layer stays resident — tokens
    match the fully-resident run. The sigmoid case (deepseek-v3 noaux_tc)
    additionally covers router_bias surviving the weight-cache round-trip
    (advisor r1: router_bias was missing from _TENSOR_FIELDS)."""
    from dnet_amd.models import ModelConfig
    from dnet_amd.parallel.ring import RingExecutor

    hf = dict(model_type="deepseek_v2", hidden_size=64, num_hidden_layers=4,
              num_attention_heads=4, num_key_value_heads=4, vocab_size=128,
              intermediate_size=64, kv_lora_rank=32, qk_nope_head_dim=16,
              qk_rope_head_dim=8, v_head_dim=16, n_routed_experts=4,
              num_experts_per_tok=2, n_shared_experts=1,
              moe_intermediate_size=32, first_k_dense_replace=1,
              routed_scaling_factor=1.0, rope_theta=10000.0)
    if scoring == "sigmoid":
        hf.update(scoring_func="sigmoid", topk_method="noaux_tc",
                  n_group=2, topk_group=1, norm_topk_prob=True)
    cfg = ModelConfig.from_hf(hf)
    toks = torch.randint(0, cfg.vocab_size, (1, 2, 6),
                         generator=torch.Generator().manual_seed(5))

    def run(residency):
        ex = RingExecutor(cfg, 0, 1, "cpu", mb_count=1, mb_size=2, smax=32,
                          seed=13, use_graphs=False, residency=residency)
        first = ex.prefill(toks.clone())
        gen = ex.decode_rounds(4)
        return torch.cat([first.unsqueeze(-1), gen], dim=-1)

    fit = run(0)
    off = run(2)
    assert torch.equal(fit, off)


def test_halda_dp_optimal_vs_bruteforce():
    """the layer-distribution DP must be makespan-
    OPTIMAL on heterogeneous profiles — asserted against brute force on
    small instances (mixed HBM speeds, capacity-limited devices that
    must host-swap)."""
    import itertools

    from dnet_amd.parallel.profiler import DeviceProfile
    from dnet_amd.parallel.solver import halda_solve

    cases = [
        # (hbm_gbps, h2d_gbps, hbm_free_gb) per device, layers, layer GB
        ([(6000, 50, 280), (6000, 50, 280), (2000, 50, 280)], 13, 0.5),
        ([(6000, 50, 4.5), (3000, 50, 280)], 12, 1.0),   # dev0 cap ~ 1
        ([(8000, 60, 280), (4000, 30, 6.0), (2000, 20, 280),
          (1000, 10, 280)], 17, 1.0),
    ]
    for devs_spec, L, gb in cases:
        devs = [DeviceProfile(instance=f"d{i}", hbm_gbps=b, h2d_gbps=h,
                              hbm_free_gb=f)
                for i, (b, h, f) in enumerate(devs_spec)]
        layer_bytes = gb * 1e9
        res = halda_solve(devs, L, layer_bytes)
        # reconstruct per-device cost exactly like the solver
        overhead = 4.0

        def t_of(i, cnt):
            d = devs[i]
            cr = layer_bytes / (d.hbm_gbps * 1e9) * 1e3
            cs = layer_bytes / (d.h2d_gbps * 1e9) * 1e3
            cap = max(int(max(d.hbm_free_gb - overhead, 0.5) * 1e9
                          // layer_bytes), 1)
            r = min(cnt, cap)
            return r * cr + (cnt - r) * cs

        got = max(t_of(i, res.w[i]) for i in range(len(devs)))
        best = min(
            max(t_of(i, c) for i, c in enumerate(combo))
            for combo in itertools.product(range(L + 1),
                                           repeat=len(devs))
            if sum(combo) == L)
        assert sum(res.w) == L
        assert abs(got - best) < 1e-9, (res.w, got, best)


def test_mla_latent_cache_matches_perhead():
    """the compressed (latent) MLA KV cache — c_kv +
    shared roped key, 576/token instead of nh*(192+128) — must produce
    the per-head path's tokens via weight absorption, at ~nh*320/576 the
    cache footprint."""
    from dnet_amd.models import ModelConfig
    from dnet_amd.parallel.ring import RingExecutor

    hf = dict(model_type="deepseek_v2", hidden_size=64, num_hidden_layers=3,
              num_attention_heads=8, num_key_value_heads=8, vocab_size=128,
              intermediate_size=64, kv_lora_rank=32, qk_nope_head_dim=16,
              qk_rope_head_dim=8, v_head_dim=16, n_routed_experts=4,
              num_experts_per_tok=2, n_shared_experts=1,
              moe_intermediate_size=32, first_k_dense_replace=1,
              routed_scaling_factor=1.0, rope_theta=10000.0)
    cfg = ModelConfig.from_hf(hf)
    toks = torch.randint(0, cfg.vocab_size, (1, 2, 6),
                         generator=torch.Generator().manual_seed(9))

    def run(perhead):
        # explicit: the default is context-aware (latent only for
        # smax >= 2048), so force each mode
        os.environ["DNET_MLA_PERHEAD"] = "1" if perhead else "0"
        try:
            ex = RingExecutor(cfg, 0, 1, "cpu", mb_count=1, mb_size=2,
                              smax=32, seed=21, use_graphs=False)
            first = ex.prefill(toks.clone(), chunk=4)  # continuation path
            gen = ex.decode_rounds(5)
            return (torch.cat([first.unsqueeze(-1), gen], dim=-1),
                    ex.kvs[0].nbytes())
        finally:
            os.environ.pop("DNET_MLA_PERHEAD", None)

    lat, lat_bytes = run(False)
    ph, ph_bytes = run(True)
    assert torch.equal(lat, ph), (lat, ph)
    # nh*(nope+rope+vd) = 8*40 = 320 per token vs lora+rope = 40
    assert lat_bytes * 4 < ph_bytes
