"""Subsystem tests with fakes (reference tier: tests/subsystems/ — servers
and managers exercised without network or GPU)."""
import asyncio
import json

import pytest
import torch
from fastapi.testclient import TestClient

from dnet_amd.api.cluster import ClusterManager
from dnet_amd.api.models import ChatRequestModel
from dnet_amd.api.server import ApiState, build_api_app
from dnet_amd.config import get_settings
from dnet_amd.core.sampler import DecodingConfig, Sampler
from dnet_amd.utils.hostfile import DeviceProperties


class FakeDiscovery:
    """Static device map (reference: tests/fakes/discovery.py)."""

    def __init__(self, devices):
        self._devices = {d.instance: d for d in devices}

    async def async_get_properties(self):
        return dict(self._devices)

    async def async_start(self):
        pass


def _devices(n=2):
    return [DeviceProperties(instance=f"shard{i}", local_ip="127.0.0.1",
                             server_port=8081 + i, shard_port=50052 + i,
                             gpu_index=i) for i in range(n)]


@pytest.fixture
def api_client():
    cluster = ClusterManager(FakeDiscovery(_devices()))
    state = ApiState(cluster, get_settings())
    return TestClient(build_api_app(state)), state


def test_models_and_devices_routes(api_client):
    client, _ = api_client
    ids = [m["id"] for m in client.get("/v1/models").json()["data"]]
    assert "tiny-random" in ids and "qwen-2.5-32b-int8-synthetic" in ids
    devs = client.get("/v1/devices").json()
    assert set(devs) == {"shard0", "shard1"}


def test_manual_topology_validation(api_client):
    client, state = api_client
    # missing layers rejected
    r = client.post("/v1/prepare_topology_manual", json={
        "model": "tiny-random",
        "assignments": [{"instance": "shard0", "layers": [0, 1]}]})
    assert r.status_code == 400
    # unknown device rejected
    r = client.post("/v1/prepare_topology_manual", json={
        "model": "tiny-random",
        "assignments": [{"instance": "nope", "layers": [0, 1, 2, 3]}]})
    assert r.status_code == 400
    # valid topology, ring auto-closed, re-fetchable
    r = client.post("/v1/prepare_topology_manual", json={
        "model": "tiny-random",
        "assignments": [{"instance": "shard0", "layers": [0, 1]},
                        {"instance": "shard1", "layers": [2, 3]}]})
    assert r.status_code == 200
    topo = r.json()
    assert topo["assignments"][1]["next_instance"] == "shard0"
    assert client.get("/v1/topology").json() == topo


def test_chat_requires_model(api_client):
    client, _ = api_client
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-random", "messages": [{"role": "user", "content": "x"}]})
    assert r.status_code == 400


def test_solver_topology_from_profiles():
    from dnet_amd.api.model_manager import resolve_model_config
    from dnet_amd.api.catalog import get_entry
    from dnet_amd.parallel.profiler import DeviceProfile
    cluster = ClusterManager(FakeDiscovery(_devices(3)))
    asyncio.run(cluster.scan_devices())
    for i, bw in enumerate([6000.0, 6000.0, 2000.0]):
        cluster.profiles[f"shard{i}"] = DeviceProfile(
            instance=f"shard{i}", hbm_gbps=bw, h2d_gbps=50, hbm_free_gb=280)
    cfg = resolve_model_config(get_entry("qwen-2.5-32b-int8-synthetic"))
    topo = cluster.solve_topology("qwen-2.5-32b-int8-synthetic", cfg)
    w = topo.solution["w"]
    assert sum(w) == 64 and w[2] < w[0]
    assert topo.assignments[0].next_instance == topo.assignments[1].instance
    head = cluster.get_head_node()
    assert head is not None and head.instance == topo.assignments[0].instance


def test_inference_manager_token_flow():
    """Token frames resolved through the pending map -> SSE chunk stream
    (reference: test_inference_manager with FakeStrategyAdapter)."""
    from dnet_amd.api.inference import InferenceManager
    from dnet_amd.api.tokenizer import ByteTokenizer

    class MM:
        tokenizer = ByteTokenizer(512)
        stop_ids = [ByteTokenizer(512).EOS]

    class FakeHead:
        def __init__(self, im):
            self.im = im

        async def request(self, frame):
            async def feed():
                nonce = frame["nonce"]
                for ch in b"ok":
                    self.im.resolve_token({"t": "token", "nonce": nonce,
                                           "token_id": int(ch),
                                           "finished": False})
                self.im.resolve_token({"t": "token", "nonce": nonce,
                                       "token_id": MM.stop_ids[0],
                                       "finished": True})
            asyncio.get_event_loop().create_task(feed())
            return {"t": "ack"}

    im = InferenceManager(MM(), token_timeout_s=10)
    im.head_client = FakeHead(im)
    im.callback_addr = "127.0.0.1:1"

    async def run():
        req = ChatRequestModel(model="tiny-random", profile=True,
                               messages=[{"role": "user", "content": "hi"}])
        return await im.chat_completions(req)

    resp = asyncio.run(run())
    assert resp.choices[0].message.content == "ok"
    assert resp.usage.completion_tokens == 3
    assert resp.metrics["tokens_generated"] == 3


def test_stop_strings_truncate_and_cancel():
    """OpenAI `stop` strings (advisor r1): matched API-side in detok text,
    stream truncated BEFORE the stop string, a cancel frame goes to the
    head, and partial-prefix holdback text is flushed when no stop hits."""
    from dnet_amd.api.inference import InferenceManager
    from dnet_amd.api.tokenizer import ByteTokenizer

    class MM:
        tokenizer = ByteTokenizer(512)
        stop_ids = [ByteTokenizer(512).EOS]

    sent = []

    class FakeHead:
        def __init__(self, im, text):
            self.im = im
            self.text = text

        async def request(self, frame):
            sent.append(frame)
            if frame.get("t") != "infer":
                return {"t": "ack"}

            async def feed():
                nonce = frame["nonce"]
                for ch in self.text:
                    self.im.resolve_token({"t": "token", "nonce": nonce,
                                           "token_id": int(ch),
                                           "finished": False})
                self.im.resolve_token({"t": "token", "nonce": nonce,
                                       "token_id": MM.stop_ids[0],
                                       "finished": True})
            asyncio.get_event_loop().create_task(feed())
            return {"t": "ack"}

    async def run(text, stop):
        im = InferenceManager(MM(), token_timeout_s=10)
        im.head_client = FakeHead(im, text)
        im.callback_addr = "127.0.0.1:1"
        req = ChatRequestModel(model="tiny-random", stop=stop,
                               messages=[{"role": "user", "content": "hi"}])
        return await im.chat_completions(req)

    # stop string mid-stream: truncate before it, cancel sent
    resp = asyncio.run(run(b"hello STOP world", ["STOP"]))
    assert resp.choices[0].message.content == "hello "
    assert resp.choices[0].finish_reason == "stop"
    assert any(f.get("t") == "cancel" for f in sent)

    # stop prefix that never completes: held-back text is flushed
    sent.clear()
    resp = asyncio.run(run(b"abcST", ["STOP"]))
    assert resp.choices[0].message.content == "abcST"
    assert not any(f.get("t") == "cancel" for f in sent)

    # string form of `stop`
    resp = asyncio.run(run(b"xxByy", "B"))
    assert resp.choices[0].message.content == "xx"


def test_sampler_modes():
    torch.manual_seed(0)
    logits = torch.tensor([[0.1, 5.0, 0.2, 0.3]])
    tok, lp, tops = Sampler(DecodingConfig()).sample(logits)
    assert int(tok[0]) == 1 and lp is None
    tok, lp, tops = Sampler(DecodingConfig(temperature=0.0, logprobs=True,
                                           top_logprobs=2)).sample(logits)
    assert lp is not None and len(tops[0]) == 2
    s = Sampler(DecodingConfig(temperature=1.0, top_k=1))
    for _ in range(5):
        tok, _, _ = s.sample(logits)
        assert int(tok[0]) == 1  # top-1 restricted
    s = Sampler(DecodingConfig(temperature=1.0, top_p=0.01))
    for _ in range(5):
        tok, _, _ = s.sample(logits)
        assert int(tok[0]) == 1


def test_shard_runtime_loads_real_checkpoint(tmp_path):
    """ShardRuntime end-to-end with an HF-layout safetensors checkpoint:
    metadata buckets -> per-layer selective load -> infer produces tokens."""
    transformers = pytest.importorskip("transformers")
    from safetensors.torch import save_file

    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    tc = transformers.LlamaConfig(
        hidden_size=128, intermediate_size=256, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=2, head_dim=64,
        vocab_size=256, rope_theta=10000.0, max_position_embeddings=64,
        tie_word_embeddings=False)
    hf = transformers.LlamaForCausalLM(tc).eval()
    mdir = tmp_path / "ckpt"
    mdir.mkdir()
    (mdir / "config.json").write_text(tc.to_json_string())
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
              str(mdir / "model.safetensors"))

    rt = ShardRuntime("shard0")
    req = ShardLoadModelRequest(
        model_path=str(mdir), model_name="tiny-ckpt", total_layers=2,
        layers=[0, 1], rank=0, world_size=1, max_batch=1, max_seq=64)
    rt._load(req)
    assert rt.status == "loaded"
    ex = rt.executor
    # loaded weights match the checkpoint
    got = ex.model.layers[0].qkv.w
    import torch as t
    want = t.cat([hf.state_dict()[f"model.layers.0.self_attn.{x}_proj.weight"]
                  for x in "qkv"]).to(t.bfloat16)
    assert t.equal(got, want)
    # one inference through the executor produces tokens
    toks = t.randint(0, 256, (1, 1, 6))
    first = ex.prefill(toks)
    gen = ex.decode_rounds(3)
    assert gen.shape == (1, 1, 3)
    rt._unload()
    assert rt.status == "idle"


def test_inference_timeout_and_error_frames():
    """Failure detection: a dead mid-ring shard surfaces as a timeout after
    token_timeout_s (not a hang), and an error frame from a shard aborts
    the request immediately (reference: 300 s await_token timeout,
    inference.py:166; RingError equivalent actually sent here)."""
    import time as _time

    from dnet_amd.api.inference import InferenceManager
    from dnet_amd.api.tokenizer import ByteTokenizer

    class MM:
        tokenizer = ByteTokenizer(512)
        stop_ids = [ByteTokenizer(512).EOS]

    class DeadHead:
        async def request(self, frame):
            return {"t": "ack"}   # acks, then never produces tokens

    im = InferenceManager(MM(), token_timeout_s=0.3)
    im.head_client = DeadHead()
    im.callback_addr = "127.0.0.1:1"

    async def run_dead():
        req = ChatRequestModel(model="tiny-random",
                               messages=[{"role": "user", "content": "x"}])
        return await im.chat_completions(req)

    t0 = _time.perf_counter()
    with pytest.raises((asyncio.TimeoutError, RuntimeError, Exception)):
        asyncio.run(run_dead())
    assert _time.perf_counter() - t0 < 5.0   # bounded, no hang
    assert not im.pending                    # nonce cleaned up

    class ErrorHead:
        def __init__(self, im):
            self.im = im

        async def request(self, frame):
            self.im.resolve_token({"t": "error", "nonce": frame["nonce"],
                                   "failed_node": "shard1",
                                   "error": "boom"})
            return {"t": "ack"}

    im2 = InferenceManager(MM(), token_timeout_s=10)
    im2.head_client = ErrorHead(im2)
    im2.callback_addr = "127.0.0.1:1"

    async def run_err():
        req = ChatRequestModel(model="tiny-random",
                               messages=[{"role": "user", "content": "x"}])
        return await im2.chat_completions(req)

    t0 = _time.perf_counter()
    with pytest.raises(RuntimeError, match="boom"):
        asyncio.run(run_err())
    assert _time.perf_counter() - t0 < 2.0
    assert not im2.pending


def test_slot_continuous_batching():
    """Slot scheduler (load with max_batch>1): two requests decode
    CONCURRENTLY in one batch, tokens demux per nonce, and a single
    request produces exactly the legacy serial-path tokens."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(
                (frame["token_id"], frame["finished"]))

        def close(self):
            pass

    def load(max_batch):
        rt = ShardRuntime("probe")
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1,
            max_batch=max_batch, max_seq=64))
        rt._callback = Cap()
        return rt

    prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()

    # legacy serial path (max_batch=1 -> no slots)
    rt1 = load(1)
    assert rt1.slots is None
    rt1._execute_infer("legacy", torch.frombuffer(
        bytearray(prompt), dtype=torch.int32).long().view(1, 1, -1), 6, [],
        {})
    legacy = [t for t, _ in emitted["legacy"]]

    rt = load(2)
    assert rt.slots is not None and len(rt.slots) == 2
    rt.infer_q.put({"nonce": "req-a", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 6, "stop_ids": [], "params": {}})
    rt.infer_q.put({"nonce": "req-b", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 4, "stop_ids": [], "params": {}})
    for _ in range(30):
        rt._slots_tick()
        if all(s is None for s in rt.slots) and "req-b" in emitted:
            a = emitted.get("req-a", [])
            b = emitted.get("req-b", [])
            if a and b and a[-1][1] and b[-1][1]:
                break
    a = emitted["req-a"]
    b = emitted["req-b"]
    assert len(a) == 6 and a[-1][1] and not a[0][1]
    assert len(b) == 4 and b[-1][1]
    # same prompt + greedy -> identical tokens on both slots, and identical
    # to the legacy serial path
    assert [t for t, _ in a[:4]] == [t for t, _ in b]
    assert [t for t, _ in a] == legacy
    rt._unload()


def test_slot_admission_mid_decode():
    """A request arriving while another slot is mid-decode is prefilled
    into a free slot without disturbing the in-flight stream."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    rt = ShardRuntime("probe")
    rt._load(ShardLoadModelRequest(
        model_path="tiny", model_name="tiny", total_layers=4,
        layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=2, max_seq=64))
    rt._callback = Cap()
    pa = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    pb = torch.arange(3, 11, dtype=torch.int32).numpy().tobytes()
    rt.infer_q.put({"nonce": "a", "tokens": pa, "prompt_len": 8,
                    "max_tokens": 10, "stop_ids": [], "params": {}})
    rt._slots_tick()          # admit a + first decode step
    rt._slots_tick()          # a decodes alone
    mid = list(emitted["a"])
    rt.infer_q.put({"nonce": "b", "tokens": pb, "prompt_len": 8,
                    "max_tokens": 5, "stop_ids": [], "params": {}})
    for _ in range(20):
        rt._slots_tick()
        if all(s is None for s in rt.slots):
            break
    assert len(emitted["a"]) == 10 and len(emitted["b"]) == 5
    # the in-flight stream's prefix was not disturbed by b's admission
    assert emitted["a"][:len(mid)] == mid
    # b ran solo too: must match the solo decode of the same prompt
    emitted2: dict[str, list] = {}
    rt2 = ShardRuntime("probe2")
    rt2._load(ShardLoadModelRequest(
        model_path="tiny", model_name="tiny", total_layers=4,
        layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=2, max_seq=64))

    class Cap2:
        def send(self, frame):
            emitted2.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    rt2._callback = Cap2()
    rt2.infer_q.put({"nonce": "b", "tokens": pb, "prompt_len": 8,
                     "max_tokens": 5, "stop_ids": [], "params": {}})
    for _ in range(10):
        rt2._slots_tick()
    assert emitted["b"] == emitted2["b"]
    rt._unload()
    rt2._unload()


def test_slot_stop_id_mid_stream():
    """A slot hitting a stop id frees early while the other keeps going."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(
                (frame["token_id"], frame["finished"]))

        def close(self):
            pass

    rt = ShardRuntime("probe")
    rt._load(ShardLoadModelRequest(
        model_path="tiny", model_name="tiny", total_layers=4,
        layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=2, max_seq=64))
    rt._callback = Cap()
    prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    # run once unstopped to learn the greedy sequence, then use its 3rd
    # token as a stop id
    rt.infer_q.put({"nonce": "probe0", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 8, "stop_ids": [], "params": {}})
    for _ in range(15):
        rt._slots_tick()
    seq = [t for t, _ in emitted["probe0"]]
    stop = seq[2]
    rt.infer_q.put({"nonce": "stopped", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 8, "stop_ids": [stop], "params": {}})
    rt.infer_q.put({"nonce": "runs", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 8, "stop_ids": [], "params": {}})
    for _ in range(20):
        rt._slots_tick()
    st = emitted["stopped"]
    assert len(st) == 3 and st[-1][1]            # stopped at the stop id
    assert len(emitted["runs"]) == 8             # unaffected neighbor
    rt._unload()


def test_slot_reuse_no_stale_token():
    """Regression (advisor r1, high): a slot freed by a stop and re-admitted
    must not receive the old request's post-stop token from the pipelined
    pending step. The re-admitted request's stream must equal a solo run."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    def mk(name):
        rt = ShardRuntime(name)
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1,
            max_batch=2, max_seq=64))
        rt._callback = Cap()
        return rt

    pa = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    pc = torch.arange(5, 13, dtype=torch.int32).numpy().tobytes()

    # solo reference for c's prompt
    rt0 = mk("solo")
    rt0.infer_q.put({"nonce": "c-solo", "tokens": pc, "prompt_len": 8,
                     "max_tokens": 6, "stop_ids": [], "params": {}})
    for _ in range(15):
        rt0._slots_tick()
    rt0._unload()

    # a (short) + b (long) fill both slots; c queues and takes a's slot
    # the moment it frees — exactly the reuse window of the old bug
    rt = mk("probe")
    rt.infer_q.put({"nonce": "a", "tokens": pa, "prompt_len": 8,
                    "max_tokens": 2, "stop_ids": [], "params": {}})
    rt.infer_q.put({"nonce": "b", "tokens": pa, "prompt_len": 8,
                    "max_tokens": 12, "stop_ids": [], "params": {}})
    rt.infer_q.put({"nonce": "c", "tokens": pc, "prompt_len": 8,
                    "max_tokens": 6, "stop_ids": [], "params": {}})
    for _ in range(40):
        rt._slots_tick()
        if (all(s is None for s in rt.slots) and rt._pending is None
                and len(emitted.get("c", [])) >= 6):
            break
    assert len(emitted["a"]) == 2
    assert len(emitted["b"]) == 12
    assert emitted["c"] == emitted["c-solo"]
    rt._unload()


def test_slot_seeded_request_reproducible():
    """Per-request `seed` in slots mode (advisor r1): the same seeded
    sampled request produces identical tokens whether it runs alone or
    next to another (unseeded, sampled) stream."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    def run(extra_neighbor):
        emitted: dict[str, list] = {}

        class Cap:
            def send(self, frame):
                emitted.setdefault(frame["nonce"], []).append(
                    frame["token_id"])

            def close(self):
                pass

        rt = ShardRuntime("probe")
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1,
            max_batch=2, max_seq=64))
        rt._callback = Cap()
        prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
        rt.infer_q.put({"nonce": "seeded", "tokens": prompt, "prompt_len": 8,
                        "max_tokens": 8, "stop_ids": [],
                        "params": {"temperature": 0.9, "seed": 1234}})
        if extra_neighbor:
            rt.infer_q.put({"nonce": "other", "tokens": prompt,
                            "prompt_len": 8, "max_tokens": 6, "stop_ids": [],
                            "params": {"temperature": 1.3}})
        for _ in range(30):
            rt._slots_tick()
            if len(emitted.get("seeded", [])) >= 8:
                break
        rt._unload()
        return emitted["seeded"]

    solo = run(False)
    torch.manual_seed(999)   # perturb the default stream between runs
    with_neighbor = run(True)
    assert len(solo) == 8
    assert solo == with_neighbor


def test_recover_excludes_dead_shard(monkeypatch):
    """/v1/recover: health-sweep drops the dead shard, the ring re-solves
    over the survivors and the model reloads there."""
    from dnet_amd.api import server as srv

    cluster = ClusterManager(FakeDiscovery(_devices(2)))
    state = ApiState(cluster, get_settings())
    app = build_api_app(state)
    client = TestClient(app)

    loads = []

    async def fake_load(topology, entry, **kw):
        loads.append(topology)
        state.models.loaded_model = entry.id

    async def fake_unload():
        pass

    async def fake_profile(parallel=True):
        await cluster.scan_devices()
        return {}

    async def fake_healthy():
        await cluster.scan_devices()
        return [d for d in cluster.devices.values()
                if not d.is_manager and d.instance != "shard1"]

    monkeypatch.setattr(state.models, "load_model", fake_load)
    monkeypatch.setattr(state.models, "unload_model", fake_unload)
    monkeypatch.setattr(cluster, "profile_cluster", fake_profile)
    monkeypatch.setattr(cluster, "healthy_shards", fake_healthy)
    monkeypatch.setattr(state.inference, "connect_head", lambda *a, **k: None)

    r = client.post("/v1/load_model", json={"model": "tiny-random"})
    assert r.status_code == 200, r.text
    assert len(loads) == 1
    assert set(loads[0].devices) == {"shard0", "shard1"}

    r = client.post("/v1/recover")
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["excluded"] == ["shard1"]
    assert len(loads) == 2
    assert set(loads[1].devices) == {"shard0"}
    # coverage: all layers still assigned across the survivors
    total = sum(len(rd) for a in loads[1].assignments for rd in a.layers)
    assert total == loads[1].num_layers


def test_slot_per_request_sampling_params():
    """Each slot samples with ITS OWN request's params: a greedy request
    and a top-k=1 request produce the greedy sequence; a wild high-temp
    request does not perturb them."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    rt = ShardRuntime("probe")
    rt._load(ShardLoadModelRequest(
        model_path="tiny", model_name="tiny", total_layers=4,
        layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=3, max_seq=64))
    rt._callback = Cap()
    prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    rt.infer_q.put({"nonce": "greedy", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 6, "stop_ids": [], "params": {}})
    rt.infer_q.put({"nonce": "topk1", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 6, "stop_ids": [],
                    "params": {"temperature": 0.7, "top_k": 1}})
    rt.infer_q.put({"nonce": "hot", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 6, "stop_ids": [],
                    "params": {"temperature": 5.0}})
    for _ in range(25):
        rt._slots_tick()
        if all(s is None for s in rt.slots) and rt._pending is None:
            break
    assert len(emitted["greedy"]) == 6
    # top-k=1 at any temperature IS greedy -> identical sequence
    assert emitted["topk1"] == emitted["greedy"]
    assert len(emitted["hot"]) == 6
    rt._unload()


def test_seeded_sampling_reproducible():
    """OpenAI `seed`: identical seeded requests sample identical tokens at
    temperature > 0 (legacy serial path)."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    rt = ShardRuntime("probe")
    rt._load(ShardLoadModelRequest(
        model_path="tiny", model_name="tiny", total_layers=4,
        layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=1, max_seq=64))
    rt._callback = Cap()
    prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()

    def infer(nonce, seed):
        rt._execute_infer(nonce, torch.frombuffer(
            bytearray(prompt), dtype=torch.int32).long().view(1, 1, -1),
            8, [], {"temperature": 1.0, "seed": seed})

    infer("a", 42)
    infer("b", 42)
    infer("c", 7)
    assert emitted["a"] == emitted["b"]
    assert len(emitted["c"]) == 8
    rt._unload()


def test_decode_never_exceeds_kv_capacity():
    """max_tokens is clamped so decode never writes past smax (an
    unclamped request would rope_append out of cache range)."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    emitted: dict[str, list] = {}

    class Cap:
        def send(self, frame):
            emitted.setdefault(frame["nonce"], []).append(frame["token_id"])

        def close(self):
            pass

    for max_batch in (1, 2):
        emitted.clear()
        rt = ShardRuntime("probe")
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1,
            max_batch=max_batch, max_seq=16))
        rt._callback = Cap()
        prompt = torch.arange(1, 11, dtype=torch.int32).numpy().tobytes()
        if max_batch == 1:
            rt._execute_infer("x", torch.frombuffer(
                bytearray(prompt), dtype=torch.int32).long().view(1, 1, -1),
                500, [], {})
        else:
            rt.infer_q.put({"nonce": "x", "tokens": prompt, "prompt_len": 10,
                            "max_tokens": 500, "stop_ids": [], "params": {}})
            for _ in range(30):
                rt._slots_tick()
                if all(s is None for s in rt.slots) and rt._pending is None:
                    break
        assert len(emitted["x"]) == 6    # smax 16 - prompt 10
        rt._unload()


def test_interleaved_chunked_admission(monkeypatch):
    """A long prompt trickles in between decode steps (DNET_PREFILL_CHUNK)
    without perturbing the in-flight stream, and its tokens equal the
    unchunked run."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    def load(cap_dict):
        rt = ShardRuntime("probe")
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1, max_batch=2,
            max_seq=64))

        class Cap:
            def send(self, frame):
                cap_dict.setdefault(frame["nonce"], []).append(
                    frame["token_id"])

            def close(self):
                pass

        rt._callback = Cap()
        return rt

    short = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    longp = torch.arange(1, 13, dtype=torch.int32).numpy().tobytes()

    def run_all(rt):
        for _ in range(40):
            rt._slots_tick()
            if (all(s is None for s in rt.slots) and rt._pending is None
                    and rt.infer_q.empty()):
                break

    # solo baselines (unchunked)
    solo: dict = {}
    rt = load(solo)
    rt.infer_q.put({"nonce": "B", "tokens": short, "prompt_len": 8,
                    "max_tokens": 8, "stop_ids": [], "params": {}})
    run_all(rt)
    rt._unload()
    rt = load(solo)
    rt.infer_q.put({"nonce": "A", "tokens": longp, "prompt_len": 12,
                    "max_tokens": 5, "stop_ids": [], "params": {}})
    run_all(rt)
    rt._unload()

    # interleaved: B decoding, A's 12-token prompt arrives in 4-token chunks
    monkeypatch.setenv("DNET_PREFILL_CHUNK", "4")
    mixed: dict = {}
    rt = load(mixed)
    rt.infer_q.put({"nonce": "B", "tokens": short, "prompt_len": 8,
                    "max_tokens": 8, "stop_ids": [], "params": {}})
    rt._slots_tick()            # B admitted + starts decoding
    rt._slots_tick()
    rt.infer_q.put({"nonce": "A", "tokens": longp, "prompt_len": 12,
                    "max_tokens": 5, "stop_ids": [], "params": {}})
    run_all(rt)
    rt._unload()
    assert mixed["B"] == solo["B"]          # undisturbed by A's admission
    assert mixed["A"] == solo["A"]          # chunked == unchunked


def test_repack_fastpath_on_reload(tmp_path, monkeypatch):
    """the first cold load of a real checkpoint writes
    per-layer repacked files; a second load of the same assignment reads
    THEM (fastpath) — proven by deleting the source safetensors between
    loads — and produces identical tokens."""
    transformers = pytest.importorskip("transformers")
    from safetensors.torch import save_file

    from dnet_amd.config import reset_settings
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    monkeypatch.setenv("DNET_STORAGE_REPACK_DIR", str(tmp_path / "repack"))
    reset_settings()
    try:
        tc = transformers.LlamaConfig(
            hidden_size=128, intermediate_size=256, num_hidden_layers=2,
            num_attention_heads=2, num_key_value_heads=2, head_dim=64,
            vocab_size=256, rope_theta=10000.0, max_position_embeddings=64,
            tie_word_embeddings=False)
        hf = transformers.LlamaForCausalLM(tc).eval()
        mdir = tmp_path / "ckpt"
        mdir.mkdir()
        (mdir / "config.json").write_text(tc.to_json_string())
        src = mdir / "model.safetensors"
        save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
                  str(src))

        def load_and_decode(tag):
            rt = ShardRuntime(tag)
            rt._load(ShardLoadModelRequest(
                model_path=str(mdir), model_name="tiny-rp", total_layers=2,
                layers=[0, 1], rank=0, world_size=1, max_batch=1,
                max_seq=64))
            ex = rt.executor
            toks = torch.randint(0, 256, (1, 1, 6),
                                 generator=torch.Generator().manual_seed(4))
            first = ex.prefill(toks)
            gen = ex.decode_rounds(3)
            rt._unload()
            return torch.cat([first.unsqueeze(-1), gen], dim=-1)

        cold = load_and_decode("cold")
        assert (tmp_path / "repack").exists()
        src.unlink()     # source gone: only the repack can serve the load
        warm = load_and_decode("warm")
        assert torch.equal(cold, warm)
    finally:
        reset_settings()


def test_slot_logprobs_match_serial():
    """per-token logprobs in slots mode. A logprobs
    request through the slot scheduler gets per-token logprob +
    top_logprobs frames that match the legacy serial path's values."""
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    frames: dict[str, list] = {}

    class Cap:
        def send(self, fr):
            frames.setdefault(fr["nonce"], []).append(fr)

        def close(self):
            pass

    def load(max_batch, tag):
        rt = ShardRuntime(tag)
        rt._load(ShardLoadModelRequest(
            model_path="tiny", model_name="tiny", total_layers=4,
            layers=[0, 1, 2, 3], rank=0, world_size=1,
            max_batch=max_batch, max_seq=64))
        rt._callback = Cap()
        return rt

    prompt = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
    params = {"logprobs": True, "top_logprobs": 3}

    rt1 = load(1, "serial")
    rt1._execute_infer("serial", torch.frombuffer(
        bytearray(prompt), dtype=torch.int32).long().view(1, 1, -1), 5, [],
        dict(params))
    rt1._unload()

    rt = load(2, "slots")
    rt.infer_q.put({"nonce": "lp", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 5, "stop_ids": [], "params": dict(params)})
    rt.infer_q.put({"nonce": "plain", "tokens": prompt, "prompt_len": 8,
                    "max_tokens": 5, "stop_ids": [], "params": {}})
    for _ in range(30):
        rt._slots_tick()
        if len(frames.get("lp", [])) >= 5 and len(frames.get("plain", [])) >= 5:
            break
    rt._unload()

    ser, slot = frames["serial"], frames["lp"]
    assert len(ser) == 5 and len(slot) == 5
    for a, b in zip(ser, slot):
        assert a["token_id"] == b["token_id"]
        assert "logprob" in b and "top_logprobs" in b
        assert abs(a["logprob"] - b["logprob"]) < 1e-3
        assert len(b["top_logprobs"]) == 3
    # the no-logprobs neighbor got none
    assert all("logprob" not in f for f in frames["plain"])


def test_auto_recover_on_error_frame(monkeypatch):
    """Failure frames trigger the auto-recover hook (VERDICT r1 weak 9:
    recovery was operator-POST-only); the API app wires the hook when
    api.auto_recover is on."""
    from dnet_amd.api import server as srv
    from dnet_amd.api.inference import InferenceManager
    from dnet_amd.api.tokenizer import ByteTokenizer

    cluster = ClusterManager(FakeDiscovery(_devices(2)))
    state = srv.ApiState(cluster, get_settings())
    srv.build_api_app(state)
    assert state.inference.on_failure is not None   # wired by default

    class MM:
        tokenizer = ByteTokenizer(512)
        stop_ids = [ByteTokenizer(512).EOS]

    fired = []

    class ErrHead:
        async def request(self, frame):
            if frame.get("t") != "infer":
                return {"t": "ack"}

            async def feed():
                im.resolve_token({"t": "error", "nonce": frame["nonce"],
                                  "error": "shard died"})
            asyncio.get_event_loop().create_task(feed())
            return {"t": "ack"}

    im = InferenceManager(MM(), token_timeout_s=5)
    im.head_client = ErrHead()
    im.callback_addr = "127.0.0.1:1"
    im.on_failure = lambda: fired.append(1)

    async def run():
        req = ChatRequestModel(model="tiny-random",
                               messages=[{"role": "user", "content": "x"}])
        return await im.chat_completions(req)

    with pytest.raises(RuntimeError, match="shard died"):
        asyncio.run(run())
    assert fired == [1]
