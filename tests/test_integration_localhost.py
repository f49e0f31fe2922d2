"""Localhost integration: real API + shard processes over HTTP/wire/gloo.

Mirrors the reference's integration tier (reference:
tests/integration/test_model_catalog.py — prepare_topology -> load_model ->
chat -> assert content -> unload, on localhost processes). Uses the
tiny-random catalog model (byte tokenizer, random-init weights, CPU).
"""
import json
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import httpx
import pytest

REPO = Path(__file__).resolve().parent.parent


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _wait_http(url, timeout=90):
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            r = httpx.get(url, timeout=2)
            if r.status_code == 200:
                return True
        except httpx.HTTPError:
            pass
        time.sleep(0.5)
    return False


@pytest.fixture(scope="module")
def cluster(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("cluster")
    ports = {f"{role}{i}": _free_port() for role in ("http", "wire")
             for i in range(2)}
    api_port, api_wire = _free_port(), _free_port()
    hostfile = tmp / "hosts"
    hostfile.write_text(
        f"shard0 127.0.0.1 {ports['http0']} {ports['wire0']} 0\n"
        f"shard1 127.0.0.1 {ports['http1']} {ports['wire1']} 0\n")
    env = dict(os.environ)
    env["DNET_LOG_DIR"] = str(tmp / "logs")
    env["DNET_TRANSPORT_MASTER_PORT"] = str(_free_port())
    procs = []
    try:
        for i in range(2):
            procs.append(subprocess.Popen(
                [sys.executable, "-m", "dnet_amd.cli.shard", "--name",
                 f"shard{i}", "--host", "127.0.0.1",
                 "--http-port", str(ports[f"http{i}"]),
                 "--wire-port", str(ports[f"wire{i}"])],
                cwd=REPO, env=env))
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "dnet_amd.cli.api", "--hostfile",
             str(hostfile), "--host", "127.0.0.1", "--port", str(api_port),
             "--wire-port", str(api_wire), "--callback-addr",
             f"127.0.0.1:{api_wire}"],
            cwd=REPO, env=env))
        for i in range(2):
            assert _wait_http(f"http://127.0.0.1:{ports[f'http{i}']}/health"), \
                f"shard{i} did not come up"
        assert _wait_http(f"http://127.0.0.1:{api_port}/health")
        yield {"api": f"http://127.0.0.1:{api_port}"}
    finally:
        for p in procs:
            p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()


@pytest.mark.timeout(300)
def test_full_flow(cluster):
    api = cluster["api"]
    with httpx.Client(timeout=180) as c:
        # models listed
        r = c.get(f"{api}/v1/models")
        ids = [m["id"] for m in r.json()["data"]]
        assert "tiny-random" in ids

        # devices discovered
        r = c.get(f"{api}/v1/devices")
        assert set(r.json()) >= {"shard0", "shard1"}

        # manual topology: 2 layers each
        r = c.post(f"{api}/v1/prepare_topology_manual", json={
            "model": "tiny-random",
            "assignments": [
                {"instance": "shard0", "layers": [0, 1]},
                {"instance": "shard1", "layers": [2, 3]}]})
        assert r.status_code == 200, r.text
        topo = r.json()
        assert topo["num_layers"] == 4
        assert topo["assignments"][0]["next_instance"] == "shard1"

        # topology is re-fetchable (the checkpoint format)
        assert c.get(f"{api}/v1/topology").json()["model"] == "tiny-random"

        # load
        r = c.post(f"{api}/v1/load_model",
                   json={"model": "tiny-random", "max_tokens": 8})
        assert r.status_code == 200, r.text

        # non-streaming chat
        r = c.post(f"{api}/v1/chat/completions", json={
            "model": "tiny-random", "stream": False, "max_tokens": 8,
            "profile": True,
            "messages": [{"role": "user", "content": "hi there"}]})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["object"] == "chat.completion"
        assert body["usage"]["completion_tokens"] >= 1
        assert body["metrics"]["ttfb_ms"] > 0

        # streaming chat (SSE)
        chunks = []
        with c.stream("POST", f"{api}/v1/chat/completions", json={
                "model": "tiny-random", "stream": True, "max_tokens": 6,
                "messages": [{"role": "user", "content": "stream please"}]
                }) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                if line.startswith("data: ") and line != "data: [DONE]":
                    chunks.append(json.loads(line[6:]))
        assert chunks and chunks[-1].get("usage") is not None

        # two concurrent chats: the shard driver serializes them per ring;
        # token frames demux by nonce and both requests complete
        import concurrent.futures as cf

        def one(i):
            rr = httpx.post(f"{api}/v1/chat/completions", json={
                "model": "tiny-random", "stream": False, "max_tokens": 6,
                "messages": [{"role": "user", "content": f"concurrent {i}"}]},
                timeout=120)
            return rr.status_code, rr.json()

        with cf.ThreadPoolExecutor(2) as pool:
            results = list(pool.map(one, range(2)))
        for code, body2 in results:
            assert code == 200, body2
            assert body2["usage"]["completion_tokens"] >= 1

        # unload
        r = c.post(f"{api}/v1/unload_model")
        assert r.status_code == 200
