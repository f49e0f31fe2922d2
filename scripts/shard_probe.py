"""ShardRuntime decode throughput, with and without the HTTP/wire server
threads — isolates serving-loop slowdowns that only appear in the shard
process (run on a GPU box with DNET_OBS_PROFILE=true to see the
[PROFILE][DECODE] lines)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dnet_amd.core.types import ShardLoadModelRequest
from dnet_amd.shard.runtime import ShardRuntime


def infer(rt, label, n=48):
    toks = torch.randint(0, 1000, (1, 1, 32))
    rt._execute_infer(f"{label}", toks, n, [-1], {})


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "gpt-oss-20b"
    rt = ShardRuntime("probe")
    req = ShardLoadModelRequest(
        model_path=model, model_name=model, total_layers=24,
        layers=list(range(24)), rank=0, world_size=1, max_batch=1,
        max_seq=1024)
    rt._load(req)
    rt._callback = None
    print("== no servers ==", flush=True)
    for i in range(3):
        infer(rt, f"bare-{i}")
    print("== with servers ==", flush=True)
    from dnet_amd.shard.server import start_servers
    start_servers(rt, "127.0.0.1", 18991, 18992)
    import time
    time.sleep(2)
    for i in range(3):
        infer(rt, f"srv-{i}")
    print("DONE", flush=True)


if __name__ == "__main__":
    main()
