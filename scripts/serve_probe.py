"""Isolate serving-loop overhead vs raw engine throughput.

decode_rounds (bench path) enqueues steps back-to-back with no host syncs;
decode_stream (serving path) syncs per token (EOS check + token emit).
This probe times both, plus the stream loop with sampling and with a
host-callback, for a given model — run on a GPU box:
    python scripts/serve_probe.py gpt-oss-20b 32
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dnet_amd.core.sampler import DecodingConfig
from dnet_amd.models import ModelConfig, PRESETS, QuantConfig
from dnet_amd.parallel.ring import RingExecutor


def timed(label, ex, toks, f, n):
    ex.reset()
    ex.prefill(toks)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    f()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{label:28s} {dt * 1e3 / n:7.2f} ms/tok  {n / dt:8.1f} tok/s",
          flush=True)


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "gpt-oss-20b"
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 32
    graphs = "--no-graphs" not in sys.argv
    hf = dict(PRESETS[model])
    quant = QuantConfig(8, 128) if model.startswith("qwen") else None
    cfg = ModelConfig.from_hf(hf, quant=quant)
    ex = RingExecutor(cfg, 0, 1, "cuda:0", mb_count=1, mb_size=1, smax=1024,
                      seed=0, use_graphs=graphs)
    print("init done", flush=True)
    toks = torch.randint(0, cfg.vocab_size, (1, 1, 64), device="cuda:0")
    ex.prefill(toks)
    torch.cuda.synchronize()
    print("prefill done", flush=True)
    ex.decode_rounds(4)  # warm graphs
    torch.cuda.synchronize()
    print("warm done", flush=True)

    lb = ex.logits_buf[0].float()
    print("logits nan:", bool(torch.isnan(lb).any()),
          "inf:", bool(torch.isinf(lb).any()),
          "absmax:", float(lb.nan_to_num().abs().max()), flush=True)
    if "--sampled-only" in sys.argv:
        def cb0(step, tok, last):
            int(tok[0])
        ex.set_decoding(DecodingConfig(temperature=0.7, top_p=0.9))
        timed("stream sampled cb", ex, toks,
              lambda: ex.decode_stream(n + 1, stop_ids=[-1], on_token=cb0), n)
        return

    timed("decode_rounds (async)", ex, toks, lambda: ex.decode_rounds(n), n)
    timed("stream greedy no-cb", ex, toks,
          lambda: ex.decode_stream(n + 1), n)
    timed("stream greedy stop-check", ex, toks,
          lambda: ex.decode_stream(n + 1, stop_ids=[-1]), n)

    def cb(step, tok, last):
        int(tok[0])

    timed("stream greedy cb", ex, toks,
          lambda: ex.decode_stream(n + 1, stop_ids=[-1], on_token=cb), n)
    ex.set_decoding(DecodingConfig(temperature=0.7, top_p=0.9))
    timed("stream sampled cb", ex, toks,
          lambda: ex.decode_stream(n + 1, stop_ids=[-1], on_token=cb), n)


if __name__ == "__main__":
    main()
