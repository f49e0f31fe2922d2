"""dnet_amd: MI355X-native distributed LLM inference engine."""


def enable_tunableop_cache() -> bool:
    """Point PyTorch TunableOp at the committed gfx950 algorithm cache
    (profiles/gfx950_tunableop0.csv — prefill GEMM shapes of the
    qwen-32b headline bench; TTFT 523 -> 476 ms measured). MUST be
    called BEFORE torch is first imported.

    NOT enabled automatically: TunableOp wraps every GEMM dispatch and
    measured a 17% decode tax on eager-path models whose shapes are not
    in the cache (deepseek-v2-lite 1493 -> 1239 tok/s) — so only
    bench.py enables it, and only for the headline model. Returns True
    if the cache file exists and the env was set."""
    import os
    import sys
    if "torch" in sys.modules:  # too late — torch reads the env at load
        return False
    # torch appends the DEVICE ORDINAL before the extension when
    # reading/writing (gfx950_tunableop.csv -> gfx950_tunableop<N>.csv);
    # profiles/ ships copies for ordinals 0-7 so every rank of a
    # multi-GPU bench run gets the tuned prefill algorithms
    tun = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..",
                       "profiles", "gfx950_tunableop.csv")
    if not os.path.exists(tun.replace(".csv", "0.csv")):
        return False
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", tun)
    return True
