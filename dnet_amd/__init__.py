"""dnet_amd: MI355X-native distributed LLM inference engine."""
import os as _os

# Pre-tuned hipBLASLt/rocBLAS algorithm cache for the prefill GEMM shapes
# (PyTorch TunableOp): measured TTFT 523 -> 475 ms on the qwen-32b
# 64x128 batched prefill. TUNING=0 means cached shapes use the tuned
# algorithm and uncached shapes fall back to the normal heuristics —
# no runtime tuning cost ever. Opt out by setting
# PYTORCH_TUNABLEOP_ENABLED yourself before import.
_tun = _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..",
                     "profiles", "gfx950_tunableop.csv")
_tun0 = _tun.replace(".csv", "0.csv")
if _os.path.exists(_tun0) and "PYTORCH_TUNABLEOP_ENABLED" not in _os.environ:
    _os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    _os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    _os.environ["PYTORCH_TUNABLEOP_FILENAME"] = _tun
