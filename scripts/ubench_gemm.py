"""Microbenchmark the decode GEMM kernels per layer shape (GPU only).

Prints achieved weight-read GB/s for the MFMA path vs the scalar GEMV for
each Qwen-32B layer shape, plus graph-capture status for one decode layer.
"""
import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

import torch

import dnet_amd.ops as ops
from dnet_amd.ops import _native

SHAPES = [  # (name, N, K)
    ("qkv", 7168, 5120),
    ("o", 5120, 5120),
    ("gateup", 55296, 5120),
    ("down", 5120, 27648),
    ("lm_head", 152064, 5120),
]


def bench(fn, reps=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=str, default="1,16,32,64")
    args = ap.parse_args()
    ms = [int(v) for v in args.m.split(",")]
    dev = "cuda:0"
    nat = _native()
    for name, N, K in SHAPES:
        wf = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 30
        q, scales = ops.quantize_int8(wf, 128)
        qp = ops.pack_int8_mfma(q)
        wbytes_i8 = q.numel() + scales.numel() * 2
        wbytes_bf = wf.numel() * 2
        for M in ms:
            x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
            out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
            scratch = ops._get_scratch(dev)

            t_pk = bench(lambda: nat.gemm_m16(x, qp, scales, None, out,
                                              scratch, 128, True, 8))
            t_old = bench(lambda: nat.gemv_int8(x, q, scales, out, 128, None, False))
            t_mfma_bf = bench(lambda: nat.gemm_m16(x, wf, None, None, out,
                                                   scratch, 0, False, 16))
            print(f"{name:8s} N={N:6d} K={K:6d} M={M:2d}  "
                  f"i8-packed {t_pk*1e6:7.1f}us {wbytes_i8/t_pk/1e9:6.0f}GB/s | "
                  f"i8-scalar {wbytes_i8/t_old/1e9:6.0f} | "
                  f"bf16-mfma {t_mfma_bf*1e6:7.1f}us {wbytes_bf/t_mfma_bf/1e9:6.0f}GB/s")
        del wf, q, scales
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
