"""CPU tests of the reference op implementations (no GPU needed)."""
import torch

import dnet_amd.ops as ops
from dnet_amd.ops import reference as ref


def test_quantize_roundtrip():
    torch.manual_seed(0)
    w = torch.randn(64, 256, dtype=torch.bfloat16)
    q, s = ops.quantize_int8(w, 64)
    assert q.dtype == torch.int8 and s.shape == (64, 4)
    wd = ref.dequant_int8(q, s, 64)
    assert (wd.float() - w.float()).abs().max().item() < 0.05


def test_rmsnorm_residual_inplace():
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    r = torch.randn(4, 64, dtype=torch.bfloat16)
    r0 = r.clone()
    y = ops.rmsnorm(x, r, torch.ones(64, dtype=torch.bfloat16), 1e-6)
    assert torch.allclose(r.float(), (x.float() + r0.float()), atol=2e-2)
    assert y.shape == x.shape


def test_attn_decode_matches_sdpa():
    torch.manual_seed(1)
    B, Hq, Hkv, D, S = 2, 4, 2, 64, 32
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, 128, D, dtype=torch.bfloat16)
    vc = torch.randn(B, Hkv, 128, D, dtype=torch.bfloat16)
    pos = torch.tensor([S, 7], dtype=torch.int32)
    out = ref.attn_decode(q, kc, vc, pos, D ** -0.5)
    # cross-check against torch sdpa for batch 0
    G = Hq // Hkv
    qe = q[0].float().view(Hkv, G, D)
    o = torch.nn.functional.scaled_dot_product_attention(
        qe, kc[0, :, :S].float(), vc[0, :, :S].float())
    assert torch.allclose(out[0].float(), o.reshape(Hq, D), atol=2e-2)


def test_rope_tables_llama3_scaling():
    cos, sin = ops.rope_tables(64, 128, 500000.0, scaling={
        "rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
        "high_freq_factor": 4.0, "original_max_position_embeddings": 8192})
    assert cos.shape == (64, 64) and torch.isfinite(cos).all()


def test_swiglu_shapes():
    gu = torch.randn(3, 5, 2 * 32, dtype=torch.bfloat16)
    y = ops.swiglu(gu)
    assert y.shape == (3, 5, 32)


def test_pack_unpack_int8_mfma():
    q = torch.randint(-127, 128, (4, 256), dtype=torch.int8)
    p = ref.pack_int8_mfma(q)
    assert not torch.equal(p, q)
    assert torch.equal(ref.unpack_int8_mfma(p), q)


def test_quantize_int4_roundtrip():
    torch.manual_seed(2)
    w = torch.randn(8, 256, dtype=torch.bfloat16)
    q4, s = ref.quantize_int4(w, 128)
    assert q4.shape == (8, 128) and q4.dtype == torch.uint8
    wd = ref.dequant_int4(q4, s, 128)
    assert (wd.float() - w.float()).abs().max().item() < 0.5
    # pack/unpack consistency through the chunk-quad layout
    p = ref.pack_int4_mfma(q4)
    vals = ref.unpack_int4(p) + 8
    v = vals.view(8, 2, 4, 4, 8).permute(0, 1, 3, 2, 4).reshape(8, 256)
    assert torch.equal(v, ref.unpack_int4(q4) + 8)


def test_attn_partials_combine_matches_full():
    """Sharding the KV sequence into partials + combine == full attention
    (the context-parallel decomposition, single process)."""
    import dnet_amd.ops as ops
    from dnet_amd.ops import reference as ref
    torch.manual_seed(0)
    B, Hq, Hkv, S, D = 2, 8, 2, 96, 64
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16)
    kc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16)
    vc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16)
    pos = torch.tensor([90, 33], dtype=torch.int32)
    full = ref.attn_decode(q, kc, vc, pos, D ** -0.5)
    # two S shards of 48; local lengths clamp per shard
    parts = []
    for r in range(2):
        s0, s1 = r * 48, (r + 1) * 48
        ln = (pos - s0).clamp(0, s1 - s0).to(torch.int32)
        parts.append(ops.attn_decode_partials(
            q, kc[:, :, s0:s1].contiguous(), vc[:, :, s0:s1].contiguous(),
            ln, D ** -0.5))
    out = ops.attn_combine(torch.cat(parts, dim=2))
    assert torch.allclose(out.float(), full.float(), atol=3e-2, rtol=3e-2)
    # sinks fold in at combine time (gpt-oss)
    sinks = torch.randn(Hq, dtype=torch.bfloat16)
    full_s = ref.attn_decode(q, kc, vc, pos, D ** -0.5, sinks=sinks)
    out_s = ops.attn_combine(torch.cat(parts, dim=2), sinks)
    assert torch.allclose(out_s.float(), full_s.float(), atol=3e-2, rtol=3e-2)
