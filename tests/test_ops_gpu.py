"""GPU numerics tests: every HIP kernel vs its plain-PyTorch fp32 reference.

Mirrors the reference repo's kernel test strategy (golden tests on tiny
tensors; see SURVEY.md §4 carry-over note).
"""
import pytest
import torch

import dnet_amd.ops as ops
from dnet_amd.ops import reference as ref

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda:0")


def test_native_loaded():
    # On a GPU box the in-tree extension must be what runs — no eager fallback.
    assert ops.has_native(), "native _C.so must load on the GPU box"


def test_rmsnorm():
    torch.manual_seed(0)
    x = torch.randn(33, 5120, dtype=torch.bfloat16, device=_dev())
    w = torch.randn(5120, dtype=torch.bfloat16, device=_dev())
    y = ops.rmsnorm(x.clone(), None, w, 1e-6)
    y_ref = ref.rmsnorm(x.cpu().clone(), None, w.cpu(), 1e-6)
    assert torch.allclose(y.float().cpu(), y_ref.float(), atol=3e-2, rtol=3e-2)


def test_rmsnorm_residual():
    torch.manual_seed(1)
    x = torch.randn(9, 1024, dtype=torch.bfloat16, device=_dev())
    r = torch.randn(9, 1024, dtype=torch.bfloat16, device=_dev())
    w = torch.randn(1024, dtype=torch.bfloat16, device=_dev())
    r_ref = r.cpu().clone()
    y_ref = ref.rmsnorm(x.cpu().clone(), r_ref, w.cpu(), 1e-6)
    y = ops.rmsnorm(x, r, w, 1e-6)
    assert torch.allclose(y.float().cpu(), y_ref.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(r.float().cpu(), r_ref.float(), atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("m", [1, 2, 7, 16, 33])
def test_gemv_bf16(m):
    torch.manual_seed(2)
    K, N = 2048, 1536
    x = torch.randn(m, K, dtype=torch.bfloat16, device=_dev())
    w = torch.randn(N, K, dtype=torch.bfloat16, device=_dev()) / 30
    out = ops.gemv_bf16(x, w)
    out_ref = ref.gemv_bf16(x.cpu(), w.cpu())
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=5e-2, rtol=3e-2)


@pytest.mark.parametrize("m,group", [(1, 64), (4, 128), (13, 64)])
def test_gemv_int8(m, group):
    torch.manual_seed(3)
    K, N = 1024, 768
    x = torch.randn(m, K, dtype=torch.bfloat16, device=_dev())
    wf = torch.randn(N, K, dtype=torch.bfloat16, device=_dev()) / 30
    q, scales = ops.quantize_int8(wf, group)
    out = ops.gemv_int8(x, q, scales, group)
    out_ref = ref.gemv_int8(x.cpu(), q.cpu(), scales.cpu(), group)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=5e-2, rtol=3e-2)


def test_dequant_int8():
    torch.manual_seed(4)
    wf = torch.randn(256, 512, dtype=torch.bfloat16, device=_dev())
    q, scales = ops.quantize_int8(wf, 64)
    wd = ops.dequant_int8(q, scales, 64)
    wd_ref = ref.dequant_int8(q.cpu(), scales.cpu(), 64)
    assert torch.allclose(wd.float().cpu(), wd_ref.float(), atol=2e-2, rtol=2e-2)
    # quantization round trip stays close to the original
    assert (wd.float() - wf.float()).abs().max().item() < 0.05


@pytest.mark.parametrize("d,hq,hkv", [(128, 10, 2), (64, 8, 8), (128, 40, 8)])
def test_attn_decode(d, hq, hkv):
    torch.manual_seed(5)
    B, Smax = 3, 256
    q = torch.randn(B, hq, d, dtype=torch.bfloat16, device=_dev())
    kc = torch.randn(B, hkv, Smax, d, dtype=torch.bfloat16, device=_dev())
    vc = torch.randn(B, hkv, Smax, d, dtype=torch.bfloat16, device=_dev())
    pos = torch.tensor([5, 200, 64], dtype=torch.int32, device=_dev())[:B]
    scale = d ** -0.5
    out = ops.attn_decode(q, kc, vc, pos, scale)
    out_ref = ref.attn_decode(q.cpu(), kc.cpu(), vc.cpu(), pos.cpu(), scale)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=3e-2, rtol=3e-2)


def test_rope_append():
    torch.manual_seed(6)
    B, Hq, Hkv, D, Smax = 4, 8, 2, 128, 128
    cos, sin = ops.rope_tables(Smax, D, 10000.0)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=_dev())
    k = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    v = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    kc = torch.zeros(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.zeros_like(kc)
    pos = torch.tensor([0, 3, 77, 127], dtype=torch.int32, device=_dev())
    qr, kr = q.cpu().clone(), k.cpu().clone()
    kcr, vcr = kc.cpu().clone(), vc.cpu().clone()
    ref.rope_append(qr, kr, v.cpu().clone(), kcr, vcr, pos.cpu(), cos, sin)
    ops.rope_append(q, k, v, kc, vc, pos, cos.to(_dev()), sin.to(_dev()))
    assert torch.allclose(q.float().cpu(), qr.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(kc.float().cpu(), kcr.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(vc.float().cpu(), vcr.float(), atol=1e-3)


def test_swiglu():
    torch.manual_seed(7)
    gu = torch.randn(17, 2 * 1024, dtype=torch.bfloat16, device=_dev())
    y = ops.swiglu(gu)
    y_ref = ref.swiglu(gu.cpu())
    assert torch.allclose(y.float().cpu(), y_ref.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("m,n,k,group,bias", [
    (1, 1000, 2048, 0, True),
    (8, 1000, 2048, 64, True),
    (16, 256, 4096, 128, False),
    (33, 4096, 5120, 128, True),
])
def test_gemm_m16_shapes(m, n, k, group, bias):
    """MFMA decode GEMM path: odd N (tail waves), split-K shapes, bias."""
    torch.manual_seed(42)
    x = torch.randn(m, k, dtype=torch.bfloat16, device=_dev())
    b = (torch.randn(n, dtype=torch.bfloat16, device=_dev()) if bias else None)
    if group:
        wf = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
        q, scales = ops.quantize_int8(wf, group)
        out = ops.gemv_int8(x, q, scales, group, b)
        out_ref = ref.gemv_int8(x.cpu(), q.cpu(), scales.cpu(), group,
                                b.cpu() if bias else None)
    else:
        w = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
        out = ops.gemv_bf16(x, w, b)
        out_ref = ref.gemv_bf16(x.cpu(), w.cpu(), b.cpu() if bias else None)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=6e-2, rtol=3e-2)


@pytest.mark.parametrize("m,n,k,group", [(1, 512, 1024, 64), (8, 1000, 5120, 128),
                                         (16, 4096, 27648, 128)])
def test_gemm_m16_packed(m, n, k, group):
    """Packed MFMA weight layout must match the plain-layout reference."""
    torch.manual_seed(9)
    x = torch.randn(m, k, dtype=torch.bfloat16, device=_dev())
    wf = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
    q, scales = ops.quantize_int8(wf, group)
    qp = ops.pack_int8_mfma(q)
    out = ops.gemv_int8(x, qp, scales, group, None, packed=True)
    out_ref = ref.gemv_int8(x.cpu(), q.cpu(), scales.cpu(), group)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=6e-2, rtol=3e-2)
    # packed dequant inverts the layout
    wd = ops.dequant_int8(qp, scales, group, packed=True)
    wd_ref = ref.dequant_int8(q.cpu(), scales.cpu(), group)
    assert torch.allclose(wd.float().cpu(), wd_ref.float(), atol=2e-2, rtol=2e-2)


def test_offload_gpu_matches_fit():
    """GPU: windowed host->HBM weight streaming == fully resident."""
    from dnet_amd.models import ModelConfig, PRESETS, QuantConfig
    from dnet_amd.parallel.ring import RingExecutor
    cfg = ModelConfig.from_hf(dict(PRESETS["tiny"]), quant=QuantConfig(8, 64))
    toks = torch.randint(0, cfg.vocab_size, (1, 2, 6),
                         generator=torch.Generator().manual_seed(3)).cuda()

    def run(residency):
        ex = RingExecutor(cfg, 0, 1, "cuda:0", mb_count=1, mb_size=2, smax=32,
                          seed=11, use_graphs=False, residency=residency)
        first = ex.prefill(toks.clone())
        gen = ex.decode_rounds(4)
        return torch.cat([first.unsqueeze(-1), gen], dim=-1)

    assert torch.equal(run(0), run(2))


def test_compression_kernels_gpu():
    from dnet_amd import compression as cz
    torch.manual_seed(1)
    x = torch.randn(8, 512, dtype=torch.bfloat16, device=_dev())
    idx, packed = cz.column_sparsify(x, 0.25)
    idx_c, packed_c = cz.column_sparsify(x.cpu(), 0.25)
    assert torch.equal(idx.cpu(), idx_c)
    assert torch.equal(packed.cpu(), packed_c)
    y = cz.column_unsparsify(packed, idx, 512)
    assert torch.equal(y.cpu(), cz.column_unsparsify(packed_c, idx_c, 512))


@pytest.mark.parametrize("window,use_sinks", [(16, False), (0, True), (8, True)])
def test_attn_decode_window_sinks(window, use_sinks):
    torch.manual_seed(11)
    B, Hq, Hkv, D, Smax = 2, 8, 2, 64, 128
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=_dev())
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    pos = torch.tensor([100, 5], dtype=torch.int32, device=_dev())
    sinks = (torch.randn(Hq, dtype=torch.bfloat16, device=_dev())
             if use_sinks else None)
    out = ops.attn_decode(q, kc, vc, pos, D ** -0.5, window, sinks)
    out_ref = ref.attn_decode(q.cpu(), kc.cpu(), vc.cpu(), pos.cpu(),
                              D ** -0.5, window,
                              sinks.cpu() if use_sinks else None)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=3e-2,
                          rtol=3e-2)


def _clone_model(src, dst):
    """Copy every layer tensor/Linear of src into dst (cross-device) —
    init_random uses device RNG, so same-seed CPU/GPU weights differ."""
    from dnet_amd.models import Linear

    def conv(x, dev):
        if isinstance(x, torch.Tensor):
            return x.detach().to(dev)
        if isinstance(x, Linear):
            return Linear(conv(x.w, dev), conv(x.bias, dev) if x.bias is not None else None,
                          conv(x.scales, dev) if x.scales is not None else None,
                          x.group, x.packed)
        if isinstance(x, list):
            return [conv(e, dev) for e in x]
        return x

    dev = dst.device
    for lid, lw in src.layers.items():
        tgt = dst.layers[lid]
        for name, val in vars(lw).items():
            setattr(tgt, name, conv(val, dev))
    if src.embed is not None:
        dst.embed = conv(src.embed, dev)
    if src.final_norm is not None:
        dst.final_norm = conv(src.final_norm, dev)
    if src.lm_head is not None:
        dst.lm_head = conv(src.lm_head, dev)


def test_gpt_oss_decode_gpu():
    """gpt-oss decode on the HIP kernel path matches the CPU reference."""
    from dnet_amd.models import ModelConfig, get_ring_model
    torch.manual_seed(12)
    hf = dict(model_type="gpt_oss", hidden_size=128, num_hidden_layers=2,
              num_attention_heads=4, num_key_value_heads=2, head_dim=64,
              intermediate_size=128, num_local_experts=2,
              num_experts_per_tok=2, vocab_size=256, sliding_window=8,
              layer_types=["sliding_attention", "full_attention"])
    cfg = ModelConfig.from_hf(hf)

    def build(dev):
        m = get_ring_model("gpt_oss")(cfg, range(2), dev, True, True, smax=64)
        m.init_random(3)
        return m

    mg, mc = build("cuda:0"), build("cpu")
    _clone_model(mg, mc)  # identical weights on both devices
    toks = torch.randint(0, 256, (2, 10))
    kvg, kvc = mg.make_kv_cache(2, 64), mc.make_kv_cache(2, 64)
    hg = mg.embed_tokens(toks.cuda()).clone()
    hc = mc.embed_tokens(toks).clone()
    mg.prefill_window(hg, mg.layer_ids, kvg, 0)
    mc.prefill_window(hc, mc.layer_ids, kvc, 0)
    kvg.pos.fill_(10)
    kvc.pos.fill_(10)
    hdg = mg.embed_tokens(toks[:, -1].cuda()).clone()
    hdc = mc.embed_tokens(toks[:, -1]).clone()
    mg.decode_window(hdg, mg.layer_ids, kvg)
    mc.decode_window(hdc, mc.layer_ids, kvc)
    lg = mg.normalize_project(hdg).float().cpu()
    lc = mc.normalize_project(hdc).float()
    cos = torch.nn.functional.cosine_similarity(lg, lc, dim=-1)
    assert (cos > 0.99).all(), f"gpu vs cpu gpt_oss decode: {cos}"


def test_kv8_kernels_vs_ref():
    """Quantized-KV rope_append + attn_decode kernels vs the CPU reference."""
    torch.manual_seed(13)
    B, Hq, Hkv, D, Smax = 2, 8, 2, 128, 128
    cos, sin = ops.rope_tables(Smax, D, 10000.0)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=_dev())
    k = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    v = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    kc = torch.zeros(B, Hkv, Smax, D, dtype=torch.int8, device=_dev())
    vc = torch.zeros_like(kc)
    ks = torch.zeros(B, Hkv, Smax, 2, dtype=torch.bfloat16, device=_dev())
    vs = torch.zeros_like(ks)
    pos = torch.tensor([0, 77], dtype=torch.int32, device=_dev())
    qr, kr = q.cpu().clone(), k.cpu().clone()
    kcr, vcr = kc.cpu().clone(), vc.cpu().clone()
    ksr, vsr = ks.cpu().clone(), vs.cpu().clone()
    ref.rope_append(qr, kr, v.cpu().clone(), kcr, vcr, pos.cpu(), cos, sin,
                    ksr, vsr)
    ops.rope_append(q, k, v, kc, vc, pos, cos.to(_dev()), sin.to(_dev()),
                    ks, vs)
    assert torch.equal(kc.cpu(), kcr) or \
        (kc.cpu().int() - kcr.int()).abs().max() <= 1  # rounding edge
    assert torch.allclose(ks.float().cpu(), ksr.float(), atol=1e-3, rtol=1e-2)
    # decode attention over a random quantized cache
    torch.manual_seed(14)
    kc2 = torch.randint(-127, 128, (B, Hkv, Smax, D), dtype=torch.int8,
                        device=_dev())
    vc2 = torch.randint(-127, 128, (B, Hkv, Smax, D), dtype=torch.int8,
                        device=_dev())
    ks2 = (torch.rand(B, Hkv, Smax, 2, device=_dev()) * 0.02 + 0.005
           ).to(torch.bfloat16)
    vs2 = (torch.rand(B, Hkv, Smax, 2, device=_dev()) * 0.02 + 0.005
           ).to(torch.bfloat16)
    pos2 = torch.tensor([100, 5], dtype=torch.int32, device=_dev())
    out = ops.attn_decode(q, kc2, vc2, pos2, D ** -0.5, 0, None, ks2, vs2)
    out_ref = ref.attn_decode(q.cpu(), kc2.cpu(), vc2.cpu(), pos2.cpu(),
                              D ** -0.5, 0, None, ks2.cpu(), vs2.cpu())
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=3e-2,
                          rtol=3e-2)


def test_attn_decode_split_s_long():
    """Long-context decode exercises the split-S partials + combine path."""
    torch.manual_seed(21)
    B, Hq, Hkv, D, Smax = 2, 40, 8, 128, 2048
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=_dev())
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    pos = torch.tensor([2000, 37], dtype=torch.int32, device=_dev())
    out = ops.attn_decode(q, kc, vc, pos, D ** -0.5)
    out_ref = ref.attn_decode(q.cpu(), kc.cpu(), vc.cpu(), pos.cpu(), D ** -0.5)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=3e-2,
                          rtol=3e-2)


@pytest.mark.parametrize("m,n,k", [(1, 512, 1024), (8, 1000, 5120),
                                   (64, 4096, 27648)])
def test_gemm_int4(m, n, k):
    """int4 chunk-quad MFMA path vs the CPU reference."""
    torch.manual_seed(31)
    group = 128
    x = torch.randn(m, k, dtype=torch.bfloat16, device=_dev())
    wf = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
    q4, scales = ops.quantize_int4(wf.cpu(), group)
    q4p = ops.pack_int4_mfma(q4).to(_dev())
    out = ops.gemv_int4(x, q4p, scales.to(_dev()), group, None, packed=True)
    out_ref = ops.ref.gemv_int4(x.cpu(), q4, scales, group)
    assert torch.allclose(out.float().cpu(), out_ref.float(), atol=8e-2,
                          rtol=4e-2)
    wd = ops.dequant_int4(q4p, scales.to(_dev()), group)
    wd_ref = ops.ref.dequant_int4(q4, scales, group)
    assert torch.allclose(wd.float().cpu(), wd_ref.float(), atol=2e-2, rtol=2e-2)


def test_model_int4_consistency_gpu():
    """End-to-end int4 model decode vs its own bf16 weights (quality check)."""
    from dnet_amd.models import ModelConfig, PRESETS, QuantConfig, get_ring_model
    hf = dict(PRESETS["tiny"])
    cfg4 = ModelConfig.from_hf(hf, quant=QuantConfig(4, 128))
    m = get_ring_model(cfg4.model_type)(cfg4, range(4), "cuda:0", True, True,
                                        smax=64)
    m.init_random(5)
    kv = m.make_kv_cache(2, 64)
    toks = torch.randint(0, cfg4.vocab_size, (2, 8), device="cuda:0")
    h = m.embed_tokens(toks).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    kv.pos.fill_(8)
    hd = m.embed_tokens(toks[:, -1]).clone()
    m.decode_window(hd, m.layer_ids, kv)
    assert torch.isfinite(m.normalize_project(hd).float()).all()


@pytest.mark.parametrize("q8,packed,glu", [
    (False, False, 0), (False, False, 1), (True, False, 0), (True, True, 1)])
def test_moe_grouped_kernels(q8, packed, glu):
    """moe_gateup + moe_down vs fp32 reference (routed experts only — the
    kernel skips unrouted experts, leaving their act rows unwritten)."""
    torch.manual_seed(7)
    E, I, K, H, M, G = 8, 256, 512, 512, 3, 64
    dev = _dev()
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    gw = torch.randn(E, 2 * I, K, dtype=torch.bfloat16, device=dev) / 8
    dw = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) / 8
    gb = torch.randn(E, 2 * I, dtype=torch.bfloat16, device=dev) / 4
    db = torch.randn(E, H, dtype=torch.bfloat16, device=dev) / 4
    # route 2 experts per row
    we = torch.zeros(M, E, dtype=torch.float32, device=dev)
    idx = torch.stack([torch.randperm(E, device=dev)[:2] for _ in range(M)])
    we.scatter_(1, idx, torch.rand(M, 2, device=dev) + 0.1)
    gs = ds = None
    if q8:
        gq, gs = ops.quantize_int8(gw.reshape(E * 2 * I, K), G)
        dq, ds = ops.quantize_int8(dw.reshape(E * H, I), G)
        if packed:
            gq = ops.pack_int8_mfma(gq)
            dq = ops.pack_int8_mfma(dq)
        gw = gq.view(E, 2 * I, K).contiguous()
        dw = dq.view(E, H, I).contiguous()
        gs = gs.view(E, 2 * I, K // G).contiguous()
        ds = ds.view(E, H, I // G).contiguous()

    def cpu(t):
        return None if t is None else t.cpu()

    act = ops.moe_gateup(x, gw, gs, gb, we, G, packed, glu)
    act_ref = ref.moe_gateup(cpu(x), cpu(gw), cpu(gs), cpu(gb), cpu(we),
                             G, packed, glu)
    routed = we.cpu().t().bool()   # [E, M]
    assert torch.allclose(act.cpu().float()[routed],
                          act_ref.float()[routed], atol=5e-2, rtol=5e-2)
    # feed the reference act into BOTH downs so gateup error doesn't compound
    act_in = act_ref.to(dev).contiguous()
    out = ops.moe_down(act_in, dw, ds, db, we, G, packed)
    out_ref = ref.moe_down(act_ref, cpu(dw), cpu(ds), cpu(db), cpu(we),
                           G, packed)
    assert torch.allclose(out.cpu(), out_ref, atol=5e-2, rtol=5e-2)


def test_moe_dense_matches_sparse_gpu():
    """Grouped dense path == per-expert sparse loop on the same weights."""
    from dnet_amd.models import ModelConfig, PRESETS
    from dnet_amd.models.gpt_oss import GptOssRingModel
    hf = dict(PRESETS["gpt-oss-20b"])
    hf["num_hidden_layers"] = 2
    cfg = ModelConfig.from_hf(hf)
    m = GptOssRingModel(cfg, [0, 1], _dev(), False, False)
    m.init_random(seed=5)
    torch.manual_seed(3)
    y = torch.randn(4, cfg.hidden_size, dtype=torch.bfloat16, device=_dev())
    lw = m.layers[0]
    dense = m._mlp(y, lw)
    try:
        GptOssRingModel.DENSE_MOE_MAX_T = 0
        sparse = m._mlp(y, lw)
    finally:
        GptOssRingModel.DENSE_MOE_MAX_T = 64
    assert torch.allclose(dense.float(), sparse.float(), atol=4e-2, rtol=4e-2)


@pytest.mark.parametrize("q8,force_splits", [(False, 1), (False, 4),
                                             (True, 1)])
def test_attn_decode_mla_shape(q8, force_splits, monkeypatch):
    """MLA head dims (qk 192 / v 128) on the decode kernel vs fp32 ref."""
    monkeypatch.setenv("DNET_ATTN_SPLITS", str(force_splits))
    torch.manual_seed(11)
    B, H, S, D, DV = 3, 4, 256, 192, 128
    dev = _dev()
    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, H, S, DV, dtype=torch.bfloat16, device=dev)
    pos = torch.tensor([S, S // 2, 7], dtype=torch.int32, device=dev)
    ks = vs = None
    kq = vq = None
    if q8:
        kq, ks = ref.quantize_kv_rows(kc.reshape(-1, D))
        vq, vs = ref.quantize_kv_rows(vc.reshape(-1, DV))
        kq = kq.view(B, H, S, D)
        ks = ks.view(B, H, S, D // 64)
        vq = vq.view(B, H, S, DV)
        vs = vs.view(B, H, S, DV // 64)
    out = ops.attn_decode(q, kq if q8 else kc, vq if q8 else vc, pos,
                          D ** -0.5, kscale=ks, vscale=vs)
    out_ref = ref.attn_decode(q.cpu(), (kq if q8 else kc).cpu(),
                              (vq if q8 else vc).cpu(), pos.cpu(),
                              D ** -0.5,
                              kscale=None if ks is None else ks.cpu(),
                              vscale=None if vs is None else vs.cpu())
    assert out.shape == (B, H, DV)
    assert torch.allclose(out.float().cpu(), out_ref.float(),
                          atol=4e-2, rtol=4e-2)


def test_attn_partials_combine_gpu():
    """Native partials + native combine across simulated sequence shards
    == fp32 reference full attention (context-parallel decomposition)."""
    torch.manual_seed(13)
    B, Hq, Hkv, S, D = 2, 8, 2, 128, 128
    dev = _dev()
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=dev)
    kc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
    pos = torch.tensor([120, 45], dtype=torch.int32, device=dev)
    full = ref.attn_decode(q.cpu(), kc.cpu(), vc.cpu(), pos.cpu(), D ** -0.5)
    parts = []
    for r in range(2):
        s0, s1 = r * 64, (r + 1) * 64
        ln = (pos - s0).clamp(0, 64).to(torch.int32)
        parts.append(ops.attn_decode_partials(
            q, kc[:, :, s0:s1].contiguous(), vc[:, :, s0:s1].contiguous(),
            ln, D ** -0.5))
    out = ops.attn_combine(torch.cat(parts, dim=2))
    assert torch.allclose(out.float().cpu(), full.float(),
                          atol=3e-2, rtol=3e-2)


def test_rope_append_wpos():
    """CP local-write mode: out-of-shard rows are rotated but not stored."""
    torch.manual_seed(21)
    B, Hq, Hkv, D, Smax = 3, 4, 2, 128, 32
    cos, sin = ops.rope_tables(64, D, 10000.0)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=_dev())
    k = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    v = torch.randn(B, Hkv, D, dtype=torch.bfloat16, device=_dev())
    kc = torch.zeros(B, Hkv, Smax, D, dtype=torch.bfloat16, device=_dev())
    vc = torch.zeros_like(kc)
    pos = torch.tensor([40, 33, 20], dtype=torch.int32, device=_dev())
    wpos = (pos - 32).int()     # rank-1 shard of cap 32: 8, 1, -12
    qr, kr = q.cpu().clone(), k.cpu().clone()
    kcr, vcr = kc.cpu().clone(), vc.cpu().clone()
    ref.rope_append(qr, kr, v.cpu().clone(), kcr, vcr, pos.cpu(), cos, sin,
                    wpos=wpos.cpu())
    ops.rope_append(q, k, v, kc, vc, pos, cos.to(_dev()), sin.to(_dev()),
                    wpos=wpos)
    assert torch.allclose(q.float().cpu(), qr.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(kc.float().cpu(), kcr.float(), atol=3e-2, rtol=3e-2)
    # in-shard rows landed at their local index; out-of-shard wrote nothing
    assert kc[0, :, 8].abs().sum() > 0      # 40-32 = 8
    assert kc[1, :, 1].abs().sum() > 0      # 33-32 = 1
    assert kc[2].abs().sum() == 0           # 20-32 < 0: other rank's row


@pytest.mark.parametrize("m", [1, 8, 16, 24, 33, 64])
@pytest.mark.parametrize("k,group", [
    (320, 64),     # 1 full 256-k tile + 1 tail pair
    (1088, 64),    # 4 full tiles + tail
    (2048, 128),   # full tiles only
    (1280, 256),   # NSC=1
    (1920, 128),   # odd tail (7 full tiles + 2 tail pairs)
])
def test_gemm_stream_matches_generic_int8(m, k, group):
    """The streamed counted-vmcnt schedule must produce bit-identical
    results to the generic kernel for every MT tier, scale-group width,
    and split/tail geometry (incl. the partial last tile served by the
    serial tail path)."""

    torch.manual_seed(m * 1000 + k)
    n = 1000
    x = torch.randn(m, k, dtype=torch.bfloat16, device=_dev())
    wf = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
    q, scales = ops.quantize_int8(wf, group)
    qp = ops.pack_int8_mfma(q)
    bias = torch.randn(n, dtype=torch.bfloat16, device=_dev())
    out_s = ops.gemv_int8(x, qp, scales, group, bias, packed=True)
    out_ref = ref.gemv_int8(x.cpu(), q.cpu(), scales.cpu(), group, bias.cpu())
    assert torch.allclose(out_s.float().cpu(), out_ref.float(), atol=6e-2,
                          rtol=3e-2), \
        (out_s.float().cpu() - out_ref.float()).abs().max()


@pytest.mark.parametrize("m", [4, 40, 64])
@pytest.mark.parametrize("k,group", [(384, 128), (1152, 128), (1536, 256)])
def test_gemm_stream_matches_generic_int4(m, k, group):
    torch.manual_seed(m + k)
    n = 768
    x = torch.randn(m, k, dtype=torch.bfloat16, device=_dev())
    wf = torch.randn(n, k, dtype=torch.bfloat16, device=_dev()) / 30
    q4, scales = ops.quantize_int4(wf.cpu(), group)
    q4p = ops.pack_int4_mfma(q4).to(_dev())
    out_s = ops.gemv_int4(x, q4p, scales.to(_dev()), group, None, packed=True)
    out_ref = ops.ref.gemv_int4(x.cpu(), q4, scales, group)
    assert torch.allclose(out_s.float().cpu(), out_ref.float(), atol=8e-2,
                          rtol=4e-2)


@pytest.mark.parametrize("hq,hkv,d,dv,t,s,qoff,window,use_sinks", [
    (8, 2, 128, 128, 128, 128, 0, 0, False),      # GQA causal from 0
    (8, 8, 64, 64, 96, 160, 64, 0, False),        # continuation (qoff>0)
    (4, 4, 128, 128, 33, 61, 28, 0, False),       # odd T/S, partial q tile
    (8, 2, 128, 128, 128, 128, 0, 48, False),     # sliding window
    (8, 2, 128, 128, 120, 120, 0, 32, True),      # window + sinks (gpt-oss)
    (10, 10, 192, 128, 64, 90, 26, 0, False),     # MLA dims (qk192/v128)
])
def test_attn_prefill_flash_vs_einsum(hq, hkv, d, dv, t, s, qoff, window,
                                      use_sinks):
    """The fused MFMA flash prefill kernel must match the chunked-einsum
    reference path (bf16 GEMMs + fp32 softmax) on every geometry the
    serving path uses: GQA, continuation offsets, partial tiles, sliding
    window, sinks, MLA head dims."""
    from dnet_amd.models.base import _chunked_causal_attention
    torch.manual_seed(hq * 100 + t)
    B = 3
    q = torch.randn(B, hq, t, d, dtype=torch.bfloat16, device=_dev())
    k = torch.randn(B, hkv, s, d, dtype=torch.bfloat16, device=_dev())
    v = torch.randn(B, hkv, s, dv, dtype=torch.bfloat16, device=_dev())
    sinks = (torch.randn(hq, dtype=torch.bfloat16, device=_dev())
             if use_sinks else None)
    scale = d ** -0.5
    out = ops.attn_prefill(q, k, v, scale, qoff, window, sinks)
    import os
    os.environ["DNET_EINSUM_PREFILL"] = "1"
    try:
        ref_out = _chunked_causal_attention(q, k, v, scale, qoff, window,
                                            sinks)
    finally:
        os.environ.pop("DNET_EINSUM_PREFILL", None)
    assert torch.allclose(out.float(), ref_out.float(), atol=4e-2,
                          rtol=3e-2), \
        (out.float() - ref_out.float()).abs().max()


def test_slots_graph_churn_matches_eager():
    """Continuous batching with hipGraph decode ON (now the default) must
    be token-exact vs eager under slot churn: staggered admits, early
    stops, slot reuse, a chunked long-prompt admission mid-stream."""
    import os
    from dnet_amd.core.types import ShardLoadModelRequest
    from dnet_amd.shard.runtime import ShardRuntime

    def run(graphs: bool):
        os.environ["DNET_SLOTS_GRAPHS"] = "1" if graphs else "0"
        frames: dict[str, list] = {}

        class Cap:
            def send(self, fr):
                frames.setdefault(fr["nonce"], []).append(fr["token_id"])

            def close(self):
                pass

        try:
            rt = ShardRuntime("churn")
            rt._load(ShardLoadModelRequest(
                model_path="tiny", model_name="tiny", total_layers=4,
                layers=[0, 1, 2, 3], rank=0, world_size=1,
                max_batch=3, max_seq=96))
            rt._callback = Cap()
            pa = torch.arange(1, 9, dtype=torch.int32).numpy().tobytes()
            pb = torch.arange(3, 11, dtype=torch.int32).numpy().tobytes()
            plong = torch.arange(1, 41, dtype=torch.int32).numpy().tobytes()
            rt.infer_q.put({"nonce": "a", "tokens": pa, "prompt_len": 8,
                            "max_tokens": 3, "stop_ids": [], "params": {}})
            rt.infer_q.put({"nonce": "b", "tokens": pb, "prompt_len": 8,
                            "max_tokens": 12, "stop_ids": [], "params": {}})
            for _ in range(4):
                rt._slots_tick()
            # a finishes -> its slot is reused by c; long prompt d admits
            # CHUNKED between decode steps (DNET_PREFILL_CHUNK)
            os.environ["DNET_PREFILL_CHUNK"] = "16"
            rt.infer_q.put({"nonce": "c", "tokens": pa, "prompt_len": 8,
                            "max_tokens": 6, "stop_ids": [], "params": {}})
            rt.infer_q.put({"nonce": "d", "tokens": plong, "prompt_len": 40,
                            "max_tokens": 5, "stop_ids": [], "params": {}})
            for _ in range(40):
                rt._slots_tick()
                if (all(s is None for s in rt.slots) and rt._pending is None
                        and len(frames.get("d", [])) >= 5):
                    break
            rt._unload()
            return frames
        finally:
            os.environ.pop("DNET_SLOTS_GRAPHS", None)
            os.environ.pop("DNET_PREFILL_CHUNK", None)

    eager = run(False)
    graphed = run(True)
    assert set(eager) == set(graphed) == {"a", "b", "c", "d"}
    for k in eager:
        assert eager[k] == graphed[k], (k, eager[k], graphed[k])
    assert len(graphed["a"]) == 3 and len(graphed["b"]) == 12
    assert len(graphed["c"]) == 6 and len(graphed["d"]) == 5


def test_mxfp4_kernels_vs_ref():
    """Native MXFP4 execution: the dequant kernel and the grouped MoE
    kernels (fused e2m1+e8m0 dequant) vs the CPU reference."""
    torch.manual_seed(12)
    N, K = 96, 256
    w = torch.randn(N, K) / 4
    p, s = ops.quantize_mxfp4(w)
    got = ops.dequant_mxfp4(p.to(_dev()), s.to(_dev())).cpu()
    want = ref.dequant_mxfp4(p, s)
    assert torch.equal(got, want)

    # grouped MoE kernels with a packed mxfp4 expert bank
    E, I, H, M = 4, 64, 128, 5
    gw = torch.randn(E, 2 * I, H) / 4
    dw = torch.randn(E, H, I) / 4
    gp = torch.stack([ops.quantize_mxfp4(gw[e])[0] for e in range(E)])
    gs = torch.stack([ops.quantize_mxfp4(gw[e])[1] for e in range(E)])
    dp = torch.stack([ops.quantize_mxfp4(dw[e])[0] for e in range(E)])
    ds = torch.stack([ops.quantize_mxfp4(dw[e])[1] for e in range(E)])
    x = torch.randn(M, H, dtype=torch.bfloat16)
    we = torch.zeros(M, E)
    we[torch.arange(M), torch.randint(0, E, (M,))] = 1.0
    gb = torch.randn(E, 2 * I, dtype=torch.bfloat16)
    db = torch.randn(E, H, dtype=torch.bfloat16)
    act = ops.moe_gateup(x.to(_dev()), gp.to(_dev()), gs.to(_dev()),
                         gb.to(_dev()), we.to(_dev()), glu=1)
    out = ops.moe_down(act, dp.to(_dev()), ds.to(_dev()), db.to(_dev()),
                       we.to(_dev()))
    act_ref = ref.moe_gateup(x, gp, gs, gb, we, glu=1)
    out_ref = ref.moe_down(act_ref, dp, ds, db, we)
    # act rows are defined only where we[m, e] != 0 (the kernel skips the
    # rest per tile; moe_down never reads them)
    valid = we.t().bool()                      # [E, M]
    assert torch.allclose(act.cpu().float()[valid], act_ref.float()[valid],
                          atol=5e-2, rtol=3e-2)
    assert torch.allclose(out.cpu(), out_ref, atol=8e-2, rtol=3e-2)


@pytest.mark.parametrize("kvq", [False, True])
def test_fused_qkv_rope_matches_separate(kvq):
    """The split-k-fused qkv GEMM + RoPE + append (gemv_qkv_rope reading
    the f32 scratch) must match the separate gemv_int8 -> rope_append
    chain: same rotated q and identical KV cache rows."""
    torch.manual_seed(7)
    B, nq, nkv, d, K, smax = 8, 10, 2, 128, 1024, 64
    N = (nq + 2 * nkv) * d
    dev = _dev()
    y = torch.randn(B, K, dtype=torch.bfloat16, device=dev)
    wf = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 30
    q8, scales = ops.quantize_int8(wf, 128)
    qp = ops.pack_int8_mfma(q8)
    bias = torch.randn(N, dtype=torch.bfloat16, device=dev)
    cos, sin = ops.rope_tables(smax, d, 1e6, dev)
    pos = torch.full((B,), 17, dtype=torch.int32, device=dev)

    def mkkv():
        kdt = torch.int8 if kvq else torch.bfloat16
        kc = torch.zeros(B, nkv, smax, d, dtype=kdt, device=dev)
        vc = torch.zeros(B, nkv, smax, d, dtype=kdt, device=dev)
        ks = vs = None
        if kvq:
            ks = torch.zeros(B, nkv, smax, d // 64, dtype=torch.bfloat16,
                             device=dev)
            vs = torch.zeros_like(ks)
        return kc, vc, ks, vs

    # reference: separate kernels
    kc1, vc1, ks1, vs1 = mkkv()
    qkv = ops.gemv_int8(y, qp, scales, 128, bias, packed=True)
    qr = qkv[:, :nq * d].view(B, nq, d).clone()
    kr = qkv[:, nq * d:(nq + nkv) * d].view(B, nkv, d)
    vr = qkv[:, (nq + nkv) * d:].view(B, nkv, d)
    ops.rope_append(qr, kr, vr, kc1, vc1, pos, cos, sin, ks1, vs1)

    # fused path (forces the deferred branch for this shape; skip if the
    # split-k pick says no deferral would happen)
    from dnet_amd.ops import _will_defer, _get_scratch
    if not _will_defer(B, N, K, 128, 8, _get_scratch(dev).numel()):
        pytest.skip("shape does not split-k; fused path == separate path")
    kc2, vc2, ks2, vs2 = mkkv()
    qf = ops.gemv_qkv_rope(y, qp, scales, 128, 8, bias, nq, nkv, d,
                           kc2, vc2, pos, cos, sin, ks2, vs2)
    assert torch.allclose(qf.float().cpu(), qr.float().cpu(), atol=3e-2,
                          rtol=2e-2), (qf.float() - qr.float()).abs().max()
    # the fused path quantizes/rounds from the f32 scratch (more precise
    # than the bf16-rounded separate chain), so compare VALUES: dequant
    # the int8 cache with its scales first
    def dq(c, sc):
        if sc is None:
            return c.float()
        return c.float() * sc.float().repeat_interleave(64, dim=-1)
    assert torch.allclose(dq(kc1, ks1).cpu(), dq(kc2, ks2).cpu(), atol=4e-2,
                          rtol=3e-2)
    assert torch.allclose(dq(vc1, vs1).cpu(), dq(vc2, vs2).cpu(), atol=4e-2,
                          rtol=3e-2)


def test_deferred_combine_consumers_match_plain():
    """The split-k deferred-combine consumers (gemv_swiglu reading the
    f32 scratch, gemv_rmsnorm, and the resid-add) must match the plain
    GEMM -> combine -> op chain, and must leave the scratch re-zeroed
    for the next split-k user."""
    from dnet_amd.ops import _get_scratch, _will_defer, _native
    torch.manual_seed(11)
    dev = _dev()
    M, K, I = 16, 1024, 512
    N = 2 * I
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    wf = torch.randn(N, K, dtype=torch.bfloat16, device=dev) / 30
    q8, sc = ops.quantize_int8(wf, 128)
    qp = ops.pack_int8_mfma(q8)
    if not _will_defer(M, N, K, 128, 8, _get_scratch(dev).numel()):
        pytest.skip("shape does not split-k")

    # suite-order hazard: leave a LARGE stale combine region first (this
    # is exactly the case that broke the boolean dirty flag — a small
    # memset cleared it while stale partials remained beyond M*N)
    xb = torch.randn(64, K, dtype=torch.bfloat16, device=dev)
    wb = torch.randn(4096, K, dtype=torch.bfloat16, device=dev) / 30
    qb, sb = ops.quantize_int8(wb, 128)
    ops.gemv_int8(xb, ops.pack_int8_mfma(qb), sb, 128, None, packed=True)

    # plain chain
    gu = ops.gemv_int8(x, qp, sc, 128, None, packed=True)
    y_ref = ops.swiglu(gu)
    # fused chain (reads f32 pre-combine values — y_ref went through a
    # bf16 rounding at the combine, so tolerance covers one bf16 ulp)
    y_fused = ops.gemv_swiglu(x, qp, sc, 128, 8)
    assert torch.allclose(y_fused.float().cpu(), y_ref.float().cpu(),
                          atol=8e-2, rtol=4e-2)

    # rmsnorm variant: residual updated in place + y match
    wproj = torch.randn(K, K, dtype=torch.bfloat16, device=dev) / 30
    pq, psc = ops.quantize_int8(wproj, 128)
    pqp = ops.pack_int8_mfma(pq)
    wn = torch.randn(K, dtype=torch.bfloat16, device=dev)
    h1 = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    h2 = h1.clone()
    o = ops.gemv_int8(x, pqp, psc, 128, None, packed=True)
    y1 = ops.rmsnorm(o, h1, wn, 1e-6)
    y2 = ops.gemv_rmsnorm(x, pqp, psc, 128, 8, h2, wn, 1e-6)
    assert torch.allclose(y2.float().cpu(), y1.float().cpu(), atol=4e-2,
                          rtol=3e-2)
    assert torch.allclose(h2.float().cpu(), h1.float().cpu(), atol=4e-2,
                          rtol=3e-2)

    # resid-add variant: h += deferred projection
    h3, h4 = h1.clone(), h1.clone()
    d_ref = ops.gemv_int8(x, pqp, psc, 128, None, packed=True)
    h3.add_(d_ref)
    ok = ops.gemv_defer(x, pqp, psc, 128, 8)
    assert ok
    ops.resid_add_scratch(h4)
    assert torch.allclose(h4.float().cpu(), h3.float().cpu(), atol=4e-2,
                          rtol=3e-2)

    # all consumers must have re-zeroed what they read
    scr = _get_scratch(dev)
    assert int((scr != 0).sum()) == 0, "scratch not re-zeroed by consumers"
