"""Model-level CPU tests: prefill/decode consistency and golden vs
transformers fp32 (the numerics acceptance for the whole layer math)."""
import pytest
import torch

from dnet_amd.models import KVCache, ModelConfig, PRESETS, QuantConfig, get_ring_model


def _tiny(quant=None, **over):
    cfg = dict(PRESETS["tiny"])
    cfg.update(over)
    return ModelConfig.from_hf(cfg, quant=quant)


def _build(cfg, smax=64, batch=2):
    cls = get_ring_model(cfg.model_type)
    m = cls(cfg, range(cfg.num_layers), "cpu", True, True, smax=smax)
    m.init_random(0)
    kv = KVCache(cfg, range(cfg.num_layers), batch, smax, "cpu")
    return m, kv


@pytest.mark.parametrize("quant", [None, QuantConfig(8, 64)])
def test_prefill_decode_consistency(quant):
    torch.manual_seed(0)
    cfg = _tiny(quant=quant)
    m, kv = _build(cfg)
    B, T = 2, 9
    tokens = torch.randint(0, cfg.vocab_size, (B, T))

    # Path A: full prefill
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    logits_a = m.normalize_project(h[:, -1].contiguous())

    # Path B: prefill T-1, decode last token
    kv2 = KVCache(cfg, range(cfg.num_layers), B, 64, "cpu")
    h2 = m.embed_tokens(tokens[:, :-1]).clone()
    m.prefill_window(h2, m.layer_ids, kv2, 0)
    kv2.pos.fill_(T - 1)
    hd = m.embed_tokens(tokens[:, -1]).clone()
    m.decode_window(hd, m.layer_ids, kv2)
    logits_b = m.normalize_project(hd)

    a, b = logits_a.float(), logits_b.float()
    cos = torch.nn.functional.cosine_similarity(a, b, dim=-1)
    assert (cos > 0.995).all(), f"prefill/decode disagree: cos={cos}"
    assert (a.argmax(-1) == b.argmax(-1)).float().mean() > 0.99


def test_vs_transformers_llama():
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(1)
    tc = transformers.LlamaConfig(
        hidden_size=128, intermediate_size=256, num_hidden_layers=3,
        num_attention_heads=2, num_key_value_heads=2, head_dim=64,
        vocab_size=256, rope_theta=10000.0, rms_norm_eps=1e-5,
        attention_bias=False, tie_word_embeddings=False,
        max_position_embeddings=128)
    hf = transformers.LlamaForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    m = get_ring_model(cfg.model_type)(cfg, range(cfg.num_layers), "cpu",
                                       True, True, smax=64)
    m.load_state_dict({k: v for k, v in hf.state_dict().items()})
    kv = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")

    tokens = torch.randint(0, 256, (1, 12))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.99).all(), f"vs transformers: cos={cos}"


def test_moe_runs():
    torch.manual_seed(2)
    cfg = _tiny(model_type="mixtral", num_local_experts=4, num_experts_per_tok=2)
    m, kv = _build(cfg)
    B, T = 2, 5
    tokens = torch.randint(0, cfg.vocab_size, (B, T))
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    kv.pos.fill_(T)
    hd = m.embed_tokens(tokens[:, -1]).clone()
    m.decode_window(hd, m.layer_ids, kv)
    logits = m.normalize_project(hd)
    assert torch.isfinite(logits.float()).all()


def test_vs_transformers_gpt_oss():
    transformers = pytest.importorskip("transformers")
    if not hasattr(transformers, "GptOssForCausalLM"):
        pytest.skip("no gpt_oss in transformers")
    import warnings
    warnings.filterwarnings("ignore")
    torch.manual_seed(5)
    tc = transformers.GptOssConfig(
        hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=64, intermediate_size=64,
        num_local_experts=4, num_experts_per_tok=2, vocab_size=128,
        sliding_window=8, max_position_embeddings=64, tie_word_embeddings=False)
    hf = transformers.GptOssForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    assert cfg.sliding_window == 8 and cfg.num_experts == 4
    m = get_ring_model("gpt_oss")(cfg, range(cfg.num_layers), "cpu",
                                  True, True, smax=64)
    m.load_state_dict(dict(hf.state_dict()))
    kv = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")
    tokens = torch.randint(0, 128, (1, 20))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.98).all(), f"gpt_oss vs transformers: cos={cos}"
    # decode path consistency: prefill 19 + decode 1 == prefill 20
    kv2 = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")
    h2 = m.embed_tokens(tokens[:, :-1]).clone()
    m.prefill_window(h2, m.layer_ids, kv2, 0)
    kv2.pos.fill_(19)
    hd = m.embed_tokens(tokens[:, -1]).clone()
    m.decode_window(hd, m.layer_ids, kv2)
    dec = m.normalize_project(hd).float()
    cos2 = torch.nn.functional.cosine_similarity(dec, ours, dim=-1)
    assert (cos2 > 0.995).all(), f"gpt_oss decode vs prefill: cos={cos2}"


def test_vs_transformers_deepseek_v2():
    transformers = pytest.importorskip("transformers")
    if not hasattr(transformers, "DeepseekV2ForCausalLM"):
        pytest.skip("no deepseek_v2 in transformers")
    import warnings
    warnings.filterwarnings("ignore")
    torch.manual_seed(6)
    tc = transformers.DeepseekV2Config(
        hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=4, vocab_size=128, intermediate_size=96,
        q_lora_rank=None, kv_lora_rank=32, qk_nope_head_dim=32,
        qk_rope_head_dim=16, v_head_dim=32, n_routed_experts=4,
        num_experts_per_tok=2, n_shared_experts=1, moe_intermediate_size=32,
        first_k_dense_replace=1, topk_method="greedy", n_group=1,
        topk_group=1, max_position_embeddings=64, tie_word_embeddings=False)
    hf = transformers.DeepseekV2ForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    assert cfg.kv_lora_rank == 32 and cfg.num_experts == 4
    m = get_ring_model("deepseek_v2")(cfg, range(cfg.num_layers), "cpu",
                                      True, True, smax=64)
    m.load_state_dict(dict(hf.state_dict()))
    kv = m.make_kv_cache(1, 64)
    tokens = torch.randint(0, 128, (1, 14))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.98).all(), f"deepseek_v2 vs transformers: cos={cos}"
    # decode-vs-prefill consistency
    kv2 = m.make_kv_cache(1, 64)
    h2 = m.embed_tokens(tokens[:, :-1]).clone()
    m.prefill_window(h2, m.layer_ids, kv2, 0)
    kv2.pos.fill_(13)
    hd = m.embed_tokens(tokens[:, -1]).clone()
    m.decode_window(hd, m.layer_ids, kv2)
    dec = m.normalize_project(hd).float()
    cos2 = torch.nn.functional.cosine_similarity(dec, ours, dim=-1)
    assert (cos2 > 0.995).all(), f"deepseek decode vs prefill: cos={cos2}"


def test_kv_cache_quantized_close_to_fp():
    """int8 group-64 KV cache tracks the bf16 cache closely (CPU ref)."""
    torch.manual_seed(8)
    cfg = _tiny()
    m, _ = _build(cfg)
    toks = torch.randint(0, cfg.vocab_size, (2, 10))

    def run(bits):
        m.kv_bits = bits
        kv = m.make_kv_cache(2, 64)
        h = m.embed_tokens(toks).clone()
        m.prefill_window(h, m.layer_ids, kv, 0)
        kv.pos.fill_(10)
        hd = m.embed_tokens(toks[:, -1]).clone()
        m.decode_window(hd, m.layer_ids, kv)
        return m.normalize_project(hd).float()

    a, b = run(16), run(8)
    cos = torch.nn.functional.cosine_similarity(a, b, dim=-1)
    assert (cos > 0.99).all(), f"kv8 vs kv16: {cos}"


def test_mxfp4_dequant_and_gpt_oss_load():
    """MXFP4 (E2M1 nibbles + E8M0 block scales) dequant, and the gpt-oss
    loader accepting *_blocks/*_scales checkpoints (reference: MXFP4 weight
    sanitization in src/dnet/core/models/gpt_oss.py)."""
    import dnet_amd.ops as ops
    from dnet_amd.ops.reference import _MXFP4_LUT

    g = torch.Generator().manual_seed(0)

    def encode(target_idx, scales_exp):
        # target_idx: [.., B, 32] LUT indices; scales_exp: [.., B] exponents
        lo, hi = target_idx[..., 0::2], target_idx[..., 1::2]
        blocks = (lo | (hi << 4)).to(torch.uint8)
        scales = (scales_exp + 127).to(torch.uint8)
        lut = torch.tensor(_MXFP4_LUT)
        expect = (lut[target_idx] *
                  torch.pow(2.0, scales_exp.float()).unsqueeze(-1))
        return blocks, scales, expect.reshape(*target_idx.shape[:-2], -1)

    idx = torch.randint(0, 16, (3, 4, 2, 32), generator=g)
    exp = torch.randint(-3, 4, (3, 4, 2), generator=g)
    blocks, scales, expect = encode(idx, exp)
    got = ops.mxfp4_dequant(blocks, scales)
    assert got.dtype == torch.bfloat16
    assert torch.equal(got.float(), expect)

    # loader: bf16 checkpoint vs the same weights as MXFP4 blocks
    from dnet_amd.models import ModelConfig
    from dnet_amd.models.gpt_oss import GptOssRingModel
    hf = dict(model_type="gpt_oss", hidden_size=64, num_hidden_layers=1,
              num_attention_heads=2, num_key_value_heads=1, head_dim=32,
              intermediate_size=64, vocab_size=64, num_local_experts=2,
              num_experts_per_tok=1, sliding_window=16, rope_theta=10000.0,
              attention_bias=True, rms_norm_eps=1e-5)
    cfg = ModelConfig.from_hf(hf)
    E, H, I2 = 2, 64, 128

    def rnd(*s):
        return torch.randn(*s, generator=g, dtype=torch.float32).to(
            torch.bfloat16)

    gi = torch.randint(0, 16, (E, I2, H // 32, 32), generator=g)
    ge = torch.randint(-2, 3, (E, I2, H // 32), generator=g)
    gub, gus, gu_vals = encode(gi, ge)     # gu_vals [E, 2I, H] output-major
    di = torch.randint(0, 16, (E, H, (I2 // 2) // 32, 32), generator=g)
    de = torch.randint(-2, 3, (E, H, (I2 // 2) // 32), generator=g)
    dnb_, dns, dn_vals = encode(di, de)    # [E, H, I]
    base = {
        "model.layers.0.input_layernorm.weight": rnd(64),
        "model.layers.0.post_attention_layernorm.weight": rnd(64),
        "model.layers.0.self_attn.q_proj.weight": rnd(64, 64),
        "model.layers.0.self_attn.k_proj.weight": rnd(32, 64),
        "model.layers.0.self_attn.v_proj.weight": rnd(32, 64),
        "model.layers.0.self_attn.q_proj.bias": rnd(64),
        "model.layers.0.self_attn.k_proj.bias": rnd(32),
        "model.layers.0.self_attn.v_proj.bias": rnd(32),
        "model.layers.0.self_attn.o_proj.weight": rnd(64, 64),
        "model.layers.0.self_attn.o_proj.bias": rnd(64),
        "model.layers.0.self_attn.sinks": rnd(2),
        "model.layers.0.mlp.router.weight": rnd(E, 64),
        "model.layers.0.mlp.router.bias": rnd(E),
        "model.layers.0.mlp.experts.gate_up_proj_bias": rnd(E, I2),
        "model.layers.0.mlp.experts.down_proj_bias": rnd(E, H),
        "model.embed_tokens.weight": rnd(64, 64),
        "model.norm.weight": rnd(64),
    }
    sd_bf = dict(base)
    sd_bf["model.layers.0.mlp.experts.gate_up_proj"] = \
        gu_vals.to(torch.bfloat16).transpose(1, 2).contiguous()  # [E,H,2I]
    sd_bf["model.layers.0.mlp.experts.down_proj"] = \
        dn_vals.to(torch.bfloat16).transpose(1, 2).contiguous()  # [E,I,H]
    sd_mx = dict(base)
    sd_mx["model.layers.0.mlp.experts.gate_up_proj_blocks"] = gub
    sd_mx["model.layers.0.mlp.experts.gate_up_proj_scales"] = gus
    sd_mx["model.layers.0.mlp.experts.down_proj_blocks"] = dnb_
    sd_mx["model.layers.0.mlp.experts.down_proj_scales"] = dns

    m1 = GptOssRingModel(cfg, [0], "cpu", True, True)
    m1.load_state_dict(sd_bf)
    # native path (default): experts stay PACKED (~4.25 bit/weight
    # resident), dequant is fused in the kernels / done transiently
    m2 = GptOssRingModel(cfg, [0], "cpu", True, True)
    m2.load_state_dict(sd_mx)
    import dnet_amd.ops as dops
    for e in range(E):
        eg = m2.layers[0].experts_gateup[e]
        ed = m2.layers[0].experts_down[e]
        assert eg.mxfp4 and eg.w.dtype == torch.uint8
        assert torch.equal(m1.layers[0].experts_gateup[e].w,
                           dops.dequant_mxfp4(eg.w, eg.scales))
        assert torch.equal(m1.layers[0].experts_down[e].w,
                           dops.dequant_mxfp4(ed.w, ed.scales))
    # resident footprint: nibbles + 1 scale byte / 32 weights
    assert (m2.layers[0].experts_gateup[0].nbytes()
            < m1.layers[0].experts_gateup[0].nbytes() / 3)

    # golden: native execution == dequantize-at-load execution
    import os
    os.environ["DNET_MXFP4_DEQUANT"] = "1"
    try:
        m3 = GptOssRingModel(cfg, [0], "cpu", True, True)
        m3.load_state_dict(sd_mx)
    finally:
        os.environ.pop("DNET_MXFP4_DEQUANT", None)
    from dnet_amd.models import KVCache
    toks = torch.randint(0, 64, (1, 5), generator=g)
    for m in (m2, m3):
        m.final_norm = m1.final_norm
        m.lm_head = m1.lm_head
        m.embed = m1.embed
    def logits_of(m):
        kv = m.make_kv_cache(1, 16)
        h = m.embed_tokens(toks)
        m.prefill_window(h, [0], kv, 0)
        return m.normalize_project(h[:, -1].contiguous())
    l_native, l_deq = logits_of(m2), logits_of(m3)
    assert torch.allclose(l_native.float(), l_deq.float(), atol=2e-2,
                          rtol=2e-2), (l_native - l_deq).abs().max()


def test_vs_transformers_mixtral():
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(4)
    tc = transformers.MixtralConfig(
        hidden_size=128, intermediate_size=256, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, head_dim=32,
        vocab_size=256, rope_theta=10000.0, rms_norm_eps=1e-5,
        num_local_experts=4, num_experts_per_tok=2,
        tie_word_embeddings=False, max_position_embeddings=128)
    hf = transformers.MixtralForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    m = get_ring_model(cfg.model_type)(cfg, range(cfg.num_layers), "cpu",
                                       True, True, smax=64)
    m.load_state_dict({k: v for k, v in hf.state_dict().items()})
    kv = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")

    tokens = torch.randint(0, 256, (1, 10))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.99).all(), f"vs transformers mixtral: cos={cos}"


def test_vs_transformers_qwen2_moe():
    """Qwen2-MoE: softmax-before-topk routing + sigmoid-gated shared
    expert (differs from mixtral on both counts)."""
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(6)
    tc = transformers.Qwen2MoeConfig(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, vocab_size=256,
        rope_theta=10000.0, rms_norm_eps=1e-5, num_experts=4,
        num_experts_per_tok=2, moe_intermediate_size=48,
        shared_expert_intermediate_size=96, norm_topk_prob=True,
        decoder_sparse_step=1, tie_word_embeddings=False,
        max_position_embeddings=128)
    hf = transformers.Qwen2MoeForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    m = get_ring_model(cfg.model_type)(cfg, range(cfg.num_layers), "cpu",
                                       True, True, smax=64)
    m.load_state_dict({k: v for k, v in hf.state_dict().items()})
    kv = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")

    tokens = torch.randint(0, 256, (1, 9))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.99).all(), f"vs transformers qwen2-moe: cos={cos}"


def test_vs_transformers_deepseek_v3():
    """DeepSeek-V3: sigmoid scoring + e_score_correction_bias +
    group-limited top-k routing (noaux_tc) — differs from v2's softmax."""
    transformers = pytest.importorskip("transformers")
    if not hasattr(transformers, "DeepseekV3ForCausalLM"):
        pytest.skip("no deepseek_v3 in transformers")
    torch.manual_seed(8)
    tc = transformers.DeepseekV3Config(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=4, vocab_size=256,
        q_lora_rank=48, kv_lora_rank=32, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, n_routed_experts=8,
        num_experts_per_tok=2, n_shared_experts=1, n_group=2, topk_group=1,
        moe_intermediate_size=32, first_k_dense_replace=1,
        routed_scaling_factor=1.5, norm_topk_prob=True,
        rope_interleave=False,
        scoring_func="sigmoid", rope_theta=10000.0, rms_norm_eps=1e-5,
        tie_word_embeddings=False, max_position_embeddings=128)
    hf = transformers.DeepseekV3ForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    m = get_ring_model(cfg.model_type)(cfg, range(cfg.num_layers), "cpu",
                                       True, True, smax=64)
    m.load_state_dict({k: v for k, v in hf.state_dict().items()})
    kv = m.make_kv_cache(1, 64)

    tokens = torch.randint(0, 256, (1, 8))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.99).all(), f"vs transformers deepseek-v3: cos={cos}"


def test_vs_transformers_qwen3_moe():
    transformers = pytest.importorskip("transformers")
    if not hasattr(transformers, "Qwen3MoeForCausalLM"):
        pytest.skip("no qwen3_moe in transformers")
    torch.manual_seed(9)
    tc = transformers.Qwen3MoeConfig(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, head_dim=16,
        vocab_size=256, rope_theta=10000.0, rms_norm_eps=1e-5,
        num_experts=4, num_experts_per_tok=2, moe_intermediate_size=48,
        norm_topk_prob=True, decoder_sparse_step=1,
        tie_word_embeddings=False, max_position_embeddings=128)
    hf = transformers.Qwen3MoeForCausalLM(tc).eval().float()
    cfg = ModelConfig.from_hf(tc.to_dict())
    m = get_ring_model(cfg.model_type)(cfg, range(cfg.num_layers), "cpu",
                                       True, True, smax=64)
    m.load_state_dict({k: v for k, v in hf.state_dict().items()})
    kv = KVCache(cfg, range(cfg.num_layers), 1, 64, "cpu")

    tokens = torch.randint(0, 256, (1, 8))
    with torch.no_grad():
        ref_logits = hf(tokens).logits[:, -1].float()
    h = m.embed_tokens(tokens).clone()
    m.prefill_window(h, m.layer_ids, kv, 0)
    ours = m.normalize_project(h[:, -1].contiguous()).float()
    cos = torch.nn.functional.cosine_similarity(ours, ref_logits, dim=-1)
    assert (cos > 0.99).all(), f"vs transformers qwen3-moe: cos={cos}"
