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
