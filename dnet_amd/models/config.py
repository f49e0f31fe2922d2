"""Model configuration parsed from HF-style config.json dicts.

Covers the reference's model families (llama, qwen2/2.5, qwen3, mixtral,
gpt-oss, deepseek-v2 — reference: src/dnet/core/models/*) with one dataclass.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class QuantConfig:
    """Weight quantization: grouped symmetric int8/int4 (W8A16/W4A16), or
    OCP MXFP4 (fmt="mxfp4": e2m1 nibbles + e8m0 32-block scales — applied
    to MoE expert banks only, like gpt-oss checkpoints)."""
    bits: int = 8
    group: int = 128
    fmt: str = ""          # "" = grouped-int, "mxfp4"

    @property
    def name(self) -> str:
        if self.fmt == "mxfp4":
            return "mxfp4"
        return f"int{self.bits}-g{self.group}"


@dataclass
class ModelConfig:
    model_type: str = "llama"
    hidden_size: int = 4096
    num_layers: int = 32
    num_q_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    intermediate_size: int = 14336
    vocab_size: int = 128256
    rms_eps: float = 1e-5
    rope_theta: float = 500000.0
    rope_scaling: Optional[dict] = None
    tie_word_embeddings: bool = False
    attention_bias: bool = False        # qwen2-style qkv bias
    qk_norm: bool = False               # qwen3-style per-head q/k rmsnorm
    max_position_embeddings: int = 32768
    # MoE (mixtral / qwen-moe / gpt-oss)
    num_experts: int = 0
    num_experts_per_tok: int = 0
    moe_intermediate_size: int = 0
    # sliding-window attention (gpt-oss / mistral): 0 = disabled
    sliding_window: int = 0
    sliding_window_pattern: Optional[list] = None  # per-layer types
    # deepseek-v2 MLA
    q_lora_rank: int = 0
    kv_lora_rank: int = 0
    qk_nope_head_dim: int = 0
    qk_rope_head_dim: int = 0
    v_head_dim: int = 0
    first_k_dense_replace: int = 0
    n_shared_experts: int = 0
    shared_expert_intermediate_size: int = 0   # qwen2-moe gated shared expert
    scoring_func: str = "softmax"              # deepseek-v3: "sigmoid"
    n_group: int = 1                           # deepseek-v3 group-limited
    topk_group: int = 1                        #   routing (noaux_tc)
    routed_scaling_factor: float = 1.0
    norm_topk_prob: bool = False
    quant: Optional[QuantConfig] = None

    @property
    def qkv_out(self) -> int:
        return (self.num_q_heads + 2 * self.num_kv_heads) * self.head_dim

    @classmethod
    def from_hf(cls, cfg: dict, quant: Optional[QuantConfig] = None) -> "ModelConfig":
        mt = cfg.get("model_type", "llama")
        hidden = cfg["hidden_size"]
        nq = cfg.get("num_attention_heads", 32)
        hd = cfg.get("head_dim") or hidden // nq
        mc = cls(
            model_type=mt,
            hidden_size=hidden,
            num_layers=cfg.get("num_hidden_layers", 32),
            num_q_heads=nq,
            num_kv_heads=cfg.get("num_key_value_heads", nq),
            head_dim=hd,
            intermediate_size=cfg.get("intermediate_size", 4 * hidden),
            vocab_size=cfg.get("vocab_size", 32000),
            rms_eps=cfg.get("rms_norm_eps", 1e-5),
            rope_theta=(cfg.get("rope_theta")
                        or (cfg.get("rope_parameters") or {}).get("rope_theta")
                        or 10000.0),
            rope_scaling=cfg.get("rope_scaling") or cfg.get("rope_parameters"),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            attention_bias=cfg.get("attention_bias", mt == "qwen2"),
            qk_norm=mt in ("qwen3", "qwen3_moe"),
            max_position_embeddings=cfg.get("max_position_embeddings", 32768),
            sliding_window=(cfg.get("sliding_window") or 0
                            if cfg.get("use_sliding_window", True) else 0),
            quant=quant,
        )
        mc.sliding_window_pattern = cfg.get("layer_types")
        if mt in ("deepseek_v2", "deepseek_v3"):
            mc.q_lora_rank = cfg.get("q_lora_rank") or 0
            mc.kv_lora_rank = cfg.get("kv_lora_rank", 512)
            mc.qk_nope_head_dim = cfg.get("qk_nope_head_dim", 128)
            mc.qk_rope_head_dim = cfg.get("qk_rope_head_dim", 64)
            mc.v_head_dim = cfg.get("v_head_dim", 128)
            mc.first_k_dense_replace = cfg.get("first_k_dense_replace", 0)
            mc.n_shared_experts = cfg.get("n_shared_experts") or 0
            mc.routed_scaling_factor = cfg.get("routed_scaling_factor", 1.0)
            mc.norm_topk_prob = cfg.get("norm_topk_prob", False)
            mc.num_experts = cfg.get("n_routed_experts") or 0
            mc.num_experts_per_tok = cfg.get("num_experts_per_tok", 0)
            mc.moe_intermediate_size = cfg.get("moe_intermediate_size", 0)
            mc.scoring_func = cfg.get("scoring_func", "softmax")
            mc.n_group = cfg.get("n_group") or 1
            mc.topk_group = cfg.get("topk_group") or 1
        if mt in ("mixtral", "qwen2_moe", "qwen3_moe", "gpt_oss"):
            mc.num_experts = cfg.get("num_local_experts",
                                     cfg.get("num_experts", 8))
            mc.num_experts_per_tok = cfg.get("num_experts_per_tok", 2)
            mc.moe_intermediate_size = cfg.get("moe_intermediate_size",
                                               cfg.get("intermediate_size"))
            mc.norm_topk_prob = cfg.get("norm_topk_prob", False)
            mc.shared_expert_intermediate_size = cfg.get(
                "shared_expert_intermediate_size", 0)
        return mc


# Known configs for synthetic benchmarking (random-init weights; no network).
PRESETS: dict[str, dict] = {
    "llama-3-8b": dict(model_type="llama", hidden_size=4096, num_hidden_layers=32,
                       num_attention_heads=32, num_key_value_heads=8,
                       intermediate_size=14336, vocab_size=128256,
                       rope_theta=500000.0, rms_norm_eps=1e-5),
    "llama-3-70b": dict(model_type="llama", hidden_size=8192, num_hidden_layers=80,
                        num_attention_heads=64, num_key_value_heads=8,
                        intermediate_size=28672, vocab_size=128256,
                        rope_theta=500000.0, rms_norm_eps=1e-5),
    "qwen-2.5-32b": dict(model_type="qwen2", hidden_size=5120, num_hidden_layers=64,
                         num_attention_heads=40, num_key_value_heads=8,
                         intermediate_size=27648, vocab_size=152064,
                         rope_theta=1000000.0, rms_norm_eps=1e-5,
                         attention_bias=True),
    "qwen3-8b": dict(model_type="qwen3", hidden_size=4096, num_hidden_layers=36,
                     num_attention_heads=32, num_key_value_heads=8, head_dim=128,
                     intermediate_size=12288, vocab_size=151936,
                     rope_theta=1000000.0, rms_norm_eps=1e-6),
    "mixtral-8x7b": dict(model_type="mixtral", hidden_size=4096,
                         num_hidden_layers=32, num_attention_heads=32,
                         num_key_value_heads=8, intermediate_size=14336,
                         vocab_size=32000, rope_theta=1000000.0,
                         num_local_experts=8, num_experts_per_tok=2),
    "opt-125m-like": dict(model_type="llama", hidden_size=768, num_hidden_layers=12,
                          num_attention_heads=12, num_key_value_heads=12,
                          head_dim=64, intermediate_size=3072, vocab_size=50272,
                          rope_theta=10000.0),
    "gpt-oss-20b": dict(model_type="gpt_oss", hidden_size=2880,
                        num_hidden_layers=24, num_attention_heads=64,
                        num_key_value_heads=8, head_dim=64,
                        intermediate_size=2880, vocab_size=201088,
                        num_local_experts=32, num_experts_per_tok=4,
                        sliding_window=128, rope_theta=150000.0,
                        attention_bias=True, rms_norm_eps=1e-5),
    "deepseek-v2-lite": dict(model_type="deepseek_v2", hidden_size=2048,
                             num_hidden_layers=27, num_attention_heads=16,
                             num_key_value_heads=16, vocab_size=102400,
                             intermediate_size=10944, kv_lora_rank=512,
                             qk_nope_head_dim=128, qk_rope_head_dim=64,
                             v_head_dim=128, n_routed_experts=64,
                             num_experts_per_tok=6, n_shared_experts=2,
                             moe_intermediate_size=1408,
                             first_k_dense_replace=1,
                             routed_scaling_factor=1.0, rope_theta=10000.0),
    "deepseek-v3-lite": dict(model_type="deepseek_v3", hidden_size=2048,
                             num_hidden_layers=27, num_attention_heads=16,
                             num_key_value_heads=16, vocab_size=102400,
                             intermediate_size=10944, q_lora_rank=0,
                             kv_lora_rank=512, qk_nope_head_dim=128,
                             qk_rope_head_dim=64, v_head_dim=128,
                             n_routed_experts=64, num_experts_per_tok=6,
                             n_shared_experts=2, moe_intermediate_size=1408,
                             first_k_dense_replace=1, n_group=8,
                             topk_group=4, scoring_func="sigmoid",
                             norm_topk_prob=True,
                             routed_scaling_factor=2.5, rope_theta=10000.0),
    "tiny": dict(model_type="llama", hidden_size=128, num_hidden_layers=4,
                 num_attention_heads=2, num_key_value_heads=2, head_dim=64,
                 intermediate_size=256, vocab_size=256, rope_theta=10000.0),
}
