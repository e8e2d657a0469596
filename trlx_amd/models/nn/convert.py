"""HF transformers interop: config + state-dict conversion both ways.

The framework's checkpoint contract (BASELINE.json north star: "trlx.train()
API and checkpoint format") is the HF directory format — ``save_pretrained``
writes a directory loadable by vanilla transformers, and ``from_pretrained``
reads one.  transformers is used ONLY here (weights/config IO); the compute
path is the native CausalTransformer.

Supported families (round 1): gpt2, llama, gptj, gpt_neox, opt.
"""

import json
import os
from typing import Dict, Optional, Tuple

import torch

from .config import PRESETS, TransformerConfig


# ---------------------------------------------------------------------------
# HF config -> TransformerConfig
# ---------------------------------------------------------------------------


def config_from_hf(hf: dict) -> TransformerConfig:
    """Build a TransformerConfig from an HF config.json dict."""
    model_type = hf.get("model_type", "gpt2")
    if model_type == "gpt2":
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["n_embd"],
            num_layers=hf["n_layer"],
            num_heads=hf["n_head"],
            intermediate_size=hf.get("n_inner") or 4 * hf["n_embd"],
            max_position_embeddings=hf["n_positions"],
            norm="layernorm",
            norm_eps=hf.get("layer_norm_epsilon", 1e-5),
            position_encoding="learned",
            activation={"gelu_new": "gelu_new", "gelu": "gelu", "relu": "relu",
                        "gelu_pytorch_tanh": "gelu_new"}.get(hf.get("activation_function", "gelu_new"), "gelu_new"),
            embd_pdrop=hf.get("embd_pdrop", 0.0),
            resid_pdrop=hf.get("resid_pdrop", 0.0),
            attn_pdrop=hf.get("attn_pdrop", 0.0),
            tie_word_embeddings=hf.get("tie_word_embeddings", True),
            arch_name="gpt2",
        )
    if model_type == "llama":
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            num_layers=hf["num_hidden_layers"],
            num_heads=hf["num_attention_heads"],
            num_kv_heads=hf.get("num_key_value_heads", hf["num_attention_heads"]),
            intermediate_size=hf["intermediate_size"],
            max_position_embeddings=hf["max_position_embeddings"],
            norm="rmsnorm",
            norm_eps=hf.get("rms_norm_eps", 1e-5),
            position_encoding="rope",
            rope_base=hf.get("rope_theta", 10000.0),
            activation="silu",
            swiglu=True,
            attn_bias=hf.get("attention_bias", False),
            mlp_bias=hf.get("mlp_bias", False),
            tie_word_embeddings=hf.get("tie_word_embeddings", False),
            arch_name="llama",
        )
    if model_type == "gptj":
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["n_embd"],
            num_layers=hf["n_layer"],
            num_heads=hf["n_head"],
            intermediate_size=hf.get("n_inner") or 4 * hf["n_embd"],
            max_position_embeddings=hf["n_positions"],
            norm="layernorm",
            norm_eps=hf.get("layer_norm_epsilon", 1e-5),
            position_encoding="rope",
            rope_interleaved=True,
            rope_pct=hf.get("rotary_dim", hf["n_embd"] // hf["n_head"]) / (hf["n_embd"] // hf["n_head"]),
            activation="gelu_new",
            parallel_residual=True,
            attn_bias=False,
            mlp_bias=True,
            tie_word_embeddings=hf.get("tie_word_embeddings", False),
            lm_head_bias=True,
            embd_pdrop=hf.get("embd_pdrop", 0.0),
            resid_pdrop=hf.get("resid_pdrop", 0.0),
            attn_pdrop=hf.get("attn_pdrop", 0.0),
            arch_name="gptj",
        )
    if model_type == "gpt_neox":
        head_dim = hf["hidden_size"] // hf["num_attention_heads"]
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            num_layers=hf["num_hidden_layers"],
            num_heads=hf["num_attention_heads"],
            intermediate_size=hf["intermediate_size"],
            max_position_embeddings=hf["max_position_embeddings"],
            norm="layernorm",
            norm_eps=hf.get("layer_norm_eps", 1e-5),
            position_encoding="rope",
            rope_interleaved=False,
            rope_pct=hf.get("rotary_pct", 0.25),
            rope_base=hf.get("rotary_emb_base", 10000.0),
            activation={"gelu": "gelu", "gelu_new": "gelu_new"}.get(hf.get("hidden_act", "gelu"), "gelu"),
            parallel_residual=hf.get("use_parallel_residual", True),
            attn_bias=True,
            mlp_bias=True,
            tie_word_embeddings=hf.get("tie_word_embeddings", False),
            arch_name="gpt_neox",
        )
    if model_type == "opt":
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            num_layers=hf["num_hidden_layers"],
            num_heads=hf["num_attention_heads"],
            intermediate_size=hf["ffn_dim"],
            max_position_embeddings=hf["max_position_embeddings"],
            norm="layernorm",
            position_encoding="learned",
            activation={"relu": "relu", "gelu": "gelu"}.get(hf.get("activation_function", "relu"), "relu"),
            tie_word_embeddings=hf.get("tie_word_embeddings", True),
            arch_name="opt",
            extra={"position_offset": 2},
        )
    if model_type == "bloom":
        H = hf["n_head"]
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf.get("hidden_size", hf.get("n_embd")),
            num_layers=hf.get("n_layer", hf.get("num_hidden_layers")),
            num_heads=H,
            max_position_embeddings=hf.get("seq_length", 2048),
            norm="layernorm",
            norm_eps=hf.get("layer_norm_epsilon", 1e-5),
            position_encoding="alibi",
            activation="gelu_new",
            tie_word_embeddings=True,
            arch_name="bloom",
            extra={"pre_embed_norm": True},
        )
    if model_type == "gpt_bigcode":
        return TransformerConfig(
            vocab_size=hf["vocab_size"],
            hidden_size=hf["n_embd"],
            num_layers=hf["n_layer"],
            num_heads=hf["n_head"],
            num_kv_heads=1 if hf.get("multi_query", True) else hf["n_head"],
            intermediate_size=hf.get("n_inner") or 4 * hf["n_embd"],
            max_position_embeddings=hf["n_positions"],
            norm="layernorm",
            norm_eps=hf.get("layer_norm_epsilon", 1e-5),
            position_encoding="learned",
            activation={"gelu_new": "gelu_new", "gelu": "gelu",
                        "gelu_pytorch_tanh": "gelu_new"}.get(hf.get("activation_function", "gelu_pytorch_tanh"), "gelu_new"),
            tie_word_embeddings=hf.get("tie_word_embeddings", True),
            arch_name="gpt_bigcode",
        )
    raise ValueError(f"Unsupported HF model_type for the native transformer: {model_type}")


def config_to_hf(cfg: TransformerConfig) -> dict:
    """TransformerConfig -> HF config.json dict (inverse of config_from_hf)."""
    if cfg.arch_name == "gpt2":
        return {
            "model_type": "gpt2",
            "architectures": ["GPT2LMHeadModel"],
            "vocab_size": cfg.vocab_size,
            "n_embd": cfg.hidden_size,
            "n_layer": cfg.num_layers,
            "n_head": cfg.num_heads,
            "n_inner": cfg.intermediate_size if cfg.intermediate_size != 4 * cfg.hidden_size else None,
            "n_positions": cfg.max_position_embeddings,
            "n_ctx": cfg.max_position_embeddings,
            "layer_norm_epsilon": cfg.norm_eps,
            "activation_function": cfg.activation,
            "embd_pdrop": cfg.embd_pdrop,
            "resid_pdrop": cfg.resid_pdrop,
            "attn_pdrop": cfg.attn_pdrop,
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    if cfg.arch_name == "llama":
        return {
            "model_type": "llama",
            "architectures": ["LlamaForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "num_hidden_layers": cfg.num_layers,
            "num_attention_heads": cfg.num_heads,
            "num_key_value_heads": cfg.num_kv_heads,
            "intermediate_size": cfg.intermediate_size,
            "max_position_embeddings": cfg.max_position_embeddings,
            "rms_norm_eps": cfg.norm_eps,
            "rope_theta": cfg.rope_base,
            "hidden_act": "silu",
            "attention_bias": cfg.attn_bias,
            "mlp_bias": cfg.mlp_bias,
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    if cfg.arch_name == "gptj":
        return {
            "model_type": "gptj",
            "architectures": ["GPTJForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "n_embd": cfg.hidden_size,
            "n_layer": cfg.num_layers,
            "n_head": cfg.num_heads,
            "n_inner": None,
            "n_positions": cfg.max_position_embeddings,
            "rotary_dim": int(cfg.head_dim * cfg.rope_pct),
            "layer_norm_epsilon": cfg.norm_eps,
            "activation_function": "gelu_new",
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    if cfg.arch_name == "gpt_neox":
        return {
            "model_type": "gpt_neox",
            "architectures": ["GPTNeoXForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "num_hidden_layers": cfg.num_layers,
            "num_attention_heads": cfg.num_heads,
            "intermediate_size": cfg.intermediate_size,
            "max_position_embeddings": cfg.max_position_embeddings,
            "layer_norm_eps": cfg.norm_eps,
            "rotary_pct": cfg.rope_pct,
            "rotary_emb_base": cfg.rope_base,
            "hidden_act": cfg.activation,
            "use_parallel_residual": cfg.parallel_residual,
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    if cfg.arch_name == "bloom":
        return {
            "model_type": "bloom",
            "architectures": ["BloomForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "n_layer": cfg.num_layers,
            "n_head": cfg.num_heads,
            "layer_norm_epsilon": cfg.norm_eps,
            "tie_word_embeddings": True,
        }
    if cfg.arch_name == "gpt_bigcode":
        return {
            "model_type": "gpt_bigcode",
            "architectures": ["GPTBigCodeForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "n_embd": cfg.hidden_size,
            "n_layer": cfg.num_layers,
            "n_head": cfg.num_heads,
            "n_inner": cfg.intermediate_size,
            "n_positions": cfg.max_position_embeddings,
            "layer_norm_epsilon": cfg.norm_eps,
            "activation_function": "gelu_pytorch_tanh",
            "multi_query": cfg.num_kv_heads == 1,
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    if cfg.arch_name == "opt":
        return {
            "model_type": "opt",
            "architectures": ["OPTForCausalLM"],
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "num_hidden_layers": cfg.num_layers,
            "num_attention_heads": cfg.num_heads,
            "ffn_dim": cfg.intermediate_size,
            "max_position_embeddings": cfg.max_position_embeddings,
            "activation_function": cfg.activation,
            "do_layer_norm_before": True,
            "word_embed_proj_dim": cfg.hidden_size,
            "tie_word_embeddings": cfg.tie_word_embeddings,
        }
    raise ValueError(f"Unsupported arch for HF export: {cfg.arch_name}")


# ---------------------------------------------------------------------------
# HF state dict -> native state dict (and back)
# ---------------------------------------------------------------------------


def _cat_qkv(q, k, v):
    return torch.cat([q, k, v], dim=0)


def state_dict_from_hf(cfg: TransformerConfig, hf: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map an HF checkpoint state dict onto the native module names."""
    out = {}
    a = cfg.arch_name
    L = cfg.num_layers
    if a == "gpt2":
        out["embed_tokens.weight"] = hf["transformer.wte.weight"]
        out["embed_positions.weight"] = hf["transformer.wpe.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "ln_1.weight"]
            out[o + "ln_1.bias"] = hf[p + "ln_1.bias"]
            out[o + "ln_2.weight"] = hf[p + "ln_2.weight"]
            out[o + "ln_2.bias"] = hf[p + "ln_2.bias"]
            # HF GPT-2 Conv1D weights are [in, out] -> transpose
            out[o + "attn.qkv_proj.weight"] = hf[p + "attn.c_attn.weight"].t().contiguous()
            out[o + "attn.qkv_proj.bias"] = hf[p + "attn.c_attn.bias"]
            out[o + "attn.o_proj.weight"] = hf[p + "attn.c_proj.weight"].t().contiguous()
            out[o + "attn.o_proj.bias"] = hf[p + "attn.c_proj.bias"]
            out[o + "mlp.fc_in.weight"] = hf[p + "mlp.c_fc.weight"].t().contiguous()
            out[o + "mlp.fc_in.bias"] = hf[p + "mlp.c_fc.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.c_proj.weight"].t().contiguous()
            out[o + "mlp.down_proj.bias"] = hf[p + "mlp.c_proj.bias"]
        out["final_norm.weight"] = hf["transformer.ln_f.weight"]
        out["final_norm.bias"] = hf["transformer.ln_f.bias"]
        if not cfg.tie_word_embeddings and "lm_head.weight" in hf:
            out["lm_head.weight"] = hf["lm_head.weight"]
    elif a == "llama":
        out["embed_tokens.weight"] = hf["model.embed_tokens.weight"]
        for i in range(L):
            p = f"model.layers.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "input_layernorm.weight"]
            out[o + "ln_2.weight"] = hf[p + "post_attention_layernorm.weight"]
            out[o + "attn.qkv_proj.weight"] = _cat_qkv(
                hf[p + "self_attn.q_proj.weight"], hf[p + "self_attn.k_proj.weight"],
                hf[p + "self_attn.v_proj.weight"])
            if cfg.attn_bias:
                out[o + "attn.qkv_proj.bias"] = _cat_qkv(
                    hf[p + "self_attn.q_proj.bias"], hf[p + "self_attn.k_proj.bias"],
                    hf[p + "self_attn.v_proj.bias"])
            out[o + "attn.o_proj.weight"] = hf[p + "self_attn.o_proj.weight"]
            out[o + "mlp.gate_up_proj.weight"] = torch.cat(
                [hf[p + "mlp.gate_proj.weight"], hf[p + "mlp.up_proj.weight"]], dim=0)
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.down_proj.weight"]
        out["final_norm.weight"] = hf["model.norm.weight"]
        if not cfg.tie_word_embeddings:
            out["lm_head.weight"] = hf["lm_head.weight"]
    elif a == "gptj":
        out["embed_tokens.weight"] = hf["transformer.wte.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "ln_1.weight"]
            out[o + "ln_1.bias"] = hf[p + "ln_1.bias"]
            out[o + "attn.qkv_proj.weight"] = _cat_qkv(
                hf[p + "attn.q_proj.weight"], hf[p + "attn.k_proj.weight"],
                hf[p + "attn.v_proj.weight"])
            out[o + "attn.o_proj.weight"] = hf[p + "attn.out_proj.weight"]
            out[o + "mlp.fc_in.weight"] = hf[p + "mlp.fc_in.weight"]
            out[o + "mlp.fc_in.bias"] = hf[p + "mlp.fc_in.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.fc_out.weight"]
            out[o + "mlp.down_proj.bias"] = hf[p + "mlp.fc_out.bias"]
        out["final_norm.weight"] = hf["transformer.ln_f.weight"]
        out["final_norm.bias"] = hf["transformer.ln_f.bias"]
        out["lm_head.weight"] = hf["lm_head.weight"]
        if "lm_head.bias" in hf:
            out["lm_head.bias"] = hf["lm_head.bias"]
    elif a == "gpt_neox":
        H, D = cfg.num_heads, cfg.head_dim
        out["embed_tokens.weight"] = hf["gpt_neox.embed_in.weight"]
        for i in range(L):
            p = f"gpt_neox.layers.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "input_layernorm.weight"]
            out[o + "ln_1.bias"] = hf[p + "input_layernorm.bias"]
            out[o + "ln_2.weight"] = hf[p + "post_attention_layernorm.weight"]
            out[o + "ln_2.bias"] = hf[p + "post_attention_layernorm.bias"]
            # NeoX fuses qkv per-head [H, 3, D, hidden] -> regroup to [3, H, D]
            w = hf[p + "attention.query_key_value.weight"].view(H, 3, D, -1)
            out[o + "attn.qkv_proj.weight"] = w.permute(1, 0, 2, 3).reshape(3 * H * D, -1).contiguous()
            b = hf[p + "attention.query_key_value.bias"].view(H, 3, D)
            out[o + "attn.qkv_proj.bias"] = b.permute(1, 0, 2).reshape(3 * H * D).contiguous()
            out[o + "attn.o_proj.weight"] = hf[p + "attention.dense.weight"]
            out[o + "attn.o_proj.bias"] = hf[p + "attention.dense.bias"]
            out[o + "mlp.fc_in.weight"] = hf[p + "mlp.dense_h_to_4h.weight"]
            out[o + "mlp.fc_in.bias"] = hf[p + "mlp.dense_h_to_4h.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.dense_4h_to_h.weight"]
            out[o + "mlp.down_proj.bias"] = hf[p + "mlp.dense_4h_to_h.bias"]
        out["final_norm.weight"] = hf["gpt_neox.final_layer_norm.weight"]
        out["final_norm.bias"] = hf["gpt_neox.final_layer_norm.bias"]
        # transformers >=5 uses lm_head.weight; older NeoX exports embed_out
        out["lm_head.weight"] = hf.get("lm_head.weight", hf.get("embed_out.weight"))
    elif a == "bloom":
        H, D = cfg.num_heads, cfg.head_dim
        out["embed_tokens.weight"] = hf["transformer.word_embeddings.weight"]
        out["embed_norm.weight"] = hf["transformer.word_embeddings_layernorm.weight"]
        out["embed_norm.bias"] = hf["transformer.word_embeddings_layernorm.bias"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "input_layernorm.weight"]
            out[o + "ln_1.bias"] = hf[p + "input_layernorm.bias"]
            out[o + "ln_2.weight"] = hf[p + "post_attention_layernorm.weight"]
            out[o + "ln_2.bias"] = hf[p + "post_attention_layernorm.bias"]
            # bloom fuses qkv per-head [H, 3, D, hidden] like NeoX
            w = hf[p + "self_attention.query_key_value.weight"].view(H, 3, D, -1)
            out[o + "attn.qkv_proj.weight"] = w.permute(1, 0, 2, 3).reshape(3 * H * D, -1).contiguous()
            b = hf[p + "self_attention.query_key_value.bias"].view(H, 3, D)
            out[o + "attn.qkv_proj.bias"] = b.permute(1, 0, 2).reshape(3 * H * D).contiguous()
            out[o + "attn.o_proj.weight"] = hf[p + "self_attention.dense.weight"]
            out[o + "attn.o_proj.bias"] = hf[p + "self_attention.dense.bias"]
            out[o + "mlp.fc_in.weight"] = hf[p + "mlp.dense_h_to_4h.weight"]
            out[o + "mlp.fc_in.bias"] = hf[p + "mlp.dense_h_to_4h.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.dense_4h_to_h.weight"]
            out[o + "mlp.down_proj.bias"] = hf[p + "mlp.dense_4h_to_h.bias"]
        out["final_norm.weight"] = hf["transformer.ln_f.weight"]
        out["final_norm.bias"] = hf["transformer.ln_f.bias"]
    elif a == "gpt_bigcode":
        out["embed_tokens.weight"] = hf["transformer.wte.weight"]
        out["embed_positions.weight"] = hf["transformer.wpe.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "ln_1.weight"]
            out[o + "ln_1.bias"] = hf[p + "ln_1.bias"]
            out[o + "ln_2.weight"] = hf[p + "ln_2.weight"]
            out[o + "ln_2.bias"] = hf[p + "ln_2.bias"]
            # gpt_bigcode uses plain Linear (NOT Conv1D): no transpose; MQA
            # layout [Hq*D + 2*D, hidden] matches the native fused qkv
            out[o + "attn.qkv_proj.weight"] = hf[p + "attn.c_attn.weight"]
            out[o + "attn.qkv_proj.bias"] = hf[p + "attn.c_attn.bias"]
            out[o + "attn.o_proj.weight"] = hf[p + "attn.c_proj.weight"]
            out[o + "attn.o_proj.bias"] = hf[p + "attn.c_proj.bias"]
            out[o + "mlp.fc_in.weight"] = hf[p + "mlp.c_fc.weight"]
            out[o + "mlp.fc_in.bias"] = hf[p + "mlp.c_fc.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "mlp.c_proj.weight"]
            out[o + "mlp.down_proj.bias"] = hf[p + "mlp.c_proj.bias"]
        out["final_norm.weight"] = hf["transformer.ln_f.weight"]
        out["final_norm.bias"] = hf["transformer.ln_f.bias"]
        if not cfg.tie_word_embeddings and "lm_head.weight" in hf:
            out["lm_head.weight"] = hf["lm_head.weight"]
    elif a == "opt":
        out["embed_tokens.weight"] = hf["model.decoder.embed_tokens.weight"]
        out["embed_positions.weight"] = hf["model.decoder.embed_positions.weight"]
        for i in range(L):
            p = f"model.decoder.layers.{i}."
            o = f"layers.{i}."
            out[o + "ln_1.weight"] = hf[p + "self_attn_layer_norm.weight"]
            out[o + "ln_1.bias"] = hf[p + "self_attn_layer_norm.bias"]
            out[o + "ln_2.weight"] = hf[p + "final_layer_norm.weight"]
            out[o + "ln_2.bias"] = hf[p + "final_layer_norm.bias"]
            out[o + "attn.qkv_proj.weight"] = _cat_qkv(
                hf[p + "self_attn.q_proj.weight"], hf[p + "self_attn.k_proj.weight"],
                hf[p + "self_attn.v_proj.weight"])
            out[o + "attn.qkv_proj.bias"] = _cat_qkv(
                hf[p + "self_attn.q_proj.bias"], hf[p + "self_attn.k_proj.bias"],
                hf[p + "self_attn.v_proj.bias"])
            out[o + "attn.o_proj.weight"] = hf[p + "self_attn.out_proj.weight"]
            out[o + "attn.o_proj.bias"] = hf[p + "self_attn.out_proj.bias"]
            out[o + "mlp.fc_in.weight"] = hf[p + "fc1.weight"]
            out[o + "mlp.fc_in.bias"] = hf[p + "fc1.bias"]
            out[o + "mlp.down_proj.weight"] = hf[p + "fc2.weight"]
            out[o + "mlp.down_proj.bias"] = hf[p + "fc2.bias"]
        out["final_norm.weight"] = hf["model.decoder.final_layer_norm.weight"]
        out["final_norm.bias"] = hf["model.decoder.final_layer_norm.bias"]
    else:
        raise ValueError(f"Unsupported arch: {a}")
    return out


def state_dict_to_hf(cfg: TransformerConfig, sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Native state dict -> HF layout (inverse of state_dict_from_hf)."""
    out = {}
    a = cfg.arch_name
    L = cfg.num_layers
    if a == "gpt2":
        out["transformer.wte.weight"] = sd["embed_tokens.weight"]
        out["transformer.wpe.weight"] = sd["embed_positions.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[p + "ln_1.weight"] = sd[o + "ln_1.weight"]
            out[p + "ln_1.bias"] = sd[o + "ln_1.bias"]
            out[p + "ln_2.weight"] = sd[o + "ln_2.weight"]
            out[p + "ln_2.bias"] = sd[o + "ln_2.bias"]
            out[p + "attn.c_attn.weight"] = sd[o + "attn.qkv_proj.weight"].t().contiguous()
            out[p + "attn.c_attn.bias"] = sd[o + "attn.qkv_proj.bias"]
            out[p + "attn.c_proj.weight"] = sd[o + "attn.o_proj.weight"].t().contiguous()
            out[p + "attn.c_proj.bias"] = sd[o + "attn.o_proj.bias"]
            out[p + "mlp.c_fc.weight"] = sd[o + "mlp.fc_in.weight"].t().contiguous()
            out[p + "mlp.c_fc.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "mlp.c_proj.weight"] = sd[o + "mlp.down_proj.weight"].t().contiguous()
            out[p + "mlp.c_proj.bias"] = sd[o + "mlp.down_proj.bias"]
        out["transformer.ln_f.weight"] = sd["final_norm.weight"]
        out["transformer.ln_f.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd.get("lm_head.weight", sd["embed_tokens.weight"])
    elif a == "llama":
        qd = cfg.num_heads * cfg.head_dim
        kd = cfg.num_kv_heads * cfg.head_dim
        out["model.embed_tokens.weight"] = sd["embed_tokens.weight"]
        for i in range(L):
            p = f"model.layers.{i}."
            o = f"layers.{i}."
            out[p + "input_layernorm.weight"] = sd[o + "ln_1.weight"]
            out[p + "post_attention_layernorm.weight"] = sd[o + "ln_2.weight"]
            w = sd[o + "attn.qkv_proj.weight"]
            out[p + "self_attn.q_proj.weight"] = w[:qd]
            out[p + "self_attn.k_proj.weight"] = w[qd : qd + kd]
            out[p + "self_attn.v_proj.weight"] = w[qd + kd :]
            out[p + "self_attn.o_proj.weight"] = sd[o + "attn.o_proj.weight"]
            gu = sd[o + "mlp.gate_up_proj.weight"]
            out[p + "mlp.gate_proj.weight"] = gu[: cfg.intermediate_size]
            out[p + "mlp.up_proj.weight"] = gu[cfg.intermediate_size :]
            out[p + "mlp.down_proj.weight"] = sd[o + "mlp.down_proj.weight"]
        out["model.norm.weight"] = sd["final_norm.weight"]
        out["lm_head.weight"] = sd.get("lm_head.weight", sd["embed_tokens.weight"])
    elif a == "gptj":
        qd = cfg.num_heads * cfg.head_dim
        out["transformer.wte.weight"] = sd["embed_tokens.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[p + "ln_1.weight"] = sd[o + "ln_1.weight"]
            out[p + "ln_1.bias"] = sd[o + "ln_1.bias"]
            w = sd[o + "attn.qkv_proj.weight"]
            out[p + "attn.q_proj.weight"] = w[:qd]
            out[p + "attn.k_proj.weight"] = w[qd : 2 * qd]
            out[p + "attn.v_proj.weight"] = w[2 * qd :]
            out[p + "attn.out_proj.weight"] = sd[o + "attn.o_proj.weight"]
            out[p + "mlp.fc_in.weight"] = sd[o + "mlp.fc_in.weight"]
            out[p + "mlp.fc_in.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "mlp.fc_out.weight"] = sd[o + "mlp.down_proj.weight"]
            out[p + "mlp.fc_out.bias"] = sd[o + "mlp.down_proj.bias"]
        out["transformer.ln_f.weight"] = sd["final_norm.weight"]
        out["transformer.ln_f.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd["lm_head.weight"]
        if "lm_head.bias" in sd:
            out["lm_head.bias"] = sd["lm_head.bias"]
    elif a == "gpt_neox":
        H, D = cfg.num_heads, cfg.head_dim
        out["gpt_neox.embed_in.weight"] = sd["embed_tokens.weight"]
        for i in range(L):
            p = f"gpt_neox.layers.{i}."
            o = f"layers.{i}."
            out[p + "input_layernorm.weight"] = sd[o + "ln_1.weight"]
            out[p + "input_layernorm.bias"] = sd[o + "ln_1.bias"]
            out[p + "post_attention_layernorm.weight"] = sd[o + "ln_2.weight"]
            out[p + "post_attention_layernorm.bias"] = sd[o + "ln_2.bias"]
            w = sd[o + "attn.qkv_proj.weight"].view(3, H, D, -1)
            out[p + "attention.query_key_value.weight"] = w.permute(1, 0, 2, 3).reshape(3 * H * D, -1).contiguous()
            b = sd[o + "attn.qkv_proj.bias"].view(3, H, D)
            out[p + "attention.query_key_value.bias"] = b.permute(1, 0, 2).reshape(3 * H * D).contiguous()
            out[p + "attention.dense.weight"] = sd[o + "attn.o_proj.weight"]
            out[p + "attention.dense.bias"] = sd[o + "attn.o_proj.bias"]
            out[p + "mlp.dense_h_to_4h.weight"] = sd[o + "mlp.fc_in.weight"]
            out[p + "mlp.dense_h_to_4h.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "mlp.dense_4h_to_h.weight"] = sd[o + "mlp.down_proj.weight"]
            out[p + "mlp.dense_4h_to_h.bias"] = sd[o + "mlp.down_proj.bias"]
        out["gpt_neox.final_layer_norm.weight"] = sd["final_norm.weight"]
        out["gpt_neox.final_layer_norm.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd["lm_head.weight"]
    elif a == "bloom":
        H, D = cfg.num_heads, cfg.head_dim
        out["transformer.word_embeddings.weight"] = sd["embed_tokens.weight"]
        out["transformer.word_embeddings_layernorm.weight"] = sd["embed_norm.weight"]
        out["transformer.word_embeddings_layernorm.bias"] = sd["embed_norm.bias"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[p + "input_layernorm.weight"] = sd[o + "ln_1.weight"]
            out[p + "input_layernorm.bias"] = sd[o + "ln_1.bias"]
            out[p + "post_attention_layernorm.weight"] = sd[o + "ln_2.weight"]
            out[p + "post_attention_layernorm.bias"] = sd[o + "ln_2.bias"]
            w = sd[o + "attn.qkv_proj.weight"].view(3, H, D, -1)
            out[p + "self_attention.query_key_value.weight"] = w.permute(1, 0, 2, 3).reshape(3 * H * D, -1).contiguous()
            b = sd[o + "attn.qkv_proj.bias"].view(3, H, D)
            out[p + "self_attention.query_key_value.bias"] = b.permute(1, 0, 2).reshape(3 * H * D).contiguous()
            out[p + "self_attention.dense.weight"] = sd[o + "attn.o_proj.weight"]
            out[p + "self_attention.dense.bias"] = sd[o + "attn.o_proj.bias"]
            out[p + "mlp.dense_h_to_4h.weight"] = sd[o + "mlp.fc_in.weight"]
            out[p + "mlp.dense_h_to_4h.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "mlp.dense_4h_to_h.weight"] = sd[o + "mlp.down_proj.weight"]
            out[p + "mlp.dense_4h_to_h.bias"] = sd[o + "mlp.down_proj.bias"]
        out["transformer.ln_f.weight"] = sd["final_norm.weight"]
        out["transformer.ln_f.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd.get("lm_head.weight", sd["embed_tokens.weight"])
    elif a == "gpt_bigcode":
        out["transformer.wte.weight"] = sd["embed_tokens.weight"]
        out["transformer.wpe.weight"] = sd["embed_positions.weight"]
        for i in range(L):
            p = f"transformer.h.{i}."
            o = f"layers.{i}."
            out[p + "ln_1.weight"] = sd[o + "ln_1.weight"]
            out[p + "ln_1.bias"] = sd[o + "ln_1.bias"]
            out[p + "ln_2.weight"] = sd[o + "ln_2.weight"]
            out[p + "ln_2.bias"] = sd[o + "ln_2.bias"]
            out[p + "attn.c_attn.weight"] = sd[o + "attn.qkv_proj.weight"]
            out[p + "attn.c_attn.bias"] = sd[o + "attn.qkv_proj.bias"]
            out[p + "attn.c_proj.weight"] = sd[o + "attn.o_proj.weight"]
            out[p + "attn.c_proj.bias"] = sd[o + "attn.o_proj.bias"]
            out[p + "mlp.c_fc.weight"] = sd[o + "mlp.fc_in.weight"]
            out[p + "mlp.c_fc.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "mlp.c_proj.weight"] = sd[o + "mlp.down_proj.weight"]
            out[p + "mlp.c_proj.bias"] = sd[o + "mlp.down_proj.bias"]
        out["transformer.ln_f.weight"] = sd["final_norm.weight"]
        out["transformer.ln_f.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd.get("lm_head.weight", sd["embed_tokens.weight"])
    elif a == "opt":
        qd = cfg.num_heads * cfg.head_dim
        out["model.decoder.embed_tokens.weight"] = sd["embed_tokens.weight"]
        out["model.decoder.embed_positions.weight"] = sd["embed_positions.weight"]
        for i in range(L):
            p = f"model.decoder.layers.{i}."
            o = f"layers.{i}."
            out[p + "self_attn_layer_norm.weight"] = sd[o + "ln_1.weight"]
            out[p + "self_attn_layer_norm.bias"] = sd[o + "ln_1.bias"]
            out[p + "final_layer_norm.weight"] = sd[o + "ln_2.weight"]
            out[p + "final_layer_norm.bias"] = sd[o + "ln_2.bias"]
            w = sd[o + "attn.qkv_proj.weight"]
            b = sd[o + "attn.qkv_proj.bias"]
            out[p + "self_attn.q_proj.weight"] = w[:qd]
            out[p + "self_attn.k_proj.weight"] = w[qd : 2 * qd]
            out[p + "self_attn.v_proj.weight"] = w[2 * qd :]
            out[p + "self_attn.q_proj.bias"] = b[:qd]
            out[p + "self_attn.k_proj.bias"] = b[qd : 2 * qd]
            out[p + "self_attn.v_proj.bias"] = b[2 * qd :]
            out[p + "self_attn.out_proj.weight"] = sd[o + "attn.o_proj.weight"]
            out[p + "self_attn.out_proj.bias"] = sd[o + "attn.o_proj.bias"]
            out[p + "fc1.weight"] = sd[o + "mlp.fc_in.weight"]
            out[p + "fc1.bias"] = sd[o + "mlp.fc_in.bias"]
            out[p + "fc2.weight"] = sd[o + "mlp.down_proj.weight"]
            out[p + "fc2.bias"] = sd[o + "mlp.down_proj.bias"]
        out["model.decoder.final_layer_norm.weight"] = sd["final_norm.weight"]
        out["model.decoder.final_layer_norm.bias"] = sd["final_norm.bias"]
        out["lm_head.weight"] = sd.get("lm_head.weight", sd["embed_tokens.weight"])
    else:
        raise ValueError(f"Unsupported arch: {a}")
    return out


# ---------------------------------------------------------------------------
# directory IO
# ---------------------------------------------------------------------------


def load_hf_dir(path: str) -> Tuple[TransformerConfig, Dict[str, torch.Tensor]]:
    """Read config.json + weights (safetensors or pytorch_model.bin) from a
    local HF-format directory."""
    with open(os.path.join(path, "config.json")) as f:
        hf_cfg = json.load(f)
    cfg = config_from_hf(hf_cfg)
    sd = {}
    st_index = os.path.join(path, "model.safetensors.index.json")
    pt_index = os.path.join(path, "pytorch_model.bin.index.json")
    st_single = os.path.join(path, "model.safetensors")
    pt_single = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(st_index):
        import safetensors.torch

        with open(st_index) as f:
            index = json.load(f)
        for shard in sorted(set(index["weight_map"].values())):
            sd.update(safetensors.torch.load_file(os.path.join(path, shard)))
    elif os.path.exists(pt_index):
        # sharded torch checkpoint (reference modeling_base.py:276-311 merges
        # the index's shards the same way)
        with open(pt_index) as f:
            index = json.load(f)
        for shard in sorted(set(index["weight_map"].values())):
            sd.update(torch.load(os.path.join(path, shard), map_location="cpu",
                                 weights_only=True))
    elif os.path.exists(st_single):
        import safetensors.torch

        sd = safetensors.torch.load_file(st_single)
    elif os.path.exists(pt_single):
        sd = torch.load(pt_single, map_location="cpu", weights_only=True)
    else:
        raise FileNotFoundError(f"No model weights found under {path}")
    return cfg, state_dict_from_hf(cfg, sd)


def save_hf_dir(path: str, cfg: TransformerConfig, sd: Dict[str, torch.Tensor],
                safe: bool = True) -> None:
    """Write an HF-format directory (config.json + model.safetensors) that
    vanilla transformers can load."""
    os.makedirs(path, exist_ok=True)
    hf_cfg = config_to_hf(cfg)
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(hf_cfg, f, indent=2)
    hf_sd = state_dict_to_hf(cfg, sd)
    hf_sd = {k: v.contiguous().cpu() for k, v in hf_sd.items()}
    if safe:
        import safetensors.torch

        # tied tensors share storage; safetensors requires unique storage
        seen = {}
        for k, v in list(hf_sd.items()):
            ptr = v.data_ptr()
            if ptr in seen and v.numel() > 0:
                hf_sd[k] = v.clone()
            seen[ptr] = k
        safetensors.torch.save_file(hf_sd, os.path.join(path, "model.safetensors"))
    else:
        torch.save(hf_sd, os.path.join(path, "pytorch_model.bin"))
