"""Convert a sharded training checkpoint to a HuggingFace Llama model.

Parity target: reference fms_to_hf_llama.py (DCP no-dist load -> HF
weights). Differences by design: our model already uses the HF RoPE
half-rotation convention, so no q/k row interleave permutation is needed
(the reference needs one because ibm-fms uses the interleaved layout,
fms_to_hf_llama.py:104-124 there); fused qkv and gate|up projections are
split here (reference analog :69-76, :89-95).

Usage: python fms_to_hf_llama.py --model_variant 7b --load_path
  /path/to/step_N_ckp --save_path /out --tokenizer_name_or_path /tok
"""

import argparse

import torch

from fms_fsdp_amd.config import get_model_config
from fms_fsdp_amd.utils.checkpointing import consolidate_checkpoint


def convert_to_hf_state_dict(sd, cfg):
    """our name->tensor dict -> HF LlamaForCausalLM state dict."""
    hd = cfg.head_dim
    nq = cfg.nheads * hd
    nkv = cfg.kvheads * hd
    out = {
        "model.embed_tokens.weight": sd["embedding.weight"],
        "model.norm.weight": sd["norm.weight"],
        "lm_head.weight": sd["lm_head.weight"],
    }
    for i in range(cfg.nlayers):
        pre = f"layers.{i}."
        hf = f"model.layers.{i}."
        qkv = sd[pre + "attn.qkv.weight"]
        out[hf + "self_attn.q_proj.weight"] = qkv[:nq]
        out[hf + "self_attn.k_proj.weight"] = qkv[nq:nq + nkv]
        out[hf + "self_attn.v_proj.weight"] = qkv[nq + nkv:]
        out[hf + "self_attn.o_proj.weight"] = sd[pre + "attn.proj.weight"]
        gu = sd[pre + "mlp.wg1.weight"]
        out[hf + "mlp.gate_proj.weight"] = gu[:cfg.hidden_dim]
        out[hf + "mlp.up_proj.weight"] = gu[cfg.hidden_dim:]
        out[hf + "mlp.down_proj.weight"] = sd[pre + "mlp.w2.weight"]
        out[hf + "input_layernorm.weight"] = sd[pre + "attn_norm.weight"]
        out[hf + "post_attention_layernorm.weight"] = sd[pre + "mlp_norm.weight"]
    return out


def hf_config_for(cfg):
    from transformers import LlamaConfig as HFLlamaConfig
    return HFLlamaConfig(
        vocab_size=cfg.src_vocab_size,
        hidden_size=cfg.emb_dim,
        intermediate_size=cfg.hidden_dim,
        num_hidden_layers=cfg.nlayers,
        num_attention_heads=cfg.nheads,
        num_key_value_heads=cfg.kvheads,
        max_position_embeddings=cfg.max_expected_seq_len,
        rms_norm_eps=cfg.norm_eps,
        rope_theta=cfg.rope_theta,
        tie_word_embeddings=False,
    )


def main(model_variant, load_path, save_path, tokenizer_name_or_path=None):
    from transformers import LlamaForCausalLM
    cfg = get_model_config(
        model_variant if model_variant.startswith(("llama", "mamba"))
        else f"llama2_{model_variant}")
    print(f"Consolidating checkpoint {load_path} ...")
    sd = consolidate_checkpoint(load_path, dtype=torch.float32)
    hf_sd = convert_to_hf_state_dict(sd, cfg)
    print("Building HF model ...")
    model = LlamaForCausalLM(hf_config_for(cfg))
    missing, unexpected = model.load_state_dict(hf_sd, strict=False)
    assert not unexpected, f"unexpected keys: {unexpected[:5]}"
    real_missing = [m for m in missing if "rotary" not in m]
    assert not real_missing, f"missing keys: {real_missing[:5]}"
    print(f"Saving HF model to {save_path} ...")
    model.save_pretrained(save_path)
    if tokenizer_name_or_path:
        from transformers import AutoTokenizer
        AutoTokenizer.from_pretrained(tokenizer_name_or_path).save_pretrained(save_path)
    print("done")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--model_variant", required=True)
    ap.add_argument("--load_path", required=True)
    ap.add_argument("--save_path", required=True)
    ap.add_argument("--tokenizer_name_or_path", default=None)
    a = ap.parse_args()
    main(a.model_variant, a.load_path, a.save_path, a.tokenizer_name_or_path)
