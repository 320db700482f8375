"""Config merging and the model-variant registry.

Parity target: reference fms_fsdp/utils/config_utils.py:6-189 (update_config
semantics incl. dotted addressing and unknown-key warnings; the same llama2/
llama3/mamba variant dimensions). The model config class here is our own
(`fms_fsdp_amd.models.llama.LlamaConfig`) — the reference delegates to the
external `ibm-fms` package, which this framework replaces.
"""

from dataclasses import fields


def update_config(config, **kwargs):
    """Merge CLI/keyword overrides into one dataclass or a tuple of them.

    Dotted keys address a specific dataclass by type name
    (e.g. ``train_config.seq_length``). Unknown keys warn, never raise.
    (reference: config_utils.py:6-22)
    """
    if isinstance(config, (tuple, list)):
        for c in config:
            update_config(c, **kwargs)
        return
    field_names = {f.name for f in fields(config)}
    for k, v in kwargs.items():
        if "." in k:
            cls_name, key = k.split(".", 1)
            if type(config).__name__ == cls_name:
                if key in field_names:
                    setattr(config, key, v)
                else:
                    print(f"Warning: config {cls_name} does not accept parameter: {key}")
        elif k in field_names:
            setattr(config, k, v)
        elif not k.startswith("_"):
            print(f"Warning: unknown parameter {k}")


def get_model_config(model_variant):
    """String variant -> model config (reference: config_utils.py:25-189).

    Llama variants return a LlamaConfig; mamba variants return a plain dict
    consumed by fms_fsdp_amd.models.mamba.
    """
    from fms_fsdp_amd.models.llama import LlamaConfig

    llama_variants = {
        "llama2_70b": dict(emb_dim=8192, multiple_of=4096, nheads=64, kvheads=8,
                           nlayers=80, hidden_grow_factor=28672 / 8192),
        "llama2_34b": dict(emb_dim=8192, nheads=64, kvheads=8, nlayers=48,
                           hidden_grow_factor=22016 / 8192,
                           max_expected_seq_len=16384, rope_theta=1000000.0),
        "llama2_13b": dict(emb_dim=5120, nheads=40, nlayers=40,
                           hidden_grow_factor=13824 / 5120),
        "llama2_7b": dict(hidden_grow_factor=11008 / 4096, kvheads=32),
        # CPU-trivial entry for contract/smoke tests
        "llama2_test": dict(src_vocab_size=256, emb_dim=64, nheads=2,
                            kvheads=2, nlayers=2, max_expected_seq_len=128),
        "llama2_1.4b": dict(emb_dim=2048, nheads=16, nlayers=24,
                            hidden_grow_factor=3, kvheads=4),
        "llama3_8b": dict(src_vocab_size=128256, emb_dim=4096, nheads=32, kvheads=8,
                          nlayers=32, hidden_grow_factor=3.5,
                          max_expected_seq_len=8192, rope_theta=500000.0),
        "llama3_1.8b": dict(src_vocab_size=128256, emb_dim=2048, nheads=16, kvheads=8,
                            nlayers=24, hidden_grow_factor=3.5,
                            max_expected_seq_len=8192, rope_theta=500000.0),
        "llama3_3.2b": dict(src_vocab_size=128256, emb_dim=3072, nheads=24, kvheads=8,
                            nlayers=24, hidden_grow_factor=8 / 3,
                            max_expected_seq_len=8192, rope_theta=500000.0),
        "llama3_70b": dict(src_vocab_size=128256, emb_dim=8192, nheads=64, kvheads=8,
                           nlayers=80, hidden_grow_factor=3.5,
                           max_expected_seq_len=8192, rope_theta=500000.0),
        "llama3_194m_4k": dict(src_vocab_size=128256, emb_dim=1024, nheads=8,
                               nlayers=10, max_expected_seq_len=4096,
                               rope_theta=500000.0),
        # tiny smoke/benchmark config (this repo; used by BASELINE.json config 1)
        "llama2_125m": dict(emb_dim=768, nheads=12, kvheads=12, nlayers=12,
                            hidden_grow_factor=8 / 3),
    }
    # llama3 *_4k variants share dims with the 8k ones, seq len 4096
    for base in ["llama3_8b", "llama3_1.8b", "llama3_3.2b", "llama3_70b"]:
        llama_variants[base + "_4k"] = dict(llama_variants[base],
                                            max_expected_seq_len=4096)

    if model_variant in llama_variants:
        return LlamaConfig(**llama_variants[model_variant])

    if model_variant == "mamba_9.8b":
        return {
            "d_model": 4096,
            "d_intermediate": 14336,
            "n_layer": 32,
            "vocab_size": 128256,
            "ssm_cfg": {"layer": "Mamba2"},
            "attn_layer_idx": [9, 18, 27],
            "attn_cfg": {
                "causal": True,
                "d_conv": 0,
                "head_dim": 128,
                "num_heads": 32,
                "num_heads_kv": 8,
                "out_proj_bias": False,
                "qkv_proj_bias": False,
                "rotary_emb_dim": 64,
            },
            "rms_norm": True,
            "residual_in_fp32": True,
            "fused_add_norm": True,
            "pad_vocab_size_multiple": 16,
            "tie_embeddings": False,
        }
    if model_variant in ("gpt_bigcode_starcoder", "gpt_bigcode_test"):
        from fms_fsdp_amd.models.gpt_bigcode import GPTBigCodeConfig
        if model_variant == "gpt_bigcode_test":
            return GPTBigCodeConfig(src_vocab_size=256, emb_dim=128, nheads=1,
                                    nlayers=2, max_expected_seq_len=256)
        return GPTBigCodeConfig()
    if model_variant in ("mixtral_8x7b", "mixtral_test"):
        from fms_fsdp_amd.models.mixtral import MixtralConfig
        if model_variant == "mixtral_test":
            return MixtralConfig(src_vocab_size=256, emb_dim=64, nheads=1,
                                 kvheads=1, nlayers=2, hidden_dim=128,
                                 num_experts=4, top_k=2,
                                 max_expected_seq_len=256)
        return MixtralConfig()
    if model_variant == "mamba_test":
        # tiny CPU-trivial hybrid used by the test suite / smoke runs
        return {
            "d_model": 64,
            "d_intermediate": 128,
            "n_layer": 2,
            "vocab_size": 512,
            "ssm_cfg": {"layer": "Mamba2"},
            "attn_layer_idx": [1],
            "attn_cfg": {"causal": True, "d_conv": 0, "head_dim": 64,
                         "num_heads": 1, "num_heads_kv": 1,
                         "rotary_emb_dim": 16},
            "rms_norm": True,
            "residual_in_fp32": True,
            "fused_add_norm": True,
            "pad_vocab_size_multiple": 16,
            "tie_embeddings": False,
            "d_state": 16,
            "headdim": 16,
            "chunk_size": 64,
        }
    if model_variant == "mamba_2.8b":
        return {
            "d_model": 2560,
            "d_intermediate": 0,
            "n_layer": 64,
            "vocab_size": 50277,
            "ssm_cfg": {"layer": "Mamba2"},
            "attn_layer_idx": [],
            "attn_cfg": {},
            "rms_norm": True,
            "residual_in_fp32": True,
            "fused_add_norm": True,
            "pad_vocab_size_multiple": 16,
            "tie_embeddings": False,
        }

    raise ValueError(f"model variant {model_variant} not supported.")
