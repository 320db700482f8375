"""Convert a sharded Mamba training checkpoint to a standalone
pretrained-format folder (parity: reference fms_to_hf_mamba.py, which
does a DCP no-dist load -> MambaLMHeadModel.save_pretrained).

Since `mamba_ssm` is not a dependency of this framework, the exporter
writes the consolidated fp32 state dict + config json in the
save_pretrained layout (pytorch_model.bin + config.json) that
mamba_ssm's MambaLMHeadModel.from_pretrained consumes.

Usage: python fms_to_hf_mamba.py --model_variant mamba_9.8b
  --load_path /path/step_N_ckp --save_path /out
"""

import argparse
import json
import os

import torch

from fms_fsdp_amd.config import get_model_config
from fms_fsdp_amd.utils.checkpointing import consolidate_checkpoint


def convert_to_mamba_ssm_state_dict(sd, cfg_dict):
    """our name->tensor dict -> mamba_ssm MambaLMHeadModel naming."""
    out = {
        "backbone.embedding.weight": sd["embedding.weight"],
        "backbone.norm_f.weight": sd["norm_f.weight"],
        "lm_head.weight": sd["lm_head.weight"],
    }
    n_layer = cfg_dict["n_layer"]
    attn_idx = set(cfg_dict.get("attn_layer_idx", []))
    for i in range(n_layer):
        pre = f"layers.{i}."
        hf = f"backbone.layers.{i}."
        out[hf + "norm.weight"] = sd[pre + "norm.weight"]
        if i in attn_idx:
            out[hf + "mixer.Wqkv.weight"] = sd[pre + "mixer.qkv.weight"]
            out[hf + "mixer.out_proj.weight"] = sd[pre + "mixer.proj.weight"]
        else:
            out[hf + "mixer.in_proj.weight"] = sd[pre + "mixer.in_proj.weight"]
            out[hf + "mixer.conv1d.weight"] = \
                sd[pre + "mixer.conv_weight"].unsqueeze(1)
            out[hf + "mixer.conv1d.bias"] = sd[pre + "mixer.conv_bias"]
            out[hf + "mixer.dt_bias"] = sd[pre + "mixer.dt_bias"]
            out[hf + "mixer.A_log"] = sd[pre + "mixer.A_log"]
            out[hf + "mixer.D"] = sd[pre + "mixer.D"]
            out[hf + "mixer.norm.weight"] = sd[pre + "mixer.norm.weight"]
            out[hf + "mixer.out_proj.weight"] = sd[pre + "mixer.out_proj.weight"]
        if pre + "mlp.wg1.weight" in sd:
            out[hf + "mlp.fc1.weight"] = sd[pre + "mlp.wg1.weight"]
            out[hf + "mlp.fc2.weight"] = sd[pre + "mlp.w2.weight"]
        if pre + "norm2.weight" in sd:
            out[hf + "norm2.weight"] = sd[pre + "norm2.weight"]
    return out


def main(model_variant, load_path, save_path):
    cfg_dict = get_model_config(model_variant)
    assert isinstance(cfg_dict, dict), "expects a mamba variant"
    print(f"Consolidating {load_path} ...")
    sd = consolidate_checkpoint(load_path, dtype=torch.float32)
    out = convert_to_mamba_ssm_state_dict(sd, cfg_dict)
    os.makedirs(save_path, exist_ok=True)
    torch.save(out, os.path.join(save_path, "pytorch_model.bin"))
    with open(os.path.join(save_path, "config.json"), "w") as f:
        json.dump(cfg_dict, f, indent=2)
    print(f"Saved to {save_path}")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--model_variant", default="mamba_9.8b")
    ap.add_argument("--load_path", required=True)
    ap.add_argument("--save_path", required=True)
    a = ap.parse_args()
    main(a.model_variant, a.load_path, a.save_path)
