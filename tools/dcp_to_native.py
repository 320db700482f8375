"""One-shot import of a reference (fms-fsdp) torch-DCP checkpoint into a
single-file state dict this framework can load for continued pretraining.

The reference saves FSDP-sharded DCP folders whose "model_state" holds an
ibm-fms LLaMA state dict (reference: fms_fsdp/utils/checkpointing_utils.py
:152-161, key layout visible in fms_to_hf_llama.py:56-130). This tool:
  1. reads the DCP folder offline (torch.distributed.checkpoint
     format_utils — no process group needed),
  2. maps the fms parameter names onto this framework's Llama names,
     fusing separate q/k/v (and wg/w1) projections where the source model
     was not fused — our attention/MLP always use fused projections,
  3. writes a plain torch.save file that `Checkpointer.load(path=<file>)`
     / `main_training_llama --ckpt_load_path=<file>` accepts.

The q/k projection rows are re-ordered from ibm-fms's INTERLEAVED RoPE
pair layout to this framework's half-rotation (HF-native) layout — the
same `view(nh, d/2, 2, in).transpose(1, 2)` transform the reference's own
HF exporter applies (fms_to_hf_llama.py:104-124 there). Pass
--model_variant so the head geometry is known.

Optimizer state is intentionally NOT imported: the reference's DCP
optimizer payload is torch-FSDP flat-parameter sharded and meaningless to
resume into a different sharding runtime; continued pretraining restarts
the optimizer (same behavior the reference applies on a "continued
training" load, checkpointing_utils.py:215-233).

Usage:
  python tools/dcp_to_native.py --dcp_path /ref/checkpoints/step_N_ckp \
      --out /tmp/imported.pth
"""

import argparse
import os
import re
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def _interleave_to_half(w, n_heads):
    """fms interleaved RoPE rows -> half-rotation rows (per head)."""
    rows, cols = w.shape
    return w.view(n_heads, rows // n_heads // 2, 2, cols) \
        .transpose(1, 2).reshape(rows, cols)


def map_fms_to_native(sd, nheads=None, kvheads=None):
    """fms LLaMA state dict -> our Llama names (+ RoPE row re-order when
    the head geometry is given). Returns (mapped, skipped)."""
    out = {}
    skipped = []
    qkv_parts = {}   # layer -> {q,k,v}
    gate_parts = {}  # layer -> {wg,w1}

    def layer_of(key):
        m = re.search(r"layers\.(\d+)\.", key)
        return int(m.group(1)) if m else None

    for k, v in sd.items():
        k0 = k.replace("_orig_mod.", "")
        base = k0.replace("base_model.", "")
        li = layer_of(base)
        if base == "embedding.weight":
            out["embedding.weight"] = v
        elif base in ("head.weight", "lm_head.weight"):
            out["lm_head.weight"] = v
        elif base in ("dec_norm.weight", "norm.weight"):
            out["norm.weight"] = v
        elif li is not None and base.endswith(".ln.weight"):
            out[f"layers.{li}.attn_norm.weight"] = v
        elif li is not None and base.endswith(".ff_ln.weight"):
            out[f"layers.{li}.mlp_norm.weight"] = v
        elif li is not None and "attn.in_proj.qkv_fused.weight" in base:
            out[f"layers.{li}.attn.qkv.weight"] = v
        elif li is not None and "attn.in_proj.query.weight" in base:
            qkv_parts.setdefault(li, {})["q"] = v
        elif li is not None and "attn.in_proj.key.weight" in base:
            qkv_parts.setdefault(li, {})["k"] = v
        elif li is not None and "attn.in_proj.value.weight" in base:
            qkv_parts.setdefault(li, {})["v"] = v
        elif li is not None and "attn.dense.weight" in base:
            out[f"layers.{li}.attn.proj.weight"] = v
        elif li is not None and "ff_sub_layer.wg1_fused.weight" in base:
            out[f"layers.{li}.mlp.wg1.weight"] = v
        elif li is not None and "ff_sub_layer.wg.weight" in base:
            gate_parts.setdefault(li, {})["wg"] = v
        elif li is not None and "ff_sub_layer.w1.weight" in base:
            gate_parts.setdefault(li, {})["w1"] = v
        elif li is not None and "ff_sub_layer.w2.weight" in base:
            out[f"layers.{li}.mlp.w2.weight"] = v
        else:
            skipped.append(k)

    for li, parts in qkv_parts.items():
        assert set(parts) == {"q", "k", "v"}, \
            f"layer {li}: incomplete q/k/v triple {list(parts)}"
        out[f"layers.{li}.attn.qkv.weight"] = torch.cat(
            [parts["q"], parts["k"], parts["v"]], dim=0)

    if nheads is not None:
        kvh = kvheads or nheads
        for k in list(out):
            if k.endswith("attn.qkv.weight"):
                w = out[k]
                qr = w.shape[0] * nheads // (nheads + 2 * kvh)
                kr = (w.shape[0] - qr) // 2
                out[k] = torch.cat(
                    [_interleave_to_half(w[:qr], nheads),
                     _interleave_to_half(w[qr:qr + kr], kvh),
                     w[qr + kr:]], dim=0)
    for li, parts in gate_parts.items():
        assert set(parts) == {"wg", "w1"}, \
            f"layer {li}: incomplete wg/w1 pair {list(parts)}"
        out[f"layers.{li}.mlp.wg1.weight"] = torch.cat(
            [parts["wg"], parts["w1"]], dim=0)
    return out, skipped


def convert(dcp_path, out_path, nheads=None, kvheads=None):
    from torch.distributed.checkpoint.format_utils import dcp_to_torch_save
    with tempfile.TemporaryDirectory() as td:
        tmp = os.path.join(td, "flat.pth")
        dcp_to_torch_save(dcp_path, tmp)
        raw = torch.load(tmp, map_location="cpu", weights_only=False)
    sd = raw.get("model_state", raw)
    if "_orig_mod" in sd:   # compiled-model nesting (fms_to_hf_llama.py:155)
        sd = sd["_orig_mod"]
    mapped, skipped = map_fms_to_native(sd, nheads, kvheads)
    if nheads is None:
        print("WARNING: no --model_variant given — q/k rows were NOT "
              "re-ordered from the fms interleaved RoPE layout")
    if skipped:
        print(f"WARNING: {len(skipped)} unmapped keys (first 10): "
              f"{skipped[:10]}")
    torch.save(mapped, out_path)
    print(f"wrote {out_path}: {len(mapped)} tensors, "
          f"{sum(t.numel() for t in mapped.values()) / 1e6:.1f}M params")
    return mapped


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--dcp_path", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--model_variant", default=None,
                    help="registry name (e.g. llama2_7b) for head geometry")
    a = ap.parse_args()
    nh = kvh = None
    if a.model_variant:
        from fms_fsdp_amd.config import get_model_config
        mc = get_model_config(a.model_variant)
        nh, kvh = mc.nheads, mc.kvheads
    convert(a.dcp_path, a.out, nh, kvh)
