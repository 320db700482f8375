"""HF export: converted model must produce identical logits to ours."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_hf_llama_logit_parity(tmp_path):
    from fms_fsdp_amd.models import Llama, LlamaBlock, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel, ShardedAdamW
    from fms_fsdp_amd.utils.checkpointing import Checkpointer, consolidate_checkpoint
    from fms_to_hf_llama import convert_to_hf_state_dict, hf_config_for
    from transformers import LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=96, emb_dim=64, nheads=4, kvheads=2,
                      nlayers=2, max_expected_seq_len=64)
    m = Llama(cfg)
    m.reset_parameters()
    sm = ShardedModel(m, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    opt = ShardedAdamW(sm, lr=1e-3)
    ck = Checkpointer(str(tmp_path), 2, "fsdp", 0, 0)
    out = ck.save(1, sm, opt, None)

    sd = consolidate_checkpoint(out)
    hf_sd = convert_to_hf_state_dict(sd, cfg)
    hf = LlamaForCausalLM(hf_config_for(cfg))
    missing, unexpected = hf.load_state_dict(hf_sd, strict=False)
    assert not unexpected
    hf.eval()

    x = torch.randint(0, 96, (2, 16))
    with torch.no_grad():
        ours = sm(x)
        theirs = hf(x).logits
    err = (ours.float() - theirs.float()).abs().max().item()
    assert err < 1e-3, err


def test_dcp_import_roundtrip(tmp_path):
    """A reference-style torch-DCP checkpoint (fms parameter names,
    unfused q/k/v and wg/w1) imports into a loadable single-file state
    dict: tools/dcp_to_native.py (VERDICT missing #5 — DCP interop)."""
    import torch
    import torch.distributed.checkpoint as dcp
    from fms_fsdp_amd.models import Llama, LlamaConfig
    from fms_fsdp_amd.parallel import ShardedModel
    from fms_fsdp_amd.models.llama import LlamaBlock
    from fms_fsdp_amd.utils.checkpointing import Checkpointer
    import tools.dcp_to_native as conv

    torch.manual_seed(0)
    cfg = LlamaConfig(src_vocab_size=64, emb_dim=32, nheads=4, kvheads=4,
                      nlayers=2, max_expected_seq_len=64)
    ref = Llama(cfg)
    ref.reset_parameters()

    def half_to_interleave(w, n_heads):
        # inverse of the importer's row re-order: our half-rotation
        # layout back to fms's interleaved pairs
        rows, cols = w.shape
        return w.view(n_heads, 2, rows // n_heads // 2, cols) \
            .transpose(1, 2).reshape(rows, cols)

    # synthesize the reference's DCP layout: fms names, unfused q/k/v,
    # INTERLEAVED RoPE rows (what a real fms checkpoint holds)
    hd = cfg.nheads * cfg.head_dim
    fms_sd = {"base_model.embedding.weight": ref.embedding.weight.detach(),
              "head.weight": ref.lm_head.weight.detach(),
              "base_model.dec_norm.weight": ref.norm.weight.detach()}
    for i, layer in enumerate(ref.layers):
        p = f"base_model.layers.{i}."
        qkv = layer.attn.qkv.weight.detach()
        fms_sd[p + "ln.weight"] = layer.attn_norm.weight.detach()
        fms_sd[p + "ff_ln.weight"] = layer.mlp_norm.weight.detach()
        fms_sd[p + "attn.in_proj.query.weight"] = \
            half_to_interleave(qkv[:hd], cfg.nheads)
        fms_sd[p + "attn.in_proj.key.weight"] = \
            half_to_interleave(qkv[hd:2 * hd], cfg.kvheads)
        fms_sd[p + "attn.in_proj.value.weight"] = qkv[2 * hd:]
        fms_sd[p + "attn.dense.weight"] = layer.attn.proj.weight.detach()
        wg1 = layer.mlp.wg1.weight.detach()
        fms_sd[p + "ff_sub_layer.wg.weight"] = wg1[:cfg.hidden_dim]
        fms_sd[p + "ff_sub_layer.w1.weight"] = wg1[cfg.hidden_dim:]
        fms_sd[p + "ff_sub_layer.w2.weight"] = layer.mlp.w2.weight.detach()
    dcp_dir = str(tmp_path / "step_100_ckp")
    dcp.save({"model_state": fms_sd}, checkpoint_id=dcp_dir, no_dist=True)

    out = str(tmp_path / "imported.pth")
    mapped = conv.convert(dcp_dir, out, nheads=cfg.nheads,
                          kvheads=cfg.kvheads)
    assert set(mapped) == {n for n, _ in ref.named_parameters()}

    # load through the Checkpointer's single-file path into a new model
    torch.manual_seed(99)
    m2 = Llama(cfg)
    m2.reset_parameters()
    sm2 = ShardedModel(m2, LlamaBlock, sharding_strategy="fsdp",
                      param_dtype=torch.float32)
    ck = Checkpointer(str(tmp_path / "save"), 2, "fsdp", 0, 0)
    ck.load(sm2, None, None, path=out)
    for n, p_ref in ref.named_parameters():
        got = dict(m2.named_parameters())[n]
        assert torch.equal(got.detach(), p_ref.detach()), n
