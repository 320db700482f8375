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
