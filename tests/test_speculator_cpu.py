import pytest
import torch

from fms_fsdp_amd.config import train_config
from fms_fsdp_amd.models import Llama, LlamaConfig
from fms_fsdp_amd.models.speculator import MLPSpeculator


def tiny_base():
    cfg = LlamaConfig(src_vocab_size=64, emb_dim=32, nheads=2, kvheads=2,
                      nlayers=2, max_expected_seq_len=128)
    m = Llama(cfg)
    m.reset_parameters()
    m.eval()
    return m, cfg


def test_speculator_forward_shapes():
    torch.manual_seed(0)
    spec = MLPSpeculator(emb_dim=32, inner_dim=48, vocab_size=64, n_predict=3)
    spec.reset_parameters()
    state = torch.randn(2, 10, 32)
    inds = torch.randint(0, 64, (2, 12))
    out = spec(state, inds)
    assert out.shape == (3, 2, 10, 64)


def test_speculator_tied_weights():
    spec = MLPSpeculator(emb_dim=32, inner_dim=48, vocab_size=64, n_predict=3,
                         tie_weights=True)
    n_tied = spec.param_count()
    spec2 = MLPSpeculator(emb_dim=32, inner_dim=48, vocab_size=64, n_predict=3,
                          tie_weights=False)
    assert spec2.param_count() > n_tied


def test_generate_with_cache_matches_full_forward():
    """Greedy cached generation must equal argmax of the full forward."""
    torch.manual_seed(1)
    m, cfg = tiny_base()
    x = torch.randint(0, 64, (2, 8))
    tokens = m.generate(x, 4, do_sample=False)
    assert tokens.shape == (2, 12)
    # step-by-step check: first generated token == argmax of full logits
    logits = m(x)
    assert torch.equal(tokens[:, 8], logits[:, -1].argmax(-1))
    # second token: full forward on extended sequence
    logits2 = m(tokens[:, :9])
    assert torch.equal(tokens[:, 9], logits2[:, -1].argmax(-1))


def test_include_embeds():
    m, cfg = tiny_base()
    x = torch.randint(0, 64, (1, 8))
    logits, emb = m(x, include_embeds=True)
    assert emb.shape == (1, 8, 32)
    toks, gemb = m.generate(x, 3, include_embeds=True)
    assert gemb.shape == (1, 3, 32)


def test_stage1_stage2_losses():
    from speculator.train_speculator_utils import stage1_loss, stage2_loss
    torch.manual_seed(2)
    m, mcfg = tiny_base()
    spec = MLPSpeculator(emb_dim=32, inner_dim=32, vocab_size=64, n_predict=2)
    spec.reset_parameters()
    cfg = train_config()
    cfg.batch_size = 2
    cfg.seq_length = 64
    cfg.stage2_batch_size = 8
    cfg.stage2_prompt_length = 8
    cfg.stage2_seq_length = 24
    inp = torch.randint(0, 64, (2, 64))
    stats = torch.zeros(2 + spec.n_predict)
    loss, stats, ntok = stage1_loss(cfg, m, spec, inp, inp, stats)
    assert torch.isfinite(loss)
    loss.backward()
    assert any(p.grad is not None for p in spec.parameters())

    for p in spec.parameters():
        p.grad = None
    loss2, stats, ntok2 = stage2_loss(cfg, m, spec, inp, inp, stats)
    assert torch.isfinite(loss2)
    loss2.backward()
    assert any(p.grad is not None for p in spec.parameters())


def test_do_ckpt_file_poll(tmp_path):
    from speculator.train_speculator_utils import do_ckpt
    d = str(tmp_path)
    assert not do_ckpt(d)
    with open(f"{d}/do_ckpt", "w") as f:
        f.write("1")
    assert do_ckpt(d)
    do_ckpt(d, reset=True)
    assert not do_ckpt(d)


def test_train_speculator_entry_smoke(tmp_path):
    """Full speculator entry point: tiny base model, dummy data, 2 stage-1
    steps single-process."""
    from speculator import train_speculator as ts
    ts.main(model_variant="llama2_125m", use_dummy_dataset=True,
            batch_size=1, seq_length=128, num_steps=2, report_interval=1,
            checkpoint_interval=100, mixed_precision=False,
            n_speculator_heads=2, speculator_width=64,
            stage2_start_step=10, model_path="/nonexistent",
            ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
            vocab_size=256, learning_rate=1e-4, sharding_strategy="fsdp")


def test_gpt_bigcode_base():
    from fms_fsdp_amd.config import get_model_config
    from fms_fsdp_amd.models.gpt_bigcode import GPTBigCode
    torch.manual_seed(3)
    m = GPTBigCode(get_model_config("gpt_bigcode_test"))
    m.reset_parameters()
    m.eval()
    x = torch.randint(0, 256, (2, 16))
    logits, emb = m(x, include_embeds=True)
    assert logits.shape == (2, 16, 256) and emb.shape == (2, 16, 128)
    toks = m.generate(x, 3, do_sample=False)
    assert toks.shape == (2, 19)
    # greedy cached generation == argmax of full forward
    assert torch.equal(toks[:, 16], m(x)[:, -1].argmax(-1))


def test_mixtral_base():
    from fms_fsdp_amd.config import get_model_config
    from fms_fsdp_amd.models.mixtral import Mixtral
    torch.manual_seed(4)
    m = Mixtral(get_model_config("mixtral_test"))
    m.reset_parameters()
    m.eval()
    x = torch.randint(0, 256, (2, 16))
    logits, emb = m(x, include_embeds=True)
    assert logits.shape == (2, 16, 256) and emb.shape == (2, 16, 64)
    toks = m.generate(x, 3, do_sample=False)
    assert torch.equal(toks[:, 16], m(x)[:, -1].argmax(-1))
    # loss path + backward
    loss = m(x, labels=torch.randint(0, 256, (2, 16)))
    loss.backward()
    assert torch.isfinite(loss)


def test_speculator_entry_bigcode(tmp_path):
    from speculator import train_speculator as ts
    ts.main(model_variant="gpt_bigcode_test", model_arch="embedgptbigcode",
            use_dummy_dataset=True, batch_size=1, seq_length=128,
            num_steps=1, report_interval=1, checkpoint_interval=100,
            mixed_precision=False, n_speculator_heads=2, speculator_width=64,
            stage2_start_step=10, model_path="/nonexistent",
            ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
            vocab_size=256, learning_rate=1e-4, sharding_strategy="fsdp")
