from fms_fsdp_amd.config import train_config, update_config, get_model_config


def test_update_config_basic():
    cfg = train_config()
    update_config(cfg, seq_length=2048, batch_size=4)
    assert cfg.seq_length == 2048 and cfg.batch_size == 4


def test_update_config_dotted_and_unknown(capsys):
    cfg = train_config()
    update_config(cfg, **{"train_config.learning_rate": 1e-4, "bogus_key": 1})
    assert cfg.learning_rate == 1e-4
    assert "unknown parameter bogus_key" in capsys.readouterr().out


def test_update_config_tuple():
    c1, c2 = train_config(), train_config()
    update_config((c1, c2), num_steps=5)
    assert c1.num_steps == 5 and c2.num_steps == 5


def test_model_registry_dims():
    c = get_model_config("llama2_7b")
    assert c.emb_dim == 4096 and c.nlayers == 32 and c.nheads == 32
    assert c.hidden_dim == 11008 and c.kvheads == 32
    c = get_model_config("llama2_70b")
    assert c.emb_dim == 8192 and c.nlayers == 80 and c.kvheads == 8
    assert c.hidden_dim == 28672
    c = get_model_config("llama2_13b")
    assert c.hidden_dim == 13824 and c.nlayers == 40
    c = get_model_config("llama3_8b")
    assert c.src_vocab_size == 128256 and c.rope_theta == 500000.0
    m = get_model_config("mamba_9.8b")
    assert m["d_model"] == 4096 and m["attn_layer_idx"] == [9, 18, 27]


def test_param_counts():
    import torch
    from fms_fsdp_amd.models import Llama
    with torch.device("meta"):
        m = Llama(get_model_config("llama2_7b"))
    n = m.param_count()
    assert 6.5e9 < n < 7.0e9, n
    with torch.device("meta"):
        m = Llama(get_model_config("llama2_70b"))
    n = m.param_count()
    assert 6.6e10 < n < 7.2e10, n


def test_all_registry_variants_construct_on_meta():
    """Every variant the reference registry defines (config_utils.py:25-189
    there) builds on the meta device with a sane parameter count —
    catches config typos without materializing 70B anywhere."""
    import torch
    from fms_fsdp_amd.config import get_model_config
    from fms_fsdp_amd.models import Llama
    expected = {"llama2_1.4b": 1.2e9, "llama2_7b": 6.5e9,
                "llama2_13b": 12e9, "llama2_34b": 32e9, "llama2_70b": 65e9,
                "llama3_194m_4k": 0.15e9, "llama3_1.8b": 1.5e9,
                "llama3_1.8b_4k": 1.5e9, "llama3_3.2b": 3.0e9,
                "llama3_3.2b_4k": 3.0e9, "llama3_8b": 7e9,
                "llama3_8b_4k": 7e9, "llama3_70b": 65e9,
                "llama3_70b_4k": 65e9}
    for name, lo in expected.items():
        cfg = get_model_config(name)
        with torch.device("meta"):
            m = Llama(cfg)
        n = sum(p.numel() for p in m.parameters())
        assert n >= lo, (name, n)
