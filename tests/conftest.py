import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def narrow_model_factory(request):
    """Tiny CPU-trivial Llama factory (the reference's 'narrow model'
    pattern, tests/conftest.py:5-22 there)."""
    nlayers = getattr(request, "param", 4)

    class F:
        @staticmethod
        def create():
            from fms_fsdp_amd.models import Llama, LlamaConfig
            cfg = LlamaConfig(src_vocab_size=32, emb_dim=16, nheads=2,
                              kvheads=1, nlayers=nlayers,
                              max_expected_seq_len=32)
            m = Llama(cfg)
            m.reset_parameters()
            return m
    return F
