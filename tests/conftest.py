import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def byte_tokenizer():
    from trlx_amd.utils.tokenizer import ByteTokenizer

    return ByteTokenizer()


def tiny_config(**overrides):
    from trlx_amd.models.nn.config import TransformerConfig

    base = dict(vocab_size=300, hidden_size=64, num_layers=2, num_heads=4,
                max_position_embeddings=256, arch_name="gpt2")
    base.update(overrides)
    return TransformerConfig(**base)


@pytest.fixture
def tiny_cfg():
    return tiny_config()


@pytest.fixture(autouse=True)
def _cuda_test_teardown():
    """Deterministic GPU cleanup between tests: each e2e test builds hipGraph
    decode/train engines whose pools otherwise get released by GC in the
    middle of a LATER test's allocations (observed segfault in
    test_ppo_dense_rewards after the examples suite on a GPU machine)."""
    yield
    if torch.cuda.is_available():
        import gc

        import trlx_amd

        # destroy captured graphs DETERMINISTICALLY before gc frees their
        # pools mid-way through the next test's allocations
        trlx_amd.release_graphs()
        gc.collect()
        torch.cuda.synchronize()
        torch.cuda.empty_cache()
