"""Generation server tests (trlx_amd/serve.py)."""

import pytest
import torch
from fastapi.testclient import TestClient

from trlx_amd.models.nn.transformer import CausalTransformer
from trlx_amd.serve import create_app
from trlx_amd.utils.tokenizer import ByteTokenizer

from conftest import tiny_config


def test_serve_generate_roundtrip():
    torch.manual_seed(0)
    model = CausalTransformer(tiny_config())
    tok = ByteTokenizer()
    app = create_app(model, tok, device=torch.device("cpu"))
    client = TestClient(app)
    assert client.get("/health").json()["ok"]
    r = client.post("/generate", json={"prompts": ["hello", "hi"], "max_new_tokens": 4,
                                       "do_sample": False})
    assert r.status_code == 200
    comps = r.json()["completions"]
    assert len(comps) == 2
    # greedy generation is deterministic
    r2 = client.post("/generate", json={"prompts": ["hello", "hi"], "max_new_tokens": 4,
                                        "do_sample": False})
    assert r2.json()["completions"] == comps


def test_serve_loads_saved_checkpoint(tmp_path):
    from trlx_amd.models.modeling_ppo import AutoModelForCausalLMWithValueHead
    from trlx_amd.serve import load

    m = AutoModelForCausalLMWithValueHead.from_config(tiny_config())
    d = str(tmp_path / "hf_model")
    m.save_pretrained(d)
    ByteTokenizer().save_pretrained(d)
    model, tok = load(d)
    app = create_app(model, tok, device=torch.device("cpu"))
    r = TestClient(app).post("/generate", json={"prompts": ["ab"], "max_new_tokens": 2})
    assert r.status_code == 200 and len(r.json()["completions"]) == 1


def test_serve_continuous_batching_mode():
    from fastapi.testclient import TestClient

    from trlx_amd.models.nn.transformer import CausalTransformer
    from trlx_amd.serve import create_app
    from trlx_amd.utils.tokenizer import get_tokenizer

    import torch
    from conftest import tiny_config

    torch.manual_seed(0)
    model = CausalTransformer(tiny_config(vocab_size=300)).eval()
    tok = get_tokenizer("byte")
    app = create_app(model, tok, continuous_slots=2, cache_len=128)
    with TestClient(app) as client:
        r = client.post("/generate", json={"prompts": ["hello", "hi there", "x"],
                                           "max_new_tokens": 5})
        assert r.status_code == 200
        outs = r.json()["completions"]
        assert len(outs) == 3
    app.state.batcher.close()
