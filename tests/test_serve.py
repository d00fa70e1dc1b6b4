"""Serving front end (serve.py): generate path + HTTP app on CPU."""
import pytest
import torch

fastapi = pytest.importorskip("fastapi")


def make_tiny_server():
    import serve
    srv = serve.make_server(None, "tiny", kv_pool_tokens=8192,
                            kv_cache_dtype="bf16")
    return serve, srv


def test_server_generate_batch():
    _, srv = make_tiny_server()
    out = srv.generate(["hello world", "another prompt"], max_tokens=6,
                       temperature=1.0, top_p=0.95, n=2)
    assert len(out) == 2 and all(len(o) == 2 for o in out)
    assert all(isinstance(s, str) and s for o in out for s in o)


def test_http_endpoints():
    from fastapi.testclient import TestClient
    serve, srv = make_tiny_server()
    app = serve.build_app(srv)
    with TestClient(app) as client:
        r = client.get("/health")
        assert r.status_code == 200 and r.json()["status"] == "ok"
        r = client.post("/generate", json={"prompt": "hi there",
                                           "max_tokens": 4, "n": 1,
                                           "temperature": 0.0})
        assert r.status_code == 200
        body = r.json()
        assert len(body["completions"]) == 1
        assert len(body["completions"][0]) == 1


def test_real_checkpoint_server(tmp_path):
    """serve over a tiny local HF checkpoint + tokenizer (the --model path)."""
    import serve
    from nanorlhf_amd.data.tokenizer import make_tiny_tokenizer
    from nanorlhf_amd.models.config import ModelConfig
    from nanorlhf_amd.models.hf_import import save_hf_checkpoint
    from nanorlhf_amd.models.qwen2 import CausalLM

    torch.manual_seed(0)
    cfg = ModelConfig(vocab_size=2048, hidden_size=64, num_layers=2,
                      num_heads=4, num_kv_heads=2, head_dim=16,
                      intermediate_size=128, rope_theta=1e4, max_position=256,
                      dtype="float32", tie_word_embeddings=True)
    save_hf_checkpoint(CausalLM(cfg), str(tmp_path))
    make_tiny_tokenizer(str(tmp_path))
    srv = serve.make_server(str(tmp_path), "tiny", kv_pool_tokens=8192,
                            kv_cache_dtype="bf16")
    out = srv.generate(["What is 2+2?"], max_tokens=5, temperature=0.8,
                       top_p=0.9, n=1)
    assert len(out) == 1 and isinstance(out[0][0], str)
