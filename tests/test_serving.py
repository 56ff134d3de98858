"""Serving endpoint: health + completion round-trip on a tiny model (CPU,
FastAPI TestClient — no sockets)."""


def test_serve_completion():
    import sys, os
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from fastapi.testclient import TestClient
    from examples.serving.serve import create_app

    cfg = {
        "tokenizer": "bytes",
        "max_new_tokens": 8,
        "model": {
            "vocab_size": 256, "hidden_size": 32, "intermediate_size": 64,
            "num_layers": 1, "num_attention_heads": 2, "num_kv_heads": 1,
            "max_position_embeddings": 64,
        },
    }
    client = TestClient(create_app(cfg))
    assert client.get("/health").json()["status"] == "ok"
    r = client.post("/v1/completions", json={"prompt": "hello", "max_tokens": 4})
    assert r.status_code == 200
    body = r.json()
    assert body["tokens"] == 4 and isinstance(body["text"], str)
    # sampling options accepted
    r2 = client.post("/v1/completions", json={
        "prompt": "hi", "max_tokens": 3, "temperature": 0.8, "top_p": 0.9})
    assert r2.status_code == 200


def test_serve_batched_prompts():
    import sys, os
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from fastapi.testclient import TestClient
    from examples.serving.serve import create_app

    cfg = {
        "tokenizer": "bytes",
        "max_new_tokens": 4,
        "model": {
            "vocab_size": 256, "hidden_size": 32, "intermediate_size": 64,
            "num_layers": 1, "num_attention_heads": 2, "num_kv_heads": 1,
            "max_position_embeddings": 64,
        },
    }
    client = TestClient(create_app(cfg))
    r = client.post("/v1/completions",
                    json={"prompt": ["hello there", "hi"], "max_tokens": 3})
    assert r.status_code == 200
    body = r.json()
    assert isinstance(body["text"], list) and len(body["text"]) == 2
    assert body["tokens"] == 3
