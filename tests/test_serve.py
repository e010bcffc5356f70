"""serve.py — the HTTP serving runtime around the cached batch decoder
(no reference analog; CPU-tested with the in-process ASGI client)."""

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from progen_amd import ProGenBase, ProGenConfig  # noqa: E402
from serve import create_app  # noqa: E402


def _app():
    cfg = ProGenConfig(num_tokens=256, dim=16, depth=2, dim_head=8, heads=2,
                       window_size=8, seq_len=64, global_mlp_depth=1)
    torch.manual_seed(5)
    module = ProGenBase(cfg).eval()
    return create_app(module, cfg, meta={"trained_sequences": 0})


def test_healthz_and_info():
    client = TestClient(_app())
    assert client.get("/healthz").json() == {"status": "ok"}
    info = client.get("/info").json()
    assert info["seq_len"] == 64
    assert info["params"] > 0
    assert info["graph"] is False  # CPU


def test_generate_single_and_batch():
    client = TestClient(_app())
    r = client.post("/generate", json={"prime": "# M", "num_tokens": 32,
                                       "seed": 7})
    assert r.status_code == 200, r.text
    body = r.json()
    assert len(body["sequences"]) == 1
    assert isinstance(body["sequences"][0], str)
    assert body["ms"] > 0
    # batch of two primes -> two rows
    rb = client.post("/generate", json={"primes": ["# M", "# G"],
                                        "num_tokens": 32, "seed": 7})
    assert rb.status_code == 200, rb.text
    assert len(rb.json()["sequences"]) == 2


def test_generate_deterministic_with_seed():
    client = TestClient(_app())
    body = {"prime": "# M", "num_tokens": 48, "seed": 123}
    a = client.post("/generate", json=body).json()
    b = client.post("/generate", json=body).json()
    assert a["tokens"] == b["tokens"]


def test_generate_validation_errors():
    client = TestClient(_app())
    assert client.post("/generate", json={}).status_code == 400
    assert client.post("/generate",
                       json={"prime": "# M",
                             "num_tokens": 1000}).status_code == 400
    assert client.post("/generate",
                       json={"prime": "X" * 40,
                             "num_tokens": 16}).status_code == 400
    assert client.post("/generate",
                       json={"prime": "# M", "num_tokens": 16,
                             "top_k": -2}).status_code == 400


def test_greedy_matches_direct_decoder():
    """top_k null -> greedy argmax; the endpoint must return exactly what
    decode.sample_cached_batch returns for the same prime."""
    from progen_amd.data import encode_tokens
    from progen_amd.decode import sample_cached_batch

    cfg = ProGenConfig(num_tokens=256, dim=16, depth=2, dim_head=8, heads=2,
                       window_size=8, seq_len=64, global_mlp_depth=1)
    torch.manual_seed(5)
    module = ProGenBase(cfg).eval()
    client = TestClient(create_app(module, cfg))
    r = client.post("/generate", json={"prime": "# M", "num_tokens": 32,
                                       "top_k": None}).json()
    prime = torch.tensor([0] + encode_tokens("# M"), dtype=torch.long)
    with torch.no_grad():
        want = sample_cached_batch(module, [prime], 32, top_k=None)
    assert r["tokens"][0] == want[0][prime.shape[0]:].tolist()


def test_generate_empty_prime_is_unconditional():
    client = TestClient(_app())
    r = client.post("/generate", json={"prime": "", "num_tokens": 16,
                                       "seed": 3})
    assert r.status_code == 200, r.text
    assert len(r.json()["tokens"][0]) == 15  # BOS + 15 generated


def test_load_module_from_checkpoint(tmp_path, monkeypatch):
    """serve.load_module rebuilds the model from the checkpoint's stored
    model_config (sample.py's recipe) and reports trained_sequences."""
    from progen_amd.checkpoint import get_checkpoint_fns, tensors_to_numpy
    from serve import load_module

    cfg_kwargs = dict(num_tokens=256, dim=16, depth=2, dim_head=8, heads=2,
                      window_size=8, seq_len=64, global_mlp_depth=1)
    torch.manual_seed(9)
    m = ProGenBase(ProGenConfig(**cfg_kwargs))
    _, _, save = get_checkpoint_fns(str(tmp_path / "ckpts"))
    save({
        "next_seq_index": 1234,
        "params": tensors_to_numpy(dict(m.state_dict())),
        "optim_state": None,
        "model_config": cfg_kwargs,
        "run_id": None,
    }, keep_last_n=1)
    module, cfg, meta = load_module(str(tmp_path / "ckpts"))
    assert cfg.seq_len == 64 and cfg.depth == 2
    assert meta["trained_sequences"] == 1234
    # weights actually loaded, not re-initialized
    got = dict(module.state_dict())["embed.weight"]
    torch.testing.assert_close(got, m.state_dict()["embed.weight"])
