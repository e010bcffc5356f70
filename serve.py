"""Serving runtime: the cached incremental decoder behind an HTTP API.

No reference analog (lucidrains/progen ships only the sample.py CLI) —
this is the deployment surface for the MI355X serving path: one model
instance, batched `decode.sample_cached_batch` per request (per-layer
recurrent caches, O(window) attention per token), optional hipGraph
replay of the per-token step on GPU.

    python serve.py --checkpoint_path ./ckpts --port 8000
    curl -s localhost:8000/generate -d '{"primes": ["# M"], "num_tokens": 256}'

Endpoints:
    GET  /healthz            liveness
    GET  /info               model config, params, device, decode mode
    POST /generate           {"primes": [str] (or "prime": str),
                              "num_tokens": int <= seq_len (default seq_len),
                              "top_k": int (default 25; 0/null = greedy),
                              "seed": int (optional, deterministic sampling)}
                             -> {"sequences": [str], "tokens": [[int]],
                                 "ms": float}

Requests are serialized through one lock (single model instance; the
batch dimension inside ONE request is where serving throughput comes
from — 2.9k tok/s aggregate at batch 128 on ProGen-1.2B, see
profiles/r02_fp8_and_decode.md). With --graph the per-token step is
captured per request (the decoder manages its own capture); the capture
cost amortizes over the generated length — persistent cross-request
capture keyed on batch shape is the next step if request shapes repeat.
"""

import threading
import time
from typing import List, Optional

import click
import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.checkpoint import get_checkpoint_fns, numpy_to_tensors
from progen_amd.data import decode_tokens, encode_tokens
from progen_amd.decode import sample_cached_batch
from progen_amd.utils import load_dotenv


def create_app(module: ProGenBase, cfg: ProGenConfig, *,
               graph: bool = False, meta: Optional[dict] = None):
    """Build the FastAPI app around a ready (device-placed, eval) module."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    app = FastAPI(title="progen-mi355x", docs_url=None, redoc_url=None)
    lock = threading.Lock()
    device = next(module.parameters()).device
    info = {
        "model_config": cfg.to_dict() if hasattr(cfg, "to_dict") else vars(cfg),
        "params": module.num_params(),
        "seq_len": cfg.seq_len,
        "device": str(device),
        "graph": bool(graph and device.type == "cuda"),
        **(meta or {}),
    }

    class GenerateRequest(BaseModel):
        primes: Optional[List[str]] = None
        prime: Optional[str] = None
        num_tokens: Optional[int] = None
        top_k: Optional[int] = 25
        seed: Optional[int] = None

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/info")
    def get_info():
        return info

    @app.post("/generate")
    def generate(req: GenerateRequest):
        primes = req.primes if req.primes is not None else (
            [req.prime] if req.prime is not None else None)
        if not primes:
            raise HTTPException(400, "provide 'primes' (list) or 'prime'")
        length = req.num_tokens or cfg.seq_len
        if not 1 <= length <= cfg.seq_len:
            raise HTTPException(400, f"num_tokens must be in [1, {cfg.seq_len}]")
        if req.top_k is not None and req.top_k < 0:
            raise HTTPException(400, "top_k must be >= 0 (0/null = greedy)")
        top_k = req.top_k if req.top_k else None
        # explicit BOS column, as the samplers' add_bos does (the byte
        # tokenizer reserves 0 for BOS/pad)
        rows = []
        for p in primes:
            toks = encode_tokens(p)
            if len(toks) + 1 >= length:
                raise HTTPException(400, f"prime longer than num_tokens: {p!r}")
            rows.append(torch.tensor([0] + toks, dtype=torch.long))
        gen = torch.Generator().manual_seed(req.seed) \
            if req.seed is not None else None
        t0 = time.perf_counter()
        with lock, torch.no_grad():
            out = sample_cached_batch(module, rows, length, top_k=top_k,
                                      generator=gen, graph=info["graph"])
        ms = (time.perf_counter() - t0) * 1e3
        seqs, toks_out = [], []
        for row, prime_row in zip(out, rows):
            tail = row[prime_row.shape[0]:]
            toks_out.append(tail.tolist())
            seqs.append(decode_tokens(tail.numpy()))
        return {"sequences": seqs, "tokens": toks_out, "ms": ms}

    return app


def load_module(checkpoint_path: str):
    """sample.py's loading recipe: lexically-last checkpoint, model FROM
    the stored model_config (reference: sample.py:46-47)."""
    _, get_last_checkpoint, _ = get_checkpoint_fns(checkpoint_path)
    last = get_last_checkpoint()
    if last is None:
        raise SystemExit(f"no checkpoints found at {checkpoint_path}")
    cfg = ProGenConfig.from_dict(last["model_config"])
    module = ProGenBase(cfg)
    module.load_state_dict(
        {k: torch.as_tensor(v)
         for k, v in numpy_to_tensors(last["params"]).items()}, strict=False)
    device = torch.device("cuda") if torch.cuda.is_available() \
        else torch.device("cpu")
    module = module.to(device)
    if device.type == "cuda":
        module = module.to(torch.bfloat16)
        module.rotary_sin = module.rotary_sin.float()
        module.rotary_cos = module.rotary_cos.float()
    module.eval()
    meta = {"trained_sequences": max(last.get("next_seq_index", 0), 0)}
    return module, cfg, meta


@click.command()
@click.option("--checkpoint_path", default="./ckpts")
@click.option("--host", default="127.0.0.1")
@click.option("--port", default=8000)
@click.option("--graph", default=False, is_flag=True,
              help="on GPU: replay the per-token decode step as one "
                   "captured hipGraph (2.1x the eager cached step)")
def main(checkpoint_path, host, port, graph):
    load_dotenv()
    from progen_amd.tuning import enable_tuned_gemms
    enable_tuned_gemms()
    import uvicorn
    module, cfg, meta = load_module(checkpoint_path)
    app = create_app(module, cfg, graph=graph, meta=meta)
    uvicorn.run(app, host=host, port=port, log_level="info")


if __name__ == "__main__":
    main()
