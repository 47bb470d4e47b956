#!/usr/bin/env python3
"""Minimal production-style serving front end for the LM decode path.

Wraps `LSTMModel.generate()` (hipGraph-captured, streaming-GEMV decode --
see README "Serving / decode") behind a FastAPI endpoint:

    python examples/serve.py --ckpt ckpt_final.pt --port 8000
    curl -s localhost:8000/generate -X POST -H 'content-type: application/json' \
         -d '{"tokens": [1, 2, 3], "steps": 16}'

With --demo (no checkpoint) it serves a random-init model -- useful for
latency/throughput testing of the serving stack itself.  Endpoints:
POST /generate (greedy/temperature/top-k/top-p/beam/EOS), POST
/generate_stream (SSE token stream; needs --continuous), GET /health,
GET /metrics (Prometheus).  --continuous turns on the continuous
batcher (lstm_tensorspark_amd/serving.py): concurrent requests share
batched decode steps.  .pt and .safetensors checkpoints both serve.

(No `from __future__ import annotations` here: FastAPI resolves the
request-model type hints at runtime, and postponed annotations of a
function-local pydantic model are unresolvable -- the endpoint would
silently degrade to a query parameter.)
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import Config, ModelConfig
from lstm_tensorspark_amd.models.lstm import LSTMModel


def build_app(model: LSTMModel, device: torch.device, batcher=None):
    import time

    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class GenRequest(BaseModel):
        tokens: list[int]
        steps: int = 32
        temperature: float = 0.0
        top_k: int = 0
        top_p: float = 0.0
        beam_size: int = 1
        eos_token: int = -1
        seed: int | None = None

    app = FastAPI(title="lstm_tensorspark_amd serving")

    # Prometheus metrics (optional: /metrics 404s if the client lib is
    # absent; the serving path itself never depends on it)
    try:
        from prometheus_client import (CONTENT_TYPE_LATEST,
                                       CollectorRegistry, Counter,
                                       Histogram, generate_latest)
        from starlette.responses import Response
        reg = CollectorRegistry()  # per-app: the default registry is
        # process-global and double-registers if build_app runs twice
        m_reqs = Counter("serve_generate_requests_total",
                         "generate requests", ["status"], registry=reg)
        m_toks = Counter("serve_generated_tokens_total", "tokens generated",
                         registry=reg)
        m_lat = Histogram("serve_generate_seconds",
                          "end-to-end /generate latency",
                          buckets=(.002, .005, .01, .025, .05, .1, .25,
                                   .5, 1., 2.5), registry=reg)

        @app.get("/metrics")
        def metrics():
            return Response(generate_latest(reg),
                            media_type=CONTENT_TYPE_LATEST)
    except ImportError:  # pragma: no cover
        m_reqs = m_toks = m_lat = None

    @app.post("/generate_stream")
    def generate_stream(req: GenRequest):
        """Server-sent events: tokens stream out as the continuous
        batcher produces them (requires --continuous; beam not streamed).
        Each event is a JSON list of new tokens; the final event is
        [DONE]."""
        import json as _json
        from starlette.responses import StreamingResponse
        if batcher is None:
            raise HTTPException(400, "streaming requires --continuous")
        if req.beam_size > 1:
            raise HTTPException(400, "beam search is not streamable")
        V = model.cfg.vocab_size
        if not req.tokens or any(t < 0 or t >= V for t in req.tokens):
            raise HTTPException(400, f"tokens must be in [0, {V})")
        if req.steps < 1 or req.steps > 4096:
            raise HTTPException(400, "steps must be in [1, 4096]")
        rid = batcher.submit(req.tokens, req.steps,
                             eos_token=req.eos_token,
                             temperature=req.temperature,
                             top_k=req.top_k, top_p=req.top_p,
                             seed=req.seed)

        def events():
            sent = len(req.tokens)
            try:
                while True:
                    toks, done = batcher.peek(rid)
                    if len(toks) > sent:
                        yield f"data: {_json.dumps(toks[sent:])}\n\n"
                        sent = len(toks)
                    if done:
                        batcher.try_result(rid)
                        yield "data: [DONE]\n\n"
                        return
                    time.sleep(0.003)
            finally:
                # client gone (GeneratorExit) or stream finished: free the
                # batch slot either way
                batcher.cancel(rid)

        return StreamingResponse(events(), media_type="text/event-stream")

    @app.get("/health")
    def health():
        r = {"ok": True, "device": str(device),
             "vocab": model.cfg.vocab_size,
             "hidden": model.cfg.hidden_size,
             "layers": model.cfg.num_layers,
             "weights_version": getattr(model, "_weights_version", 0),
             "continuous_batching": batcher is not None}
        if batcher is not None:
            r["engine"] = batcher.stats()
        return r

    @app.post("/generate")
    def generate(req: GenRequest):
        t0 = time.perf_counter()
        V = model.cfg.vocab_size
        if not req.tokens or any(t < 0 or t >= V for t in req.tokens):
            if m_reqs:
                m_reqs.labels(status="bad_request").inc()
            raise HTTPException(400, f"tokens must be in [0, {V})")
        if req.steps < 1 or req.steps > 4096:
            if m_reqs:
                m_reqs.labels(status="bad_request").inc()
            raise HTTPException(400, "steps must be in [1, 4096]")
        score = None
        if batcher is not None and req.beam_size <= 1:
            # continuous batching: greedy AND sampled requests share
            # batched decode steps (sampled ones use their own generator)
            rid = batcher.submit(req.tokens, req.steps,
                                 eos_token=req.eos_token,
                                 temperature=req.temperature,
                                 top_k=req.top_k, top_p=req.top_p,
                                 seed=req.seed)
            toks = batcher.result(rid, timeout=120.0)
            if m_reqs:
                m_reqs.labels(status="ok").inc()
                m_toks.inc(len(toks) - len(req.tokens))  # EOS may end early
                m_lat.observe(time.perf_counter() - t0)
            return {"tokens": toks, "generated": toks[len(req.tokens):]}
        prompt = torch.tensor(req.tokens, dtype=torch.long,
                              device=device).unsqueeze(1)  # (Tp, B=1)
        with torch.no_grad():
            if req.beam_size > 1:
                out, score = model.generate_beam(prompt, steps=req.steps,
                                                 beam_size=req.beam_size,
                                                 eos_token=req.eos_token)
            else:
                out = model.generate(prompt, steps=req.steps,
                                     temperature=req.temperature,
                                     top_k=req.top_k, top_p=req.top_p,
                                     eos_token=req.eos_token)
        if m_reqs:
            m_reqs.labels(status="ok").inc()
            m_toks.inc(out.shape[0] - len(req.tokens))
            m_lat.observe(time.perf_counter() - t0)
        resp = {"tokens": out[:, 0].tolist(),
                "generated": out[len(req.tokens):, 0].tolist()}
        if score is not None:
            resp["score"] = score
        return resp

    return app


def load_model(args) -> tuple[LSTMModel, torch.device]:
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    if args.demo:
        cfg = ModelConfig(input_size=args.hidden, hidden_size=args.hidden,
                          num_layers=2, head="lm", vocab_size=args.vocab)
        model = LSTMModel(cfg)
    elif args.ckpt.endswith(".safetensors"):
        # weights-only deployment artifact: model config travels in the
        # file's metadata (json) if the exporter put it there, else from CLI
        import json
        from safetensors import safe_open
        from lstm_tensorspark_amd.checkpoint import load_weights_safetensors
        with safe_open(args.ckpt, framework="pt", device="cpu") as f:
            meta = dict(f.metadata() or {})
        if "config" in meta:
            cfg = ModelConfig(**json.loads(meta["config"]))
        else:
            cfg = ModelConfig(input_size=args.hidden, hidden_size=args.hidden,
                              num_layers=2, head="lm", vocab_size=args.vocab)
        model = LSTMModel(cfg)
        load_weights_safetensors(args.ckpt, model)
    else:
        payload = torch.load(args.ckpt, map_location="cpu",
                             weights_only=False)
        cfg = Config.from_dict(payload["config"]).model
        model = LSTMModel(cfg)
        for name, p in model.weight_dict().items():
            p.data.copy_(payload["weights"][name])
    model = model.to(device)
    if device.type == "cuda":
        model = model.bfloat16()
    model.eval()
    return model, device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ckpt", default="", help="checkpoint (.pt) to serve")
    ap.add_argument("--demo", action="store_true",
                    help="serve a random-init model (stack testing)")
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--continuous", action="store_true",
                    help="continuous batching: concurrent requests (greedy "
                         "and sampled) share batched decode steps")
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args()
    if not args.demo and not args.ckpt:
        print("need --ckpt or --demo", file=sys.stderr)
        sys.exit(2)
    model, device = load_model(args)
    batcher = None
    if args.continuous:
        from lstm_tensorspark_amd.serving import ContinuousBatcher
        batcher = ContinuousBatcher(model, max_batch=args.max_batch)
        batcher.start()
    app = build_app(model, device, batcher=batcher)
    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
