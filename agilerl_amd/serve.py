"""Checkpoint serving: HTTP inference endpoint for trained agents.

Reference parity: the Arena SDK's deploy/infer surface
(``agilerl-arena/agilerl/arena/client.py`` deploy/inference endpoints) —
here as a self-hosted FastAPI app serving a classic-RL checkpoint:

    python -m agilerl_amd.serve ckpt.pt --host 0.0.0.0 --port 8000

Endpoints:
    GET  /healthz  -> {"status": "ok"}
    GET  /info     -> algorithm name, spaces, fitness history tail
    POST /predict  -> {"obs": [...]} => {"action": [...]} (deterministic)
    POST /reload   -> {"path": "..."} hot-swaps the checkpoint

Metrics: if prometheus_client is installed, request counts/latency are
exported on the same app at GET /metrics.
"""

from __future__ import annotations

import argparse
import threading
import time
from typing import Any, Dict, Optional

import numpy as np
from pydantic import BaseModel

__all__ = ["load_agent", "create_app", "main"]


class PredictRequest(BaseModel):
    obs: Any
    batch: bool = True


class ReloadRequest(BaseModel):
    path: str
    device: str = "cpu"


class GenerateRequest(BaseModel):
    prompt: Optional[str] = None
    messages: Optional[list] = None
    input_ids: Optional[list] = None
    max_new_tokens: int = 128
    temperature: float = 1.0
    do_sample: bool = True


def load_agent(path: str, device: str = "cpu"):
    """Rebuild any classic-RL agent from a single-file checkpoint."""
    from .algorithms.core.base import EvolvableAlgorithm

    return EvolvableAlgorithm.load(path, device=device)


def _space_summary(space) -> Dict[str, Any]:
    out: Dict[str, Any] = {"type": type(space).__name__}
    if hasattr(space, "shape") and getattr(space, "shape", None) is not None:
        out["shape"] = list(space.shape)
    if hasattr(space, "n"):
        out["n"] = int(space.n)
    return out


_METRICS_CACHE: Dict[str, Any] = {}


def _serve_metrics():
    """Process-wide Prometheus metrics (create_app may run many times)."""
    if "req" in _METRICS_CACHE:
        return _METRICS_CACHE["req"], _METRICS_CACHE["lat"]
    try:
        from prometheus_client import Counter, Histogram
    except ImportError:  # pragma: no cover
        _METRICS_CACHE["req"] = _METRICS_CACHE["lat"] = None
        return None, None
    _METRICS_CACHE["req"] = Counter("agilerl_serve_requests_total", "predict requests")
    _METRICS_CACHE["lat"] = Histogram("agilerl_serve_latency_seconds", "predict latency")
    return _METRICS_CACHE["req"], _METRICS_CACHE["lat"]


def create_app(agent, lock: Optional[threading.Lock] = None):
    """Build the FastAPI app around a loaded agent (swappable via /reload)."""
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="agilerl-amd agent server")
    state = {"agent": agent, "requests": 0, "started": time.time()}
    lock = lock or threading.Lock()

    req_counter, latency = _serve_metrics()
    if req_counter is not None:
        from prometheus_client import make_asgi_app

        app.mount("/metrics", make_asgi_app())

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "uptime_s": round(time.time() - state["started"], 1)}

    @app.get("/info")
    def info():
        a = state["agent"]
        return {
            "algo": type(a).__name__,
            "index": getattr(a, "index", None),
            "observation_space": _space_summary(a.observation_space),
            "action_space": _space_summary(a.action_space),
            "fitness_tail": [float(f) for f in getattr(a, "fitness", [])[-5:]],
            "requests_served": state["requests"],
        }

    @app.post("/predict")
    def predict(req: PredictRequest):
        t0 = time.time()
        obs = req.obs
        if isinstance(obs, dict):
            obs = {k: np.asarray(v, dtype=np.float32) for k, v in obs.items()}
            if not req.batch:
                obs = {k: v[None] for k, v in obs.items()}
        else:
            obs = np.asarray(obs, dtype=np.float32)
            if not req.batch:
                obs = obs[None]
        with lock:
            state["requests"] += 1
            try:
                action = state["agent"].get_action(obs, training=False)
            except Exception as exc:  # surface shape errors as 400s
                raise HTTPException(status_code=400, detail=str(exc))
        if isinstance(action, tuple):
            action = action[0]
        action = np.asarray(action)
        if not req.batch:
            action = action[0]
        if req_counter is not None:
            req_counter.inc()
            latency.observe(time.time() - t0)
        return {"action": action.tolist()}

    @app.post("/generate")
    def generate(req: GenerateRequest):
        """Text/token generation for LLM agents (has .generate + tokenizer)."""
        import torch

        a = state["agent"]
        if not hasattr(a, "generate"):
            raise HTTPException(status_code=400, detail="agent has no generate()")
        tok = getattr(a, "tokenizer", None)
        if req.input_ids is not None:
            ids = torch.as_tensor(req.input_ids, dtype=torch.long)
            if ids.dim() == 1:
                ids = ids.unsqueeze(0)
        elif req.prompt is not None or req.messages is not None:
            if tok is None:
                raise HTTPException(status_code=400,
                                    detail="agent has no tokenizer; pass input_ids")
            text = req.prompt
            if req.messages is not None:
                if getattr(tok, "chat_template", None):
                    text = tok.apply_chat_template(
                        req.messages, tokenize=False, add_generation_prompt=True
                    )
                else:
                    text = "\n".join(
                        f"{m.get('role', 'user').capitalize()}: {m.get('content', '')}"
                        for m in req.messages
                    ) + "\nAssistant:"
            ids = torch.as_tensor([tok.encode(text)], dtype=torch.long)
        else:
            raise HTTPException(status_code=400, detail="prompt, messages or input_ids required")
        ids = ids.to(getattr(a, "device", "cpu"))
        mask = torch.ones_like(ids)
        gen = a.generate_paged if getattr(a, "generation", "hf") == "paged" and \
            hasattr(a, "generate_paged") else a.generate
        with lock, torch.no_grad():
            state["requests"] += 1
            out = gen(ids, mask, max_new_tokens=req.max_new_tokens,
                      do_sample=req.do_sample, temperature=req.temperature)
        completion = out[:, ids.shape[1]:]
        resp = {"output_ids": out.cpu().tolist(),
                "completion_ids": completion.cpu().tolist()}
        if tok is not None:
            resp["completion"] = [tok.decode(row) for row in completion.cpu().tolist()]
        return resp

    @app.post("/generate/stream")
    def generate_stream(req: GenerateRequest):
        """NDJSON token stream: one {"token_id", "text"} line per generated
        token, then a final {"done": true, "completion": ...} line."""
        import json as _json

        import torch
        from fastapi.responses import StreamingResponse

        a = state["agent"]
        if not hasattr(a, "stream_generate"):
            raise HTTPException(status_code=400, detail="agent cannot stream")
        tok = getattr(a, "tokenizer", None)
        if req.input_ids is not None:
            ids = torch.as_tensor(req.input_ids, dtype=torch.long).reshape(-1)
        elif req.prompt is not None and tok is not None:
            ids = torch.as_tensor(tok.encode(req.prompt), dtype=torch.long)
        else:
            raise HTTPException(status_code=400,
                                detail="input_ids (or prompt + tokenizer) required")
        ids = ids.to(getattr(a, "device", "cpu"))
        mask = torch.ones_like(ids)

        def event_lines():
            produced = []
            with lock, torch.no_grad():
                state["requests"] += 1
                for t in a.stream_generate(
                    ids, mask, max_new_tokens=req.max_new_tokens,
                    do_sample=req.do_sample, temperature=req.temperature,
                ):
                    produced.append(t)
                    event = {"token_id": t}
                    if tok is not None:
                        event["text"] = tok.decode([t])
                    yield _json.dumps(event) + "\n"
            final = {"done": True, "completion_ids": produced}
            if tok is not None:
                final["completion"] = tok.decode(produced)
            yield _json.dumps(final) + "\n"

        return StreamingResponse(event_lines(), media_type="application/x-ndjson")

    @app.post("/reload")
    def reload(req: ReloadRequest):
        try:
            fresh = load_agent(req.path, device=req.device)
        except Exception as exc:
            raise HTTPException(status_code=400, detail=str(exc))
        with lock:
            state["agent"] = fresh
        return {"status": "reloaded", "algo": type(fresh).__name__}

    return app


def main(argv=None):
    p = argparse.ArgumentParser(description="Serve a trained agent over HTTP")
    p.add_argument("checkpoint")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--device", default="cpu")
    args = p.parse_args(argv)

    import uvicorn

    agent = load_agent(args.checkpoint, device=args.device)
    uvicorn.run(create_app(agent), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
