"""OpenAI-compatible HTTP serving layer over the local engine.

Beyond-reference completeness: the reference (k_llms) is a CLIENT library
pointed at api.openai.com; this module puts the same consensus engine
behind the matching SERVER surface, so any stock OpenAI SDK can call it:

    python -m kllms_amd.server --model llama-3-8b --port 8000
    curl localhost:8000/v1/chat/completions -d '{"model": "llama-3-8b",
        "messages": [{"role": "user", "content": "hi"}], "n": 5}'

Routes:
- POST /v1/chat/completions — n>1 returns the k-LLMs consensus shape
  (choices[0] = consensus, choices[1..n] = originals, `likelihoods`);
  supports response_format json_schema/json_object (constrained decoding),
  tools (forced tool calls), logprobs, logit_bias, stop, seed.
- GET /v1/models — the loaded model (OpenAI list shape).
- GET /health — engine liveness + scheduler stats.

Concurrency: requests go through AsyncKLLMs, so concurrent HTTP calls
merge into the engine's continuous-batching scheduler exactly like the
async client path the bench measures.
"""

import argparse
import time
from typing import Any, Dict

__all__ = ["create_app", "main"]


def _error_response(message: str, status: int, err_type: str = "invalid_request_error",
                    code: str | None = None):
    from fastapi.responses import JSONResponse

    return JSONResponse(
        status_code=status,
        content={"error": {"message": message, "type": err_type,
                           "param": None, "code": code}},
    )


def create_app(client):
    """Build the FastAPI app over an AsyncKLLMs client (tests construct it
    with a tiny CPU model; `main()` with the configured model)."""
    from fastapi import FastAPI, Request

    from .engine.api import ContextLengthExceededError

    app = FastAPI(title="kllms_amd", docs_url=None, redoc_url=None)
    app.state.client = client
    app.state.started = time.time()

    @app.get("/health")
    async def health():
        sched = getattr(client.client, "_scheduler", None)
        return {
            "status": "ok",
            "model": client.client.config.model,
            "uptime_s": round(time.time() - app.state.started, 1),
            "scheduler": sched.stats if sched is not None else None,
        }

    @app.get("/v1/models")
    async def models():
        return {
            "object": "list",
            "data": [{
                "id": client.client.config.model,
                "object": "model",
                "created": int(app.state.started),
                "owned_by": "kllms_amd",
            }],
        }

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        import asyncio as _asyncio

        try:
            body: Dict[str, Any] = await request.json()
        except Exception:
            return _error_response("request body must be JSON", 400)
        inp = body.get("input")
        if isinstance(inp, str):
            inp = [inp]
        if not isinstance(inp, list) or not all(isinstance(t, str) for t in inp):
            return _error_response("'input' must be a string or list of strings", 400)
        model = body.get("model", "text-embedding-3-small")
        from .client import MAX_TOKENS_PER_MODEL

        if model not in MAX_TOKENS_PER_MODEL:
            return _error_response(
                f"unknown embedding model {model!r} "
                f"(available: {sorted(MAX_TOKENS_PER_MODEL)})", 404, code="model_not_found")
        try:
            resp = await _asyncio.to_thread(
                client.client.embeddings_create, input=inp, model=model)
        except ValueError as e:
            return _error_response(str(e), 400)
        return resp.model_dump()

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        try:
            body: Dict[str, Any] = await request.json()
        except Exception:
            return _error_response("request body must be JSON", 400)
        if body.get("stream"):
            return _error_response(
                "streaming is not supported (the reference forces stream=False)", 400)
        messages = body.get("messages")
        if not isinstance(messages, list) or not messages:
            return _error_response("'messages' must be a non-empty list", 400)
        known = {"model", "messages", "n", "temperature", "top_p", "top_k", "timeout",
                 "max_tokens", "max_completion_tokens", "stop", "seed",
                 "frequency_penalty", "presence_penalty", "logprobs",
                 "top_logprobs", "logit_bias", "response_format", "tools",
                 "tool_choice", "user", "stream"}
        unknown = set(body) - known
        if unknown:
            return _error_response(
                f"unknown parameter(s): {', '.join(sorted(unknown))}", 400)
        kwargs = {k: v for k, v in body.items()
                  if k in known and k not in ("stream", "user", "max_completion_tokens")}
        if "max_completion_tokens" in body and "max_tokens" not in body:
            kwargs["max_tokens"] = body["max_completion_tokens"]
        kwargs.setdefault("model", client.client.config.model)
        try:
            completion = await client.chat.completions.create(**kwargs)
        except ContextLengthExceededError as e:
            return _error_response(str(e), 400, code="context_length_exceeded")
        except ValueError as e:
            msg = str(e)
            if "model_not_found" in msg:
                return _error_response(msg, 404, code="model_not_found")
            return _error_response(msg, 400)
        return completion.model_dump()

    return app


def main(argv=None):
    ap = argparse.ArgumentParser(description="OpenAI-compatible server over the local engine")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--device", default=None, help="cuda:0 (default on GPU) or cpu")
    ap.add_argument("--max-kv-blocks", type=int, default=None)
    args = ap.parse_args(argv)

    import torch
    import uvicorn

    from . import AsyncKLLMs

    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    kw: Dict[str, Any] = dict(model=args.model, device=device,
                              use_hip_graphs=device.startswith("cuda"))
    if args.max_kv_blocks:
        kw["max_kv_blocks"] = args.max_kv_blocks
    client = AsyncKLLMs(**kw)
    uvicorn.run(create_app(client), host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
