"""REST text-generation server.

Capability analog of reference megatron/core/inference/text_generation_server/
(Flask MegatronServer + tools/run_text_generation_server.py), built on FastAPI/
uvicorn (in-image). POST /api/generate {"prompts": [...], "max_tokens": ...,
"temperature"/"top_k"/"top_p"/"greedy"/"logprobs"} -> generations. The engine
runs under a lock; continuous batching batches concurrent requests naturally
when driven through the dynamic engine's queue.
"""

import threading

from fastapi import FastAPI, Request

from megatron_amd.inference.sampling import SamplingParams


def create_app(engine, tokenizer=None):

    app = FastAPI(title="megatron_amd text generation")
    lock = threading.Lock()

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.post("/api/generate")
    async def generate(request: Request):
        req = await request.json()
        prompts = req["prompts"]
        logprobs = bool(req.get("logprobs", False))
        params = SamplingParams(
            max_tokens=int(req.get("max_tokens", 64)),
            temperature=float(req.get("temperature", 1.0)),
            top_k=int(req.get("top_k", 0)), top_p=float(req.get("top_p", 0.0)),
            greedy=bool(req.get("greedy", False)), return_log_probs=logprobs,
            top_n_logprobs=int(req.get("top_n_logprobs", 0)),
            repetition_penalty=float(req.get("repetition_penalty", 1.0)),
            min_p=float(req.get("min_p", 0.0)),
            logit_bias={int(k): float(v) for k, v in (req.get("logit_bias") or {}).items()} or None,
            stop_strings=tuple(req.get("stop_strings", ())),
            stop_on_eod=bool(req.get("stop_on_eod", True)),
            seed=req.get("seed"))
        with lock:
            results = engine.generate(prompts, params)
        return {
            "generations": [
                {
                    "request_id": r.request_id,
                    "tokens": r.output_tokens,
                    "text": r.text,
                    "logprobs": r.log_probs if logprobs else None,
                    "top_logprobs": r.top_logprobs or None,
                }
                for r in results
            ]
        }

    @app.post("/api/chat")
    async def chat(request: Request):
        """Chat completion: messages are rendered with the tokenizer's chat
        template and generated as one continuation."""
        req = await request.json()
        assert tokenizer is not None, "chat endpoint needs a tokenizer"
        from megatron_amd.tokenizers import apply_chat_template

        prompt_ids = apply_chat_template(tokenizer, req["messages"], add_generation_prompt=True)
        params = SamplingParams(
            max_tokens=int(req.get("max_tokens", 256)),
            temperature=float(req.get("temperature", 1.0)),
            top_k=int(req.get("top_k", 0)), top_p=float(req.get("top_p", 0.0)),
            greedy=bool(req.get("greedy", False)), seed=req.get("seed"))
        with lock:
            r = engine.generate([prompt_ids], params)[0]
        return {"message": {"role": "assistant",
                            "content": tokenizer.detokenize(r.output_tokens)},
                "tokens": r.output_tokens}

    def _sp(req, defaults_max=256):
        return SamplingParams(
            max_tokens=int(req.get("max_tokens", defaults_max)),
            temperature=float(req.get("temperature", 1.0)),
            top_k=int(req.get("top_k", 0)), top_p=float(req.get("top_p", 0.0)),
            greedy=float(req.get("temperature", 1.0)) == 0.0,
            return_log_probs=bool(req.get("logprobs", False)),
            min_p=float(req.get("min_p", 0.0)),
            logit_bias={int(k): float(v) for k, v in (req.get("logit_bias") or {}).items()} or None,
            stop_strings=tuple(req.get("stop", []) if isinstance(req.get("stop", []), list)
                               else [req["stop"]]),
            seed=req.get("seed"))

    @app.post("/v1/completions")
    async def v1_completions(request: Request):
        """OpenAI-compatible completions endpoint (prompt string or token
        list; temperature 0 = greedy; `stop` strings honored)."""
        req = await request.json()
        assert tokenizer is not None, "/v1 endpoints need a tokenizer"
        prompt = req.get("prompt", "")
        prompts = prompt if isinstance(prompt, list) and prompt and isinstance(prompt[0], (list, str)) \
            else [prompt]
        n = int(req.get("n", 1))
        prompts = [p for p in prompts for _ in range(n)]
        with lock:
            results = engine.generate(prompts, _sp(req, defaults_max=16))
        return {
            "object": "text_completion",
            "model": req.get("model", "megatron_amd"),
            "choices": [
                {"index": i, "text": r.text,
                 "finish_reason": ("length" if len(r.output_tokens) >= int(req.get("max_tokens", 16))
                                   else "stop"),
                 "logprobs": ({"token_logprobs": r.log_probs}
                              if req.get("logprobs") else None)}
                for i, r in enumerate(results)
            ],
            "usage": {
                "prompt_tokens": sum(len(r.prompt_tokens) for r in results),
                "completion_tokens": sum(len(r.output_tokens) for r in results),
            },
        }

    @app.post("/v1/chat/completions")
    async def v1_chat_completions(request: Request):
        """OpenAI-compatible chat endpoint over the tokenizer's template."""
        req = await request.json()
        assert tokenizer is not None, "/v1 endpoints need a tokenizer"
        from megatron_amd.tokenizers import apply_chat_template

        prompt_ids = apply_chat_template(tokenizer, req["messages"],
                                         add_generation_prompt=True)
        with lock:
            r = engine.generate([prompt_ids], _sp(req))[0]
        return {
            "object": "chat.completion",
            "model": req.get("model", "megatron_amd"),
            "choices": [{"index": 0,
                         "message": {"role": "assistant",
                                     "content": tokenizer.detokenize(r.output_tokens)},
                         "finish_reason": ("length" if len(r.output_tokens) >= int(req.get("max_tokens", 256))
                                           else "stop")}],
            "usage": {"prompt_tokens": len(prompt_ids),
                      "completion_tokens": len(r.output_tokens)},
        }

    @app.get("/metrics")
    async def metrics():
        """Prometheus-format serving metrics (reference observability role;
        prometheus_client is in-image)."""
        from fastapi.responses import PlainTextResponse

        lines = []

        def gauge(name, value, help_):
            lines.append(f"# HELP {name} {help_}")
            lines.append(f"# TYPE {name} gauge")
            lines.append(f"{name} {value}")

        gauge("megatron_amd_active_requests", len(getattr(engine, "active", [])),
              "requests currently decoding")
        gauge("megatron_amd_waiting_requests", len(getattr(engine, "waiting", [])),
              "requests queued for prefill")
        gauge("megatron_amd_finished_requests", len(getattr(engine, "finished", {})),
              "finished results not yet collected")
        gauge("megatron_amd_preempted_requests", len(getattr(engine, "preempted", [])),
              "requests offloaded to host")
        ctx = getattr(engine, "context", None)
        if ctx is not None:
            alloc = ctx.allocator
            gauge("megatron_amd_kv_blocks_free", alloc.num_free, "free KV blocks")
            gauge("megatron_amd_kv_blocks_total", alloc.num_blocks, "total KV blocks")
            if hasattr(alloc, "hits"):
                gauge("megatron_amd_prefix_cache_hits", alloc.hits, "prefix-cache block hits")
                gauge("megatron_amd_prefix_cache_misses", alloc.misses, "prefix-cache block misses")
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.post("/api/stream")
    async def stream(request: Request):
        """SSE token streaming for a single prompt (dynamic engine only):
        each generated token is emitted as a `data:` event as soon as its
        decode step completes; the final event carries done+text."""
        import json as _json

        from fastapi.responses import StreamingResponse

        req = await request.json()
        prompt = req["prompt"]
        params = SamplingParams(
            max_tokens=int(req.get("max_tokens", 64)),
            temperature=float(req.get("temperature", 1.0)),
            top_k=int(req.get("top_k", 0)), top_p=float(req.get("top_p", 0.0)),
            greedy=bool(req.get("greedy", False)), seed=req.get("seed"))

        detok = None
        if tokenizer is not None:
            from megatron_amd.tokenizers import IncrementalDetokenizer

            detok = IncrementalDetokenizer(tokenizer)

        def emit(t):
            ev = {"token": t}
            if detok is not None:
                delta = detok.put(t)
                if delta:
                    ev["text"] = delta  # stable utf-8 prefix delta
            return f"data: {_json.dumps(ev)}\n\n"

        def gen():
            with lock:
                rid = engine.add_request(prompt, params)
                sent = 0
                while True:
                    engine.step()
                    if rid in engine.finished:
                        r = engine.finished.pop(rid)
                        for t in r.output_tokens[sent:]:
                            yield emit(t)
                        tail = detok.flush() if detok is not None else None
                        yield f"data: {_json.dumps({'done': True, 'text': r.text, 'tail': tail})}\n\n"
                        return
                    cur = next((q for q in engine.active if q.rid == rid), None)
                    if cur is not None and len(cur.result.output_tokens) > sent:
                        for t in cur.result.output_tokens[sent:]:
                            yield emit(t)
                        sent = len(cur.result.output_tokens)

        return StreamingResponse(gen(), media_type="text/event-stream")

    return app


def run_server(engine, tokenizer=None, host: str = "127.0.0.1", port: int = 5000):
    import uvicorn

    uvicorn.run(create_app(engine, tokenizer), host=host, port=port, log_level="warning")
