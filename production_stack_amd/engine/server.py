"""OpenAI-compatible HTTP front for the MI355X engine.

API surface the router depends on (SURVEY.md section 2.9 / reference
main_router proxy targets): /v1/completions, /v1/chat/completions (SSE with
usage in the final chunk), /v1/models, /tokenize, /detokenize, /health,
/metrics (vllm:* series the router scrapes), /is_sleeping, /sleep,
/wake_up, /v1/load_lora_adapter, /v1/unload_lora_adapter.
"""

from __future__ import annotations

import argparse
import asyncio
import contextlib
import json
import logging
import time
import uuid
from typing import Optional

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse

from production_stack_amd.engine.async_engine import AsyncEngine
from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.tokenizer import render_chat

logger = logging.getLogger("engine.server")


def _params_from_body(body: dict, max_model_len: int) -> SamplingParams:
    max_tokens = (
        body.get("max_tokens")
        or body.get("max_completion_tokens")
        or 128
    )
    temperature = body.get("temperature")
    if temperature is None:
        temperature = 1.0
    stop = body.get("stop")
    if isinstance(stop, str):
        stop = [stop]
    return SamplingParams(
        max_tokens=int(max_tokens),
        temperature=float(temperature),
        top_p=float(body.get("top_p") or 1.0),
        top_k=int(body.get("top_k") or -1),
        ignore_eos=bool(body.get("ignore_eos", False)),
        stop=list(stop) if stop else [],
        stop_token_ids=list(body.get("stop_token_ids") or []),
        seed=body.get("seed"),
        presence_penalty=float(body.get("presence_penalty") or 0.0),
        frequency_penalty=float(body.get("frequency_penalty") or 0.0),
        repetition_penalty=float(body.get("repetition_penalty") or 1.0),
        logit_bias={int(k): float(v)
                    for k, v in (body.get("logit_bias") or {}).items()},
        min_tokens=int(body.get("min_tokens") or 0),
        min_p=float(body.get("min_p") or 0.0),
        prompt_logprobs=(
            int(body["prompt_logprobs"])
            if body.get("prompt_logprobs") is not None else None
        ),
        logprobs=_parse_logprobs(body),
        response_format=_guided_from_body(body),
    )


def _guided_from_body(body: dict):
    """response_format json modes, plus vLLM-style guided_choice /
    guided_regex mapped onto the same guided-state machinery."""
    rf = body.get("response_format")
    if isinstance(rf, dict) and rf.get("type") in ("json_object",
                                                   "json_schema"):
        return rf
    if body.get("guided_choice"):
        return {"type": "choice",
                "choices": list(body["guided_choice"])}
    if body.get("guided_regex"):
        return {"type": "regex", "pattern": str(body["guided_regex"])}
    return None


def _parse_logprobs(body: dict):
    """completions: logprobs=<int>; chat: logprobs=true [+ top_logprobs]."""
    lp = body.get("logprobs")
    if lp is None or lp is False:
        return None
    if lp is True:
        return int(body.get("top_logprobs") or 1)
    return int(lp)


def build_server(engine: LLMEngine, served_model: str,
                 chat_template: Optional[str] = None,
                 extra_model_names: Optional[list] = None) -> FastAPI:
    # custom chat template (--chat-template FILE): vLLM-compatible jinja2
    # rendering with messages/tools/add_generation_prompt; falls back to
    # the built-in role-tagged template (tokenizer.render_chat)
    _tpl_cache = {}

    def _render_messages(messages, tools=None):
        if chat_template:
            tpl = _tpl_cache.get("t")
            if tpl is None:
                import jinja2

                tpl = jinja2.Environment(
                    trim_blocks=True, lstrip_blocks=True
                ).from_string(chat_template)
                _tpl_cache["t"] = tpl
            return tpl.render(
                messages=messages, tools=tools,
                add_generation_prompt=True, bos_token="", eos_token="",
            )
        return render_chat(messages, tools=tools)

    async_engine = AsyncEngine(engine)

    @contextlib.asynccontextmanager
    async def lifespan(app: FastAPI):
        async_engine.start(asyncio.get_running_loop())
        cfg = engine.config
        if getattr(cfg, "kv_transfer", None):
            from production_stack_amd.parallel.kv_transfer import (
                KVTransferService,
            )

            kt = cfg.kv_transfer
            app.state.kv_service = KVTransferService(
                engine,
                kv_rank=kt["rank"],
                kv_world=kt["world"],
                master_port=kt.get("master_port", 14500),
                side_port=kt.get("side_port", 14001),
                host=kt.get("host", "127.0.0.1"),
            )
            await app.state.kv_service.start_side_channel()
        if getattr(cfg, "kv_controller_url", None):
            host, _, port = cfg.kv_controller_url.rpartition(":")
            from production_stack_amd.kvpool.client import EngineReporter

            app.state.kv_reporter = EngineReporter(
                engine,
                url=getattr(cfg, "advertise_url", "http://127.0.0.1:8000"),
                host=host or "127.0.0.1",
                port=int(port),
            )
            app.state.kv_reporter.start()
        yield
        reporter = getattr(app.state, "kv_reporter", None)
        if reporter is not None:
            await reporter.stop()
        kv_service = getattr(app.state, "kv_service", None)
        if kv_service is not None:
            await kv_service.stop()
        async_engine.stop()

    app = FastAPI(
        title=f"production-stack-amd engine ({served_model})",
        lifespan=lifespan,
    )

    import os

    api_key = os.environ.get("VLLM_API_KEY")
    if api_key:
        @app.middleware("http")
        async def _auth(request: Request, call_next):
            if request.url.path not in ("/health", "/metrics"):
                auth = request.headers.get("authorization", "")
                if auth != f"Bearer {api_key}":
                    return JSONResponse(
                        status_code=401,
                        content={"error": "invalid or missing API key"},
                    )
            return await call_next(request)
    app.state.engine = engine
    app.state.async_engine = async_engine
    app.state.served_model = served_model
    app.state.lora_adapters = {}
    app.state.start_time = time.time()

    # ------------------------------------------------------------------
    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/v1/models")
    async def models():
        data = [
            {
                "id": name,
                "object": "model",
                "created": int(app.state.start_time),
                "owned_by": "production-stack-amd",
            }
            for name in [served_model] + list(extra_model_names or [])
        ]
        for name in app.state.lora_adapters:
            data.append(
                {
                    "id": name,
                    "object": "model",
                    "created": int(time.time()),
                    "owned_by": "production-stack-amd",
                    "parent": served_model,
                }
            )
        return {"object": "list", "data": data}

    @app.get("/metrics")
    async def metrics():
        m = engine.engine_metrics()
        lines = []
        for k, v in m.items():
            name = f"vllm:{k}"
            kind = "counter" if k.endswith("_total") else "gauge"
            lines.append(f"# TYPE {name} {kind}")
            lines.append(
                f'{name}{{model_name="{served_model}"}} {v}'
            )
        q = m.get("gpu_prefix_cache_queries_total", 0.0)
        h = m.get("gpu_prefix_cache_hits_total", 0.0)
        lines.append("# TYPE vllm:gpu_prefix_cache_hit_rate gauge")
        lines.append(
            f'vllm:gpu_prefix_cache_hit_rate{{model_name="{served_model}"}} '
            f"{h / q if q else 0.0}"
        )
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.post("/tokenize")
    async def tokenize(request: Request):
        body = await request.json()
        if "messages" in body:
            text = _render_messages(body["messages"])
        else:
            text = str(body.get("prompt", ""))
        tokens = engine.tokenizer.encode(text)
        return {"tokens": tokens, "count": len(tokens), "max_model_len":
                engine.config.max_model_len}

    @app.post("/detokenize")
    async def detokenize(request: Request):
        body = await request.json()
        return {"prompt": engine.tokenizer.decode(body.get("tokens", []))}

    # ---- sleep / wake -------------------------------------------------
    @app.get("/is_sleeping")
    async def is_sleeping():
        return {"is_sleeping": engine.is_sleeping}

    @app.post("/sleep")
    async def sleep(level: int = 1):
        engine.sleep(level)
        return {"status": "ok"}

    @app.post("/wake_up")
    async def wake_up():
        engine.wake_up()
        return {"status": "ok"}

    # ---- LoRA (load/unload surface for the operator's LoraAdapter CRD) --
    @app.post("/v1/load_lora_adapter")
    async def load_lora(request: Request):
        body = await request.json()
        name = body.get("lora_name")
        if not name:
            return JSONResponse(
                status_code=400, content={"error": "lora_name required"}
            )
        path = body.get("lora_path", "")
        try:
            if path:
                engine.load_lora(name, path)
        except (OSError, ValueError) as e:
            return JSONResponse(
                status_code=400,
                content={"error": f"failed to load adapter: {e}"},
            )
        app.state.lora_adapters[name] = path
        return {"status": "ok"}

    @app.post("/v1/unload_lora_adapter")
    async def unload_lora(request: Request):
        body = await request.json()
        name = body.get("lora_name")
        engine.unload_lora(name)
        app.state.lora_adapters.pop(name, None)
        return {"status": "ok"}

    # ---- embeddings / rerank / score -----------------------------------
    def _embed_texts(texts):
        import torch

        from production_stack_amd.engine.models.llama import BatchMeta

        vecs = []
        with engine.lock:
            runner = engine.runner
            bm = engine.block_manager
            for text in texts:
                toks = engine.tokenizer.encode(text)[
                    : engine.config.max_model_len - 1
                ] or [engine.model_cfg.bos_token_id]
                # mean-pool final hidden states over a scratch forward that
                # borrows cache blocks and returns them immediately
                n_blocks = (len(toks) + bm.block_size - 1) // bm.block_size
                blocks = []
                for _ in range(n_blocks):
                    b = bm._pop_block()
                    if b is None:
                        break
                    blocks.append(b)
                if len(blocks) < n_blocks:
                    for b in blocks:
                        bm.free.append(b)
                    raise RuntimeError("no KV blocks free for embedding")
                dev = runner.device
                T = len(toks)
                bs = bm.block_size
                slots = [
                    blocks[i // bs] * bs + i % bs for i in range(T)
                ]
                meta = BatchMeta(
                    positions=torch.arange(
                        T, dtype=torch.int32, device=dev
                    ),
                    slot_mapping=torch.tensor(
                        slots, dtype=torch.long, device=dev
                    ),
                    num_prefill_tokens=T,
                    prefill_token_seq=torch.zeros(
                        T, dtype=torch.int32, device=dev
                    ),
                    prefill_token_pos=torch.arange(
                        T, dtype=torch.int32, device=dev
                    ),
                    prefill_block_tables=torch.tensor(
                        [blocks], dtype=torch.int32, device=dev
                    ),
                    num_decode_seqs=0,
                    decode_seq_lens=None,
                    decode_block_tables=None,
                    prefill_tiles=torch.tensor(
                        [[0, t0, t0, min(64, T - t0)]
                         for t0 in range(0, T, 64)],
                        dtype=torch.int32,
                        device=dev,
                    ),
                )
                hidden = runner.model(
                    torch.tensor(toks, dtype=torch.long, device=dev),
                    meta,
                    runner.kv_caches,
                )
                v = hidden.float().mean(dim=0)
                v = v / (v.norm() + 1e-6)
                vecs.append(v.cpu().tolist())
                for b in blocks:
                    bm.free.append(b)
        return vecs

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        body = await request.json()
        inp = body.get("input")
        texts = [inp] if isinstance(inp, str) else list(inp or [])
        try:
            vecs = await asyncio.to_thread(_embed_texts, texts)
        except RuntimeError as e:
            return JSONResponse(status_code=503, content={"error": str(e)})
        if body.get("encoding_format") == "base64":
            import base64 as _b64
            import struct as _struct

            def enc(v):
                return _b64.b64encode(
                    _struct.pack(f"<{len(v)}f", *v)).decode()
        else:
            def enc(v):
                return v
        return {
            "object": "list",
            "model": body.get("model", served_model),
            "data": [
                {"object": "embedding", "index": i, "embedding": enc(v)}
                for i, v in enumerate(vecs)
            ],
            "usage": {"prompt_tokens": sum(len(t.split()) for t in texts),
                      "total_tokens": sum(len(t.split()) for t in texts)},
        }

    async def _rerank_impl(request: Request):
        body = await request.json()
        query = body.get("query", "")
        docs = body.get("documents") or []
        texts = [query] + [
            d if isinstance(d, str) else d.get("text", "") for d in docs
        ]
        try:
            vecs = await asyncio.to_thread(_embed_texts, texts)
        except RuntimeError as e:
            return JSONResponse(status_code=503, content={"error": str(e)})
        import math

        qv = vecs[0]
        results = []
        for i, dv in enumerate(vecs[1:]):
            score = sum(a * b for a, b in zip(qv, dv))
            results.append(
                {"index": i, "relevance_score": score,
                 "document": docs[i] if isinstance(docs[i], dict)
                 else {"text": docs[i]}}
            )
        results.sort(key=lambda r: -r["relevance_score"])
        return {"model": body.get("model", served_model),
                "results": results}

    @app.post("/v1/rerank")
    async def rerank(request: Request):
        return await _rerank_impl(request)

    @app.post("/rerank")
    async def rerank_alias(request: Request):
        return await _rerank_impl(request)

    async def _score_impl(request: Request):
        body = await request.json()
        t1 = body.get("text_1", "")
        t2s = body.get("text_2")
        t2s = [t2s] if isinstance(t2s, str) else list(t2s or [])
        try:
            vecs = await asyncio.to_thread(_embed_texts, [t1] + t2s)
        except RuntimeError as e:
            return JSONResponse(status_code=503, content={"error": str(e)})
        qv = vecs[0]
        return {
            "object": "list",
            "model": body.get("model", served_model),
            "data": [
                {"index": i,
                 "score": sum(a * b for a, b in zip(qv, dv))}
                for i, dv in enumerate(vecs[1:])
            ],
        }

    @app.post("/v1/score")
    async def score(request: Request):
        return await _score_impl(request)

    @app.post("/score")
    async def score_alias(request: Request):
        return await _score_impl(request)

    # ---- completions ---------------------------------------------------
    async def _run_completion(request: Request, chat: bool):
        body = await request.json()
        params = _params_from_body(body, engine.config.max_model_len)
        rid = (
            request.headers.get("x-request-id")
            or f"cmpl-{uuid.uuid4().hex[:24]}"
        )
        mm_embeds = None
        tools = body.get("tools") if chat else None
        if body.get("tool_choice") == "none":
            tools = None
        if chat:
            messages = body.get("messages") or []
            if _has_media(messages):
                prompt, mm_embeds = _assemble_multimodal(messages, engine)
            else:
                prompt = _render_messages(messages, tools=tools)
        else:
            p = body.get("prompt", "")
            if isinstance(p, list) and p and isinstance(p[0], int):
                prompt = p
            elif isinstance(p, list):
                prompt = "\n".join(str(x) for x in p)
            else:
                prompt = str(p)
        if isinstance(prompt, str):
            prompt_tokens = engine.tokenizer.encode(prompt)
        else:
            prompt_tokens = prompt
        if len(prompt_tokens) + 1 > engine.config.max_model_len:
            return JSONResponse(
                status_code=400,
                content={
                    "error": {
                        "message": (
                            f"prompt ({len(prompt_tokens)} tokens) exceeds "
                            f"max_model_len {engine.config.max_model_len}"
                        ),
                        "type": "invalid_request_error",
                    }
                },
            )
        created = int(time.time())
        model_name = body.get("model", served_model)
        obj = "chat.completion" if chat else "text_completion"
        # OpenAI classic `echo`: prepend the prompt text to the output
        # (completions only; prompt logprobs are not echoed)
        echo_text = ""
        if not chat and body.get("echo"):
            echo_text = (
                prompt if isinstance(prompt, str)
                else engine.tokenizer.decode(list(prompt_tokens))
            )

        # disaggregated prefill: decode role pulls the prefiller's KV blocks
        # into the local prefix cache before scheduling (parallel/kv_transfer)
        kvp = body.get("kv_transfer_params") or {}
        kv_service = getattr(app.state, "kv_service", None)
        if (
            kv_service is not None
            and kvp.get("do_remote_prefill")
            and kvp.get("remote_request_id")
        ):
            try:
                await kv_service.pull_into_prefix_cache(
                    kvp["remote_request_id"],
                    list(prompt_tokens),
                    kvp.get("remote_host") or "127.0.0.1",
                    int(kvp.get("remote_port") or 14001),
                    int(kvp.get("remote_engine_id") or 0),
                )
            except (ConnectionError, OSError, asyncio.TimeoutError) as e:
                logger.warning("KV pull failed (%s); recomputing prefill", e)

        if body.get("stream"):
            # OpenAI stream_options: usage rides the final chunk only
            # when asked for (the bundled benchmark harness asks; the
            # reference harness depends on it, multi_round_qa.py)
            include_usage = bool(
                (body.get("stream_options") or {}).get("include_usage",
                                                       True)
            )

            async def gen():
                try:
                    first = True
                    n_out = 0
                    # with tools, buffer the whole turn so tool_calls can
                    # be parsed and emitted as one structured chunk
                    buffered = "" if (chat and tools) else None
                    async for out in async_engine.generate(
                        rid, prompt_tokens, params, mm_embeds=mm_embeds
                    ):
                        n_out = out.num_output_tokens
                        if buffered is not None:
                            buffered += out.text_delta
                            if not out.finished:
                                continue
                            rest, tcs = parse_tool_calls(buffered)
                            if tcs:
                                for j, tc in enumerate(tcs):
                                    tc["index"] = j
                                delta = {"role": "assistant",
                                         "content": rest or None,
                                         "tool_calls": tcs}
                                fr = "tool_calls"
                            else:
                                delta = {"role": "assistant",
                                         "content": buffered}
                                fr = out.finish_reason
                            choice = {"index": 0, "delta": delta,
                                      "finish_reason": fr}
                        elif chat:
                            delta = (
                                {"role": "assistant", "content": out.text_delta}
                                if first
                                else {"content": out.text_delta}
                            )
                            choice = {
                                "index": 0,
                                "delta": delta,
                                "finish_reason": (
                                    out.finish_reason if out.finished else None
                                ),
                            }
                        else:
                            choice = {
                                "index": 0,
                                "text": (
                                    echo_text + out.text_delta
                                    if first else out.text_delta
                                ),
                                "finish_reason": (
                                    out.finish_reason if out.finished else None
                                ),
                            }
                        first = False
                        if params.logprobs is not None and out.new_logprobs:
                            tok_strs = [
                                engine.tokenizer.decode_token(t)
                                for t in out.new_token_ids
                            ]

                            def _tops(j):
                                tl = out.new_top_logprobs
                                if not tl or j >= len(tl) or not tl[j]:
                                    return None
                                return [
                                    {"token":
                                     engine.tokenizer.decode_token(t),
                                     "logprob": v}
                                    for t, v in tl[j]
                                ]

                            if chat:
                                choice["logprobs"] = {"content": [
                                    {"token": ts, "logprob": lv,
                                     **({"top_logprobs": _tops(j)}
                                        if _tops(j) is not None else {})}
                                    for j, (ts, lv) in enumerate(zip(
                                        tok_strs, out.new_logprobs))
                                ]}
                            else:
                                choice["logprobs"] = {
                                    "tokens": tok_strs,
                                    "token_logprobs": list(
                                        out.new_logprobs),
                                }
                        chunk = {
                            "id": rid,
                            "object": obj + ".chunk" if chat else obj,
                            "created": created,
                            "model": model_name,
                            "choices": [choice],
                        }
                        if out.finished and include_usage:
                            chunk["usage"] = {
                                "prompt_tokens": out.num_prompt_tokens,
                                "completion_tokens": n_out,
                                "total_tokens": out.num_prompt_tokens + n_out,
                            }
                        yield f"data: {json.dumps(chunk)}\n\n".encode()
                    yield b"data: [DONE]\n\n"
                except asyncio.CancelledError:
                    async_engine.abort(rid)
                    raise

            return StreamingResponse(gen(), media_type="text/event-stream")

        # non-streaming (n parallel choices share the prompt's KV via the
        # prefix cache; each choice is its own engine request)
        n_choices = max(int(body.get("n") or 1), 1)

        async def run_one(idx: int):
            import dataclasses as _dc

            p_i = params
            if idx > 0:
                p_i = _dc.replace(
                    params,
                    seed=(params.seed + idx) if params.seed is not None
                    else None,
                )
            r_i = rid if idx == 0 else f"{rid}-{idx}"
            text, toks, lps, tops = "", [], [], []
            plp = None
            reason, npr = None, len(prompt_tokens)
            try:
                async for out in async_engine.generate(
                    r_i, prompt_tokens, p_i, mm_embeds=mm_embeds
                ):
                    text += out.text_delta
                    toks.extend(out.new_token_ids)
                    if out.prompt_logprobs is not None:
                        plp = out.prompt_logprobs
                    if out.new_logprobs:
                        lps.extend(out.new_logprobs)
                        tl = out.new_top_logprobs or []
                        tops.extend(
                            tl + [None] * (len(out.new_logprobs)
                                           - len(tl))
                        )
                    if out.finished:
                        reason = out.finish_reason
                        npr = out.num_prompt_tokens or npr
            except asyncio.CancelledError:
                # client disconnected mid-generation: free the engine
                # slot instead of decoding tokens nobody will read
                async_engine.abort(r_i)
                raise
            return text, toks, lps, reason, npr, plp, tops

        results = await asyncio.gather(
            *[run_one(i) for i in range(n_choices)]
        )
        text, tokens, _, finish_reason, n_prompt = results[0][:5]
        choices = []
        for i, (txt, toks, lps, reason, _, plp, tops) in enumerate(results):
            if chat:
                message = {"role": "assistant", "content": txt}
                if tools:
                    rest, tcs = parse_tool_calls(txt)
                    if tcs:
                        message = {"role": "assistant",
                                   "content": rest or None,
                                   "tool_calls": tcs}
                        reason = "tool_calls"
                choice = {
                    "index": i,
                    "finish_reason": reason,
                    "message": message,
                }
            else:
                choice = {
                    "index": i,
                    "finish_reason": reason,
                    "text": (echo_text + txt) if echo_text else txt,
                }
            if params.logprobs is not None and lps:
                tok_strs = [engine.tokenizer.decode_token(t) for t in toks]

                def top_fmt(pos):
                    if pos >= len(tops) or not tops[pos]:
                        return None
                    return [
                        {"token": engine.tokenizer.decode_token(t),
                         "logprob": v}
                        for t, v in tops[pos]
                    ]

                if chat:
                    choice["logprobs"] = {
                        "content": [
                            {"token": ts, "logprob": lp,
                             **({"top_logprobs": top_fmt(j)}
                                if top_fmt(j) is not None else {})}
                            for j, (ts, lp) in enumerate(
                                zip(tok_strs, lps))
                        ]
                    }
                else:
                    lgp = {
                        "tokens": tok_strs,
                        "token_logprobs": lps,
                    }
                    if tops:
                        lgp["top_logprobs"] = [
                            {engine.tokenizer.decode_token(t): v
                             for t, v in ((tops[j] or [])
                                          if j < len(tops) else [])}
                            for j in range(len(toks))
                        ]
                    choice["logprobs"] = lgp
            if plp is not None:
                choice["prompt_logprobs"] = plp
            choices.append(choice)
        total_out = sum(len(r[1]) for r in results)
        resp = {
            "id": rid,
            "object": obj,
            "created": created,
            "model": model_name,
            "choices": choices,
            "usage": {
                "prompt_tokens": n_prompt,
                "completion_tokens": total_out,
                "total_tokens": n_prompt + total_out,
            },
        }
        if kvp.get("do_remote_decode"):
            if kv_service is not None:
                resp["kv_transfer_params"] = kv_service.register_prefilled(
                    rid, list(prompt_tokens)
                )
            else:
                resp["kv_transfer_params"] = handle_kv_transfer_params(
                    app, kvp, rid
                )
        return JSONResponse(resp)

    # ---- multimodal: vision content parts + audio endpoints -----------
    def _has_media(messages) -> bool:
        for m in messages:
            c = m.get("content")
            if isinstance(c, list):
                for part in c:
                    if isinstance(part, dict) and part.get("type") in (
                            "image_url", "input_audio"):
                        return True
        return False

    def _assemble_multimodal(messages, eng):
        """Token stream with per-media placeholder ids + (offset, embeds)
        pairs (vision/audio adapter injection; engine/models/multimodal)."""
        import base64 as b64

        from production_stack_amd.engine.models.multimodal import (
            decode_image,
            decode_wav,
            media_placeholder_tokens,
        )

        toks: list = []
        mm: list = []
        vocab = eng.model_cfg.vocab_size

        def add_text(t: str) -> None:
            toks.extend(eng.tokenizer.encode(t))

        for m in messages:
            add_text(f"<|{m.get('role', 'user')}|> ")
            c = m.get("content")
            parts = c if isinstance(c, list) else [
                {"type": "text", "text": c or ""}]
            for part in parts:
                if not isinstance(part, dict):
                    continue
                kind = part.get("type")
                if kind == "text":
                    add_text(part.get("text", ""))
                elif kind == "image_url":
                    url = (part.get("image_url") or {}).get("url", "")
                    if not url.startswith("data:"):
                        raise ValueError(
                            "only data: image URLs are supported offline")
                    data = b64.b64decode(url.split(",", 1)[1])
                    emb = eng.get_vision_encoder()(
                        decode_image(data).to(eng.device))
                    mm.append((len(toks), emb))
                    toks.extend(media_placeholder_tokens(
                        data, emb.shape[0], vocab))
                elif kind == "input_audio":
                    data = b64.b64decode(
                        (part.get("input_audio") or {}).get("data", ""))
                    emb = eng.get_audio_encoder()(
                        decode_wav(data).to(eng.device))
                    mm.append((len(toks), emb))
                    toks.extend(media_placeholder_tokens(
                        data, emb.shape[0], vocab))
        add_text("<|assistant|>")
        return toks, mm

    def _parse_multipart(body: bytes, content_type: str):
        """Minimal multipart/form-data parser (python-multipart is not in
        the image). Returns {field: bytes}."""
        import re as _re

        m = _re.search(r'boundary="?([^";,]+)"?', content_type)
        if not m:
            return {}
        sep = b"--" + m.group(1).encode()
        fields = {}
        for chunk in body.split(sep):
            if b"\r\n\r\n" not in chunk:
                continue
            head, _, payload = chunk.partition(b"\r\n\r\n")
            nm = _re.search(rb'name="([^"]+)"', head)
            if nm:
                fields[nm.group(1).decode()] = payload.rstrip(b"\r\n-")
        return fields

    async def _transcribe_impl(request: Request, translate: bool):
        ctype = request.headers.get("content-type", "")
        raw = await request.body()
        language = None
        fmt = "json"
        if ctype.startswith("multipart/form-data"):
            fields = _parse_multipart(raw, ctype)
            wav = fields.get("file", b"")
            language = (fields.get("language") or b"").decode() or None
            fmt = (fields.get("response_format") or b"json").decode()
        elif ctype.startswith("application/json"):
            body = json.loads(raw or b"{}")
            import base64 as b64

            wav = b64.b64decode(body.get("file", ""))
            language = body.get("language")
            fmt = body.get("response_format", "json")
        else:  # raw audio body
            wav = raw
        if not wav:
            return JSONResponse(status_code=400, content={
                "error": {"message": "no audio payload",
                          "type": "invalid_request_error"}})
        from production_stack_amd.engine.models.multimodal import (
            decode_wav,
            media_placeholder_tokens,
        )

        emb = engine.get_audio_encoder()(
            decode_wav(wav).to(engine.device))
        task = "translate" if translate else "transcribe"
        head = engine.tokenizer.encode(f"<|audio|> {task} ")
        toks = list(head) + media_placeholder_tokens(
            wav, emb.shape[0], engine.model_cfg.vocab_size)
        mm = [(len(head), emb)]
        rid = f"transcribe-{uuid.uuid4().hex[:16]}"
        text = ""
        params = SamplingParams(max_tokens=48, temperature=0.0,
                                ignore_eos=False)
        async for out in async_engine.generate(rid, toks, params,
                                               mm_embeds=mm):
            text += out.text_delta
        if fmt == "text":
            return PlainTextResponse(text.strip())
        return {"text": text.strip(),
                **({"language": language} if language else {})}

    @app.post("/v1/audio/transcriptions")
    async def audio_transcriptions(request: Request):
        return await _transcribe_impl(request, translate=False)

    @app.post("/v1/audio/translations")
    async def audio_translations(request: Request):
        return await _transcribe_impl(request, translate=True)

    @app.get("/v1/audio/voices")
    async def audio_voices():
        """Voice inventory for the TTS endpoint (reference proxies
        /v1/audio/voices; the tone-vocoder keys pitch off the voice)."""
        return {"voices": ["alloy", "echo", "fable", "onyx", "nova",
                           "shimmer"]}

    @app.post("/v1/audio/speech")
    async def audio_speech(request: Request):
        """OpenAI TTS API shape (reference routes /v1/audio/speech to the
        engine, main_router.py:265-267). Synthesis here is a
        deterministic tone-sequence vocoder — one 40 ms voiced segment
        per input token, pitch keyed to the token hash — not a neural
        TTS: it keeps the endpoint API-complete and pipeline-testable
        (valid WAV out, duration ∝ input length)."""
        import io
        import math
        import wave

        body = await request.json()
        text = str(body.get("input", ""))
        speed = float(body.get("speed") or 1.0)
        speed = min(max(speed, 0.25), 4.0)
        fmt = body.get("response_format", "wav")
        if fmt not in ("wav",):
            return JSONResponse(
                status_code=400,
                content={"error": {"message":
                         f"response_format {fmt!r} unsupported (wav only)",
                         "type": "invalid_request_error"}},
            )
        sr = 16000
        seg = max(int(sr * 0.04 / speed), 1)
        tokens = engine.tokenizer.encode(text) or [0]
        samples = []
        import zlib

        voice = str(body.get("voice", "alloy"))
        for t in tokens[:2048]:
            # stable across processes (hash() is seed-randomized)
            f = 110.0 + (zlib.crc32(f"{t}:{voice}".encode()) % 520)
            for i in range(seg):
                env = min(i, seg - i, seg // 8 + 1) / (seg // 8 + 1)
                samples.append(int(
                    12000 * env * math.sin(2 * math.pi * f * i / sr)
                ))
        buf = io.BytesIO()
        with wave.open(buf, "wb") as w:
            w.setnchannels(1)
            w.setsampwidth(2)
            w.setframerate(sr)
            w.writeframes(b"".join(
                s.to_bytes(2, "little", signed=True) for s in samples
            ))
        return Response(content=buf.getvalue(), media_type="audio/wav")

    @app.get("/version")
    async def version():
        return {"version": "production-stack-amd 0.1 (MI355X)"}

    @app.post("/v1/responses")
    async def responses(request: Request):
        """Minimal OpenAI Responses API: `input` string or message list
        runs through the chat path; output in Responses format."""
        body = await request.json()
        inp = body.get("input", "")
        if isinstance(inp, str):
            messages = [{"role": "user", "content": inp}]
        else:
            messages = inp
        chat_body = dict(body)
        chat_body.pop("input", None)
        chat_body["messages"] = messages
        prompt_tokens = engine.tokenizer.encode(_render_messages(messages))
        params = _params_from_body(chat_body, engine.config.max_model_len)
        rid = f"resp-{uuid.uuid4().hex[:12]}"
        text = ""
        async for out in async_engine.generate(rid, prompt_tokens, params):
            text += out.text_delta
        return JSONResponse({
            "id": rid,
            "object": "response",
            "created_at": int(time.time()),
            "model": body.get("model", served_model),
            "status": "completed",
            "output": [{
                "type": "message",
                "role": "assistant",
                "content": [{"type": "output_text", "text": text}],
            }],
            "output_text": text,
        })

    @app.post("/v1/messages")
    async def anthropic_messages(request: Request):
        """Anthropic-style Messages API mapped onto the chat pipeline
        (reference proxies /v1/messages verbatim to the engine,
        main_router.py:51-300; vLLM serves it natively). Supports
        string/content-block messages, system prompts, stop_sequences
        and the Anthropic SSE event framing when stream=true."""
        body = await request.json()

        def _blocks_to_text(content) -> str:
            if isinstance(content, list):
                return "".join(
                    b.get("text", "") for b in content
                    if isinstance(b, dict) and b.get("type") == "text"
                )
            return str(content)

        messages = []
        if body.get("system"):
            messages.append(
                {"role": "system",
                 "content": _blocks_to_text(body["system"])}
            )
        for m in body.get("messages", []):
            messages.append(
                {"role": m.get("role", "user"),
                 "content": _blocks_to_text(m.get("content", ""))}
            )
        chat_body = {
            "max_tokens": body.get("max_tokens", 256),
            "temperature": body.get("temperature", 1.0),
            "top_p": body.get("top_p", 1.0),
            "stop": body.get("stop_sequences") or [],
        }
        if body.get("top_k") is not None:
            chat_body["top_k"] = body["top_k"]
        prompt_tokens = engine.tokenizer.encode(_render_messages(messages))
        if len(prompt_tokens) + 1 > engine.config.max_model_len:
            return JSONResponse(
                status_code=400,
                content={"type": "error", "error": {
                    "type": "invalid_request_error",
                    "message": "prompt exceeds max_model_len"}},
            )
        params = _params_from_body(chat_body, engine.config.max_model_len)
        rid = f"msg-{uuid.uuid4().hex[:12]}"
        model_name = body.get("model", served_model)
        stop_map = {"stop": "end_turn", "length": "max_tokens",
                    "stop_sequence": "stop_sequence"}

        if body.get("stream"):

            async def gen():
                def ev(name, payload):
                    return (f"event: {name}\n"
                            f"data: {json.dumps(payload)}\n\n")

                yield ev("message_start", {
                    "type": "message_start",
                    "message": {
                        "id": rid, "type": "message", "role": "assistant",
                        "content": [], "model": model_name,
                        "stop_reason": None, "stop_sequence": None,
                        "usage": {"input_tokens": len(prompt_tokens),
                                  "output_tokens": 0},
                    }})
                yield ev("content_block_start", {
                    "type": "content_block_start", "index": 0,
                    "content_block": {"type": "text", "text": ""}})
                n_out, reason = 0, "end_turn"
                async for out in async_engine.generate(
                    rid, prompt_tokens, params
                ):
                    n_out = out.num_output_tokens
                    if out.finished:
                        reason = stop_map.get(
                            out.finish_reason or "stop", "end_turn")
                    if out.text_delta:
                        yield ev("content_block_delta", {
                            "type": "content_block_delta", "index": 0,
                            "delta": {"type": "text_delta",
                                      "text": out.text_delta}})
                yield ev("content_block_stop",
                         {"type": "content_block_stop", "index": 0})
                yield ev("message_delta", {
                    "type": "message_delta",
                    "delta": {"stop_reason": reason,
                              "stop_sequence": None},
                    "usage": {"output_tokens": n_out}})
                yield ev("message_stop", {"type": "message_stop"})

            return StreamingResponse(gen(), media_type="text/event-stream")

        text, n_out, reason = "", 0, "end_turn"
        async for out in async_engine.generate(rid, prompt_tokens, params):
            text += out.text_delta
            n_out = out.num_output_tokens
            if out.finished:
                reason = stop_map.get(out.finish_reason or "stop",
                                      "end_turn")
        return JSONResponse({
            "id": rid,
            "type": "message",
            "role": "assistant",
            "content": [{"type": "text", "text": text}],
            "model": model_name,
            "stop_reason": reason,
            "stop_sequence": None,
            "usage": {"input_tokens": len(prompt_tokens),
                      "output_tokens": n_out},
        })

    @app.post("/v1/completions")
    async def completions(request: Request):
        return await _run_completion(request, chat=False)

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        return await _run_completion(request, chat=True)

    return app


_TOOL_CALL_RE = None


def parse_tool_calls(text: str):
    """Extract Hermes-style <tool_call>{json}</tool_call> spans emitted by
    the model into OpenAI `message.tool_calls`; returns (remaining_text,
    tool_calls). Engine-side counterpart of the reference stack's vLLM
    tool-call parsers (reference tutorials/13-tool-enabled-installation.md
    surface)."""
    global _TOOL_CALL_RE
    import re

    if _TOOL_CALL_RE is None:
        _TOOL_CALL_RE = re.compile(
            r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.DOTALL
        )
    calls = []
    for i, m in enumerate(_TOOL_CALL_RE.finditer(text)):
        try:
            obj = json.loads(m.group(1))
        except json.JSONDecodeError:
            continue
        if not isinstance(obj, dict) or "name" not in obj:
            continue
        calls.append({
            "id": f"call_{uuid.uuid4().hex[:24]}",
            "type": "function",
            "function": {
                "name": str(obj["name"]),
                "arguments": json.dumps(obj.get("arguments") or {}),
            },
        })
    remaining = _TOOL_CALL_RE.sub("", text).strip()
    return remaining, calls


def handle_kv_transfer_params(app, kv_params: dict, request_id: str) -> dict:
    """Echo the PD handshake shape (reference request.py:793-851). The
    actual block transfer over RCCL/xGMI lives in parallel/kv_transfer.py;
    in the HTTP server we advertise this engine as the KV source."""
    cfg = app.state.engine.config
    out = dict(kv_params)
    if kv_params.get("do_remote_decode"):
        out.update(
            {
                "do_remote_decode": False,
                "do_remote_prefill": True,
                "remote_engine_id": getattr(cfg, "engine_id", "engine-0"),
                "remote_block_ids": [],
                "remote_host": None,
                "remote_port": getattr(cfg, "kv_transfer_port", 14001),
            }
        )
    return out


def main() -> None:
    import uvicorn

    ap = argparse.ArgumentParser(
        description="MI355X-native OpenAI-compatible serving engine"
    )
    ap.add_argument("model", nargs="?", default="llama-3-8b")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--served-model-name", default=None, nargs="+",
                    help="one or more public names for this model "
                         "(vLLM-style; first is primary, the rest alias)")
    ap.add_argument("--max-model-len", type=int, default=4096)
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--max-num-seqs", type=int, default=256)
    ap.add_argument("--max-num-batched-tokens", type=int, default=2048)
    ap.add_argument("--gpu-memory-utilization", type=float, default=0.85)
    ap.add_argument("--num-gpu-blocks", type=int, default=None)
    ap.add_argument("--enable-prefix-caching", action="store_true",
                    default=True)
    ap.add_argument("--no-enable-prefix-caching", dest="enable_prefix_caching",
                    action="store_false")
    ap.add_argument("--enable-chunked-prefill", action="store_true",
                    default=True)
    ap.add_argument("--tensor-parallel-size", type=int, default=1)
    ap.add_argument("--pipeline-parallel-size", type=int, default=1)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--device", default=None, help="cpu forces CPU mode")
    ap.add_argument("--weights-path", default=None)
    ap.add_argument("--tokenizer", default="synthetic")
    ap.add_argument("--kv-controller-url", default=None,
                    help="host:port of the KV controller for kvaware routing")
    ap.add_argument("--advertise-url", default=None,
                    help="this engine's URL as seen by the router")
    ap.add_argument("--kv-role", default=None,
                    choices=[None, "kv_producer", "kv_consumer"])
    ap.add_argument("--kv-rank", type=int, default=None,
                    help="rank in the KV-transfer group (disagg prefill)")
    ap.add_argument("--kv-world", type=int, default=0)
    ap.add_argument("--kv-master-port", type=int, default=14500)
    ap.add_argument("--kv-side-port", type=int, default=14001)
    ap.add_argument("--cpu-offload-gb", type=float, default=0.0)
    ap.add_argument("--offload-dtype", default="bf16",
                    choices=["bf16", "int8"],
                    help="host-pool record dtype (int8 = rowwise-quant)")
    ap.add_argument("--remote-kv-url", default=None,
                    help="cacheserver data plane host:port (shared KV tier)")
    ap.add_argument("--kv-cache-dtype", default="auto",
                    choices=["auto", "bf16", "fp8", "fp8_e4m3"])
    ap.add_argument("--quantization", default=None,
                    choices=[None, "fp8"],
                    help="fp8 = OCP e4m3 weights through the fp8 MFMA pipe")
    ap.add_argument("--no-unified-mixed-steps", dest="unified_mixed_steps",
                    action="store_false", default=True)
    ap.add_argument("--async-scheduling", action="store_true",
                    help="one-step-lagged sampling (greedy-exact overlap)")
    ap.add_argument("--num-speculative-tokens", type=int, default=0,
                    help="n-gram (prompt-lookup) speculative decoding")
    ap.add_argument("--speculative-model", default=None,
                    help="draft model architecture for model-based "
                         "speculation (same vocab; engine/draft.py)")
    ap.add_argument("--speculative-weights-path", default=None)
    ap.add_argument("--enable-lora", action="store_true",
                    help="enable graph-safe BGMV adapter slots")
    ap.add_argument("--max-loras", type=int, default=4)
    ap.add_argument("--max-lora-rank", type=int, default=16)
    ap.add_argument("--kv-transfer-config", default=None,
                    help="vLLM-style JSON; kv_role/kv_rank/kv_parallel_size "
                         "are mapped onto the native RCCL KV transfer")
    ap.add_argument("--chat-template", default=None,
                    help="path to a chat template file (tokenizer)")
    # accepted for `vllm serve` CLI compatibility (no-ops here)
    ap.add_argument("--revision", default=None)
    ap.add_argument("--runner", default=None)
    ap.add_argument("--convert", default=None)
    ap.add_argument("--trust-remote-code", action="store_true")
    args = ap.parse_args()
    if args.kv_transfer_config:
        import json as _json

        ktc = _json.loads(args.kv_transfer_config)
        args.kv_role = ktc.get("kv_role", args.kv_role)
        if ktc.get("kv_rank") is not None:
            args.kv_rank = int(ktc["kv_rank"])
        if ktc.get("kv_parallel_size"):
            args.kv_world = int(ktc["kv_parallel_size"])

    from production_stack_amd.engine.config import ParallelConfig

    cfg = EngineConfig(
        model=args.model,
        max_model_len=args.max_model_len,
        seed=args.seed,
        weights_path=args.weights_path,
        tokenizer=args.tokenizer,
        cache=CacheConfig(
            num_gpu_blocks=args.num_gpu_blocks,
            gpu_memory_utilization=args.gpu_memory_utilization,
            enable_prefix_caching=args.enable_prefix_caching,
            cpu_offload_gb=args.cpu_offload_gb,
            offload_dtype=args.offload_dtype,
            remote_kv_url=args.remote_kv_url,
            kv_cache_dtype=args.kv_cache_dtype,
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=args.max_num_seqs,
            max_num_batched_tokens=args.max_num_batched_tokens,
            enable_chunked_prefill=args.enable_chunked_prefill,
            num_speculative_tokens=args.num_speculative_tokens,
        ),
        async_scheduling=args.async_scheduling,
        unified_mixed_steps=args.unified_mixed_steps,
        speculative_model=args.speculative_model,
        speculative_weights_path=args.speculative_weights_path,
        quantization=args.quantization,
        enable_lora=args.enable_lora,
        max_loras=args.max_loras,
        max_lora_rank=args.max_lora_rank,
        parallel=ParallelConfig(
            tensor_parallel_size=args.tensor_parallel_size,
            pipeline_parallel_size=args.pipeline_parallel_size,
            kv_role=args.kv_role,
        ),
    )
    cfg.kv_controller_url = args.kv_controller_url  # type: ignore[attr-defined]
    if args.kv_rank is not None and args.kv_world > 1:
        cfg.kv_transfer = {  # type: ignore[attr-defined]
            "rank": args.kv_rank,
            "world": args.kv_world,
            "master_port": args.kv_master_port,
            "side_port": args.kv_side_port,
        }
    cfg.advertise_url = (  # type: ignore[attr-defined]
        args.advertise_url or f"http://127.0.0.1:{args.port}"
    )
    engine = LLMEngine(cfg, device=args.device)
    if engine.is_tp_worker:
        logging.basicConfig(level=logging.INFO)
        engine.run_tp_worker()
        return
    if engine.is_pp_worker:
        # pipeline ranks > 0 serve activations, not HTTP
        logging.basicConfig(level=logging.INFO)
        engine.run_pp_worker()
        return
    smn = args.served_model_name
    if isinstance(smn, list):
        served = smn[0] if smn else args.model
        extra_names = smn[1:]
    else:
        served = smn or args.model
        extra_names = []
    tpl_src = None
    if args.chat_template:
        with open(args.chat_template) as f:
            tpl_src = f.read()
    app = build_server(engine, served, chat_template=tpl_src,
                       extra_model_names=extra_names)
    logging.basicConfig(level=logging.INFO)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
