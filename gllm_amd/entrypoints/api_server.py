"""OpenAI-compatible API server (reference: entrypoints/api_server.py).

Routes: /v1/chat/completions, /v1/completions, /v1/models, /health,
/version, /server_info, /start_profile, /stop_profile.

    python -m gllm_amd.entrypoints.api_server --model <hf_dir> \
        --pp-size 4 --schedule-method token_throttling
"""

import argparse
import asyncio
import json
from typing import Optional

from gllm_amd import __version__
from gllm_amd.config import EngineConfig
from gllm_amd.engine.server_engine import AsyncLLMEngine
from gllm_amd.entrypoints.protocol import (
    ChatCompletionRequest, ChatCompletionResponse,
    ChatCompletionResponseChoice, ChatCompletionStreamChoice,
    ChatCompletionStreamResponse, ChatMessage, CompletionRequest,
    CompletionResponse, CompletionResponseChoice, DeltaMessage, ModelCard,
    ModelList, UsageInfo)
from gllm_amd.logger import logger
from gllm_amd.sequence import SamplingParams

engine: Optional[AsyncLLMEngine] = None
served_model = ""


def build_app():
    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse, StreamingResponse

    app = FastAPI(title="gllm_amd")

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/version")
    async def version():
        return {"version": __version__}

    @app.get("/server_info")
    async def server_info():
        c = engine.config
        return {"model": c.model, "pp_size": c.pp_size, "tp_size": c.tp_size,
                "dp_size": c.dp_size, "schedule_method": c.schedule_method,
                "page_size": c.page_size}

    @app.get("/stats")
    async def stats():
        return {"engine": engine.latest_stats,
                "requests_total": engine.request_counter,
                "tokens_total": engine.token_counter,
                "active_requests": len(engine.requests)}

    @app.get("/metrics")
    async def metrics():
        from fastapi.responses import PlainTextResponse
        st = engine.latest_stats or {}
        lines = [
            "# TYPE gllm_requests_total counter",
            f"gllm_requests_total {engine.request_counter}",
            "# TYPE gllm_generated_tokens_total counter",
            f"gllm_generated_tokens_total {engine.token_counter}",
            "# TYPE gllm_active_requests gauge",
            f"gllm_active_requests {len(engine.requests)}",
        ]
        for k, v in st.items():
            lines.append(f"# TYPE gllm_{k} gauge")
            lines.append(f"gllm_{k} {v}")
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/v1/models")
    async def models():
        return ModelList(data=[ModelCard(id=served_model)]).model_dump()

    @app.post("/start_profile")
    async def start_profile():
        engine.send_command("start_profile")
        return {"status": "started"}

    @app.post("/stop_profile")
    async def stop_profile():
        engine.send_command("stop_profile")
        return {"status": "stopped"}

    def _sampling_from(req, default_max: int = 512) -> SamplingParams:
        stops = req.stop
        if isinstance(stops, str):
            stops = [stops]
        max_tokens = getattr(req, "max_completion_tokens", None) or \
            req.max_tokens or default_max
        rp = req.repetition_penalty
        if rp is None:
            rp = 1.0
        lp = req.logprobs
        if isinstance(lp, bool):  # chat: bool + top_logprobs count
            lp = (getattr(req, "top_logprobs", None) or 1) if lp else None
        vocab = getattr(engine.hf_config, "vocab_size", None)
        bias = None
        if getattr(req, "logit_bias", None):
            bias = {int(k): float(v) for k, v in req.logit_bias.items()}
            if vocab is not None:
                bad = [k for k in bias if not 0 <= k < vocab]
                if bad:
                    raise ValueError(
                        f"logit_bias token id(s) {bad[:5]} out of range "
                        f"[0, {vocab})")
        allowed = getattr(req, "allowed_token_ids", None)
        if allowed and vocab is not None:
            bad = [i for i in allowed if not 0 <= i < vocab]
            if bad:
                raise ValueError(
                    f"allowed_token_ids {bad[:5]} out of range [0, {vocab})")
        bad_words_ids = None
        if getattr(req, "bad_words", None):
            tok = engine.tokenizer
            if tok is None:
                raise ValueError("bad_words needs a tokenizer")
            bad_words_ids = []
            for w in req.bad_words:
                # ban both the word-initial and the mid-text piece
                # sequences (tokenizers split " word" and "word"
                # differently)
                for v in {w, " " + w}:
                    ids = tok.encode(v, add_special_tokens=False)
                    if ids:
                        bad_words_ids.append(list(ids))
        return SamplingParams(
            temperature=req.temperature if req.temperature is not None
            else 1.0,
            top_p=req.top_p if req.top_p is not None else 1.0,
            top_k=req.top_k if req.top_k is not None else -1,
            min_p=getattr(req, "min_p", None) or 0.0,
            repetition_penalty=rp,
            presence_penalty=getattr(req, "presence_penalty", 0.0) or 0.0,
            frequency_penalty=getattr(req, "frequency_penalty", 0.0)
            or 0.0,
            max_tokens=max_tokens,
            min_tokens=req.min_tokens or 0,
            ignore_eos=req.ignore_eos,
            stop=stops, stop_token_ids=req.stop_token_ids,
            include_stop_str_in_output=getattr(
                req, "include_stop_str_in_output", False),
            seed=req.seed,
            logprobs=lp,
            prompt_logprobs=getattr(req, "prompt_logprobs", None),
            logit_bias=bias,
            allowed_token_ids=allowed,
            bad_words_token_ids=bad_words_ids,
            skip_special_tokens=getattr(req, "skip_special_tokens", True))

    def _truncate(req, token_ids):
        """OpenAI truncate_prompt_tokens: keep only the LAST t prompt
        tokens (skipped for multimodal prompts — cutting an image span
        would desync the embedding merge)."""
        t = getattr(req, "truncate_prompt_tokens", None)
        return token_ids[-t:] if t else token_ids

    def _vary_seed(sampling: SamplingParams, j: int) -> SamplingParams:
        """Choice j of an n>1 request: distinct seed per choice when one
        was given, shared params otherwise."""
        import dataclasses as _dc
        if j == 0:
            return sampling
        return _dc.replace(sampling,
                           seed=None if sampling.seed is None
                           else sampling.seed + j)

    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest, raw: Request):
        messages = [m.model_dump(exclude_none=True) for m in req.messages]
        kwargs = req.chat_template_kwargs or {}
        kwargs.setdefault("add_generation_prompt",
                          req.add_generation_prompt)
        if req.continue_final_message:
            kwargs["continue_final_message"] = True
        if req.tools:
            kwargs["tools"] = [t.model_dump() for t in req.tools]
        try:
            messages, images = engine.extract_images(messages)
            token_ids = engine.apply_chat_template(messages, **kwargs)
            # preprocessing + (disagg) remote encode off the event loop
            import asyncio as _aio
            token_ids, mm = await _aio.to_thread(
                engine.process_images, token_ids, images)
            if not mm:
                token_ids = _truncate(req, token_ids)
        except Exception as e:
            return JSONResponse(status_code=400,
                                content={"error": str(e)})
        try:
            sampling = _sampling_from(req)
        except ValueError as e:
            return JSONResponse(status_code=400,
                                content={"error": str(e)})
        if req.stream:
            return StreamingResponse(
                _chat_stream(req, raw, token_ids, sampling, mm=mm),
                media_type="text/event-stream")
        n = req.n or 1
        import asyncio as _aio
        results = await _aio.gather(*[
            _collect(raw, token_ids, _vary_seed(sampling, j), mm=mm)
            for j in range(n)])
        choices = []
        total_out = 0
        for j, (text, finish, n_out, _plp, lps) in enumerate(results):
            total_out += n_out
            message = ChatMessage(role="assistant", content=text)
            if req.tools:
                from gllm_amd.tokenizers.deepseek_v32 import \
                    load_dsv32_encoder
                from gllm_amd.tokenizers.tool_parsers import \
                    parse_tool_calls
                parsed_text, calls = parse_tool_calls(
                    text, served_model,
                    tools=[t.model_dump() for t in req.tools],
                    encoder=load_dsv32_encoder(engine.config.model))
                if calls:
                    message = ChatMessage(role="assistant",
                                          content=parsed_text or None,
                                          tool_calls=calls)
                    finish = "tool_calls"
            choices.append(ChatCompletionResponseChoice(
                index=j, message=message, finish_reason=finish,
                logprobs={"content": lps} if lps else None))
        resp = ChatCompletionResponse(
            model=req.model or served_model,
            choices=choices,
            usage=UsageInfo(prompt_tokens=len(token_ids),
                            completion_tokens=total_out,
                            total_tokens=len(token_ids) + total_out))
        return resp.model_dump()

    def _lp_entry(chunk):
        chosen, topk = chunk.logprob
        return {"token": chunk.text, "logprob": chosen,
                "top_logprobs": [{"token_id": int(t), "logprob": v}
                                 for t, v in topk.items()]}

    async def _collect(raw, token_ids, sampling, mm=None):
        text_parts = []
        finish = None
        n_out = 0
        plp = None
        lps = [] if sampling.logprobs else None
        async for chunk in engine.generate_stream(token_ids, sampling,
                                                  mm=mm):
            if await raw.is_disconnected():
                break
            text_parts.append(chunk.text)
            finish = chunk.finish_reason or finish
            n_out = chunk.n_output_tokens
            plp = chunk.prompt_logprobs or plp
            if lps is not None and chunk.logprob is not None:
                lps.append(_lp_entry(chunk))
        text = "".join(text_parts)
        from gllm_amd.engine.detokenizer import check_stop_strings
        _, text = check_stop_strings(
            text, sampling.stop,
            include_stop=sampling.include_stop_str_in_output)
        return text, finish, n_out, plp, lps

    async def _merged_stream(raw, token_ids, sampling, n, mm=None):
        """Run n generations concurrently, yield (choice_idx, chunk) in
        arrival order (n>1 streaming: interleaved choices)."""
        import asyncio as _aio
        if n <= 1:
            async for chunk in engine.generate_stream(token_ids, sampling,
                                                      mm=mm):
                if await raw.is_disconnected():
                    return
                yield 0, chunk
            return
        q: _aio.Queue = _aio.Queue()

        async def pump(j):
            async for chunk in engine.generate_stream(
                    token_ids, _vary_seed(sampling, j), mm=mm):
                await q.put((j, chunk))
            await q.put((j, None))

        tasks = [_aio.ensure_future(pump(j)) for j in range(n)]
        live = n
        try:
            while live:
                j, chunk = await q.get()
                if chunk is None:
                    live -= 1
                    continue
                if await raw.is_disconnected():
                    return
                yield j, chunk
        finally:
            for t in tasks:
                t.cancel()

    async def _chat_stream(req, raw, token_ids, sampling, mm=None):
        resp_id = None
        first_for = set()
        n = req.n or 1
        n_out = 0
        # streaming tool-call parse: content before the first marker
        # streams as it arrives; each completed call becomes one
        # tool_calls delta (reference stream_parser semantics)
        stream_parsers = {}
        cum_text = {}
        if req.tools:
            from gllm_amd.tokenizers.deepseek_v32 import load_dsv32_encoder
            from gllm_amd.tokenizers.tool_parsers import get_tool_parser
            tool_schemas = [t.model_dump() for t in req.tools]
            enc = load_dsv32_encoder(engine.config.model)
            for j in range(n):
                stream_parsers[j] = get_tool_parser(
                    served_model, encoder=enc).stream(tool_schemas)
                cum_text[j] = ""

        def _mk_out(j, delta, finish, chunk=None):
            return ChatCompletionStreamResponse(
                model=req.model or served_model,
                choices=[ChatCompletionStreamChoice(
                    index=j, delta=delta, finish_reason=finish,
                    logprobs={"content": [_lp_entry(chunk)]}
                    if chunk is not None and chunk.logprob is not None
                    else None)])

        async for j, chunk in _merged_stream(raw, token_ids, sampling, n,
                                             mm=mm):
            n_out += 1 if chunk.token_id >= 0 else 0
            deltas = []
            if j in stream_parsers:
                cum_text[j] += chunk.text or ""
                for d in stream_parsers[j].feed(cum_text[j]):
                    if "content" in d:
                        deltas.append(DeltaMessage(content=d["content"]))
                    else:
                        deltas.append(DeltaMessage(
                            tool_calls=[d["tool_call"]]))
                finish = chunk.finish_reason
                if finish and stream_parsers[j].emitted_tool_calls:
                    finish = "tool_calls"
                if not deltas and finish is None:
                    continue
                if not deltas:
                    deltas = [DeltaMessage()]
            else:
                deltas = [DeltaMessage(content=chunk.text)]
                finish = chunk.finish_reason
            if j not in first_for:
                deltas[0].role = "assistant"
                first_for.add(j)
            for i, delta in enumerate(deltas):
                out = _mk_out(j, delta,
                              finish if i == len(deltas) - 1 else None,
                              chunk if not stream_parsers else None)
                if resp_id is None:
                    resp_id = out.id
                else:
                    out.id = resp_id
                yield f"data: {out.model_dump_json(exclude_none=True)}\n\n"
        if req.stream_options and req.stream_options.get("include_usage"):
            out = ChatCompletionStreamResponse(
                id=resp_id or "", model=req.model or served_model,
                choices=[],
                usage=UsageInfo(prompt_tokens=len(token_ids),
                                completion_tokens=n_out,
                                total_tokens=len(token_ids) + n_out))
            yield f"data: {out.model_dump_json(exclude_none=True)}\n\n"
        yield "data: [DONE]\n\n"

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        prompts = req.prompt
        if isinstance(prompts, str):
            prompts = [prompts]
        elif prompts and isinstance(prompts[0], int):
            prompts = [prompts]
        try:
            sampling = _sampling_from(req, default_max=16)
        except ValueError as e:
            return JSONResponse(status_code=400,
                                content={"error": str(e)})
        if req.stream:
            token_ids = prompts[0] if isinstance(prompts[0], list) \
                else engine.encode(prompts[0])
            token_ids = _truncate(req, token_ids)
            return StreamingResponse(
                _completion_stream(req, raw, token_ids, sampling),
                media_type="text/event-stream")
        n = req.n or 1
        choices = []
        total_p = total_c = 0
        for i, p in enumerate(prompts):
            token_ids = p if isinstance(p, list) else engine.encode(p)
            token_ids = _truncate(req, token_ids)
            import asyncio as _aio
            results = await _aio.gather(*[
                _collect(raw, token_ids, _vary_seed(sampling, j))
                for j in range(n)])
            for j, (text, finish, n_out, plp, lps) in enumerate(results):
                if req.echo and not isinstance(p, list):
                    text = p + text
                choices.append(CompletionResponseChoice(
                    index=i * n + j, text=text, finish_reason=finish,
                    logprobs={"content": lps} if lps else None,
                    prompt_logprobs=plp if req.prompt_logprobs else None))
                total_c += n_out
            total_p += len(token_ids)
        resp = CompletionResponse(
            model=req.model or served_model, choices=choices,
            usage=UsageInfo(prompt_tokens=total_p, completion_tokens=total_c,
                            total_tokens=total_p + total_c))
        return resp.model_dump()

    async def _completion_stream(req, raw, token_ids, sampling):
        n = req.n or 1
        total_c = 0
        async for j, chunk in _merged_stream(raw, token_ids, sampling, n):
            if chunk.finish_reason is not None:
                total_c += chunk.n_output_tokens
            resp = CompletionResponse(
                model=req.model or served_model,
                choices=[CompletionResponseChoice(
                    index=j, text=chunk.text,
                    finish_reason=chunk.finish_reason)])
            yield f"data: {resp.model_dump_json(exclude_none=True)}\n\n"
        opts = getattr(req, "stream_options", None) or {}
        if opts.get("include_usage"):
            usage = CompletionResponse(
                model=req.model or served_model, choices=[],
                usage=UsageInfo(
                    prompt_tokens=len(token_ids),
                    completion_tokens=total_c,
                    total_tokens=len(token_ids) + total_c))
            yield f"data: {usage.model_dump_json(exclude_none=True)}\n\n"
        yield "data: [DONE]\n\n"

    return app


def make_arg_parser():
    p = argparse.ArgumentParser(description="gllm_amd OpenAI API server")
    p.add_argument("--model", type=str, required=True)
    p.add_argument("--served-model-name", type=str, default=None)
    p.add_argument("--host", type=str, default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--zmq-port", type=int, default=28700)
    p.add_argument("--pp-size", "--pp", type=int, default=1)
    p.add_argument("--tp-size", "--tp", type=int, default=1)
    p.add_argument("--dp-size", "--dp", type=int, default=1)
    p.add_argument("--use-ep", action="store_true")
    p.add_argument("--load-format", choices=["auto", "dummy"],
                   default="auto")
    p.add_argument("--dtype", type=str, default="bfloat16")
    p.add_argument("--schedule-method", type=str,
                   choices=["split_pd", "chunked_prefill",
                            "token_throttling"],
                   default="token_throttling")
    p.add_argument("--maxp", type=int, default=8192)
    p.add_argument("--maxd", type=int, default=1024)
    p.add_argument("--minp", type=int, default=512)
    p.add_argument("--iterp", type=int, default=16)
    p.add_argument("--page-size", type=int, default=16)
    p.add_argument("--gpu-memory-util", type=float, default=0.9)
    p.add_argument("--model-max-length", type=int, default=None)
    p.add_argument("--disable-prefix-caching", action="store_true")
    p.add_argument("--enforce-eager", action="store_true")
    p.add_argument("--max-graph-bs", type=int, default=256)
    p.add_argument("--assigned-layers", type=str, default=None)
    p.add_argument("--master-addr", type=str, default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=29500)
    # multi-node: master hosts the frontend + --worker-ranks; each slave
    # node hosts its own --worker-ranks and bridges to the master's
    # control-plane relay (engine/multinode.py)
    p.add_argument("--launch-mode", choices=["normal", "master", "slave"],
                   default="normal")
    p.add_argument("--worker-ranks", type=str, default=None,
                   help="comma-separated global ranks on this node, "
                        "e.g. '0,1' (master/slave modes)")
    p.add_argument("--relay-port", type=int, default=None,
                   help="control-plane TCP port (default master_port+1)")
    # encoder disaggregation (disagg/)
    p.add_argument("--mm-encoder-addr", type=str, default=None,
                   help="remote vision-encoder 'host:port'")
    p.add_argument("--discovery-addr", type=str, default=None,
                   help="discovery server 'host:port' to resolve an "
                        "encoder from")
    p.add_argument("--skip-visual", action="store_true", default=None,
                   help="LM node of an encoder-disagg deployment: do "
                        "not load the vision tower (default: on when "
                        "an encoder addr/discovery is configured)")
    p.add_argument("--mla-mode", choices=["absorbed", "decompressed"],
                   default="absorbed",
                   help="MLA execution form (reference --mla-backend): "
                        "absorbed = 576-dim latent MQA cache, "
                        "decompressed = per-head K/V")
    p.add_argument("--seed", type=int, default=0)
    return p


def config_from_args(args) -> EngineConfig:
    return EngineConfig(
        model=args.model, load_format=args.load_format, dtype=args.dtype,
        pp_size=args.pp_size, tp_size=args.tp_size, dp_size=args.dp_size,
        use_ep=args.use_ep, schedule_method=args.schedule_method,
        maxp=args.maxp, maxd=args.maxd, minp=args.minp, iterp=args.iterp,
        page_size=args.page_size, gpu_memory_util=args.gpu_memory_util,
        model_max_length=args.model_max_length,
        enable_prefix_caching=not args.disable_prefix_caching,
        enforce_eager=args.enforce_eager, max_graph_bs=args.max_graph_bs,
        assigned_layers=args.assigned_layers,
        master_addr=args.master_addr, master_port=args.master_port,
        launch_mode=args.launch_mode,
        worker_ranks=[int(r) for r in args.worker_ranks.split(",")]
        if args.worker_ranks else None,
        relay_port=args.relay_port,
        mm_encoder_addr=args.mm_encoder_addr,
        skip_visual=(args.skip_visual if args.skip_visual is not None
                     else bool(args.mm_encoder_addr
                               or args.discovery_addr)),
        discovery_addr=args.discovery_addr, seed=args.seed,
        mla_mode=args.mla_mode,
        device="cuda" if _has_gpu() else "cpu")


def _has_gpu() -> bool:
    import torch
    return torch.cuda.is_available()


def main():
    global engine, served_model
    args = make_arg_parser().parse_args()
    served_model = args.served_model_name or args.model
    config = config_from_args(args)
    if config.launch_mode == "slave":
        # no HTTP frontend on slave nodes: run workers + control bridge
        from gllm_amd.engine.multinode import run_slave_node
        run_slave_node(config)
        return
    engine = AsyncLLMEngine(config, base_port=args.zmq_port)
    engine.start()
    app = build_app()
    import uvicorn
    try:
        uvicorn.run(app, host=args.host, port=args.port, log_level="info")
    finally:
        engine.stop()


if __name__ == "__main__":
    main()
