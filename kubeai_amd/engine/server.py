"""OpenAI-compatible HTTP server wrapping LLMEngine — the per-Model engine
container contract the control plane depends on:

  POST /v1/completions, /v1/chat/completions (stream + non-stream)
  POST /v1/embeddings
  GET  /health                      (startup/readiness/liveness probes)
  GET  /metrics                     (Prometheus; queue depth + KV occupancy
                                     gauges consumed by the autoscaler)
  GET  /v1/models
  POST /v1/load_lora_adapter        ("already loaded" error semantics)
  POST /v1/unload_lora_adapter      ("cannot be found" error semantics)

Run:  python -m kubeai_amd.engine.server --model llama-tiny --port 8000
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import queue
import re
import threading
import time
import uuid
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse
import prometheus_client as prom

from .engine import EngineConfig, LLMEngine
from .scheduler import SamplingParams
from .tokenizer import StreamDecoder, apply_chat_template, load_tokenizer


class _StopScanner:
    """Earliest stop-string hit with O(new text) scanning per step."""

    def __init__(self, stops: list[str]):
        self.stops = stops
        self.max_len = max(len(s) for s in stops)
        self._from = 0

    def scan(self, text: str) -> int:
        start = max(0, self._from - self.max_len + 1)
        best = -1
        for s in self.stops:
            i = text.find(s, start)
            if i >= 0 and (best < 0 or i < best):
                best = i
        self._from = len(text)
        return best


def _holdback_len(text: str, stops: list[str]) -> int:
    """Longest trailing substring of `text` that is a proper prefix of a
    stop string — streamed deltas hold it back so a stop spanning chunk
    boundaries never leaks its leading characters to the client."""
    m = 0
    for s in stops:
        k = min(len(s) - 1, len(text))
        while k > m:
            if text.endswith(s[:k]):
                m = k
                break
            k -= 1
    return m

REGISTRY = prom.CollectorRegistry()
M_WAITING = prom.Gauge("kubeai_engine_num_requests_waiting", "queue depth", ["model"], registry=REGISTRY)
M_RUNNING = prom.Gauge("kubeai_engine_num_requests_running", "running requests", ["model"], registry=REGISTRY)
M_KV = prom.Gauge("kubeai_engine_kv_cache_usage_perc", "KV cache occupancy", ["model"], registry=REGISTRY)
M_HIT = prom.Gauge("kubeai_engine_prefix_cache_hit_rate", "prefix cache hit rate", ["model"], registry=REGISTRY)
M_GEN = prom.Counter("kubeai_engine_generation_tokens_total", "generated tokens", ["model"], registry=REGISTRY)
M_PROMPT = prom.Counter("kubeai_engine_prompt_tokens_total", "prompt tokens", ["model"], registry=REGISTRY)
M_TTFT = prom.Histogram(
    "kubeai_engine_time_to_first_token_seconds", "TTFT", ["model"], registry=REGISTRY,
    buckets=(0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0, 30.0),
)


class EngineServer:
    """Engine + stepping thread + async request plumbing.

    Tensor-parallel serving (--tensor-parallel-size N): rank 0 owns the
    HTTP surface and broadcasts each iteration's (new requests, aborts)
    to N-1 worker processes over torch.distributed; every rank then runs
    the identical scheduler step in lockstep (all-reduced activations are
    rank-identical, so sampling and scheduling never diverge — the same
    invariant the TP bench path relies on, kubeai_amd/parallel/tp.py).
    """

    def __init__(self, cfg: EngineConfig, served_model_name: str, tp_size: int = 1,
                 tp_port: int = 0, task: str = "generate"):
        self.cfg = cfg
        self.served_model_name = served_model_name
        self.tp_size = tp_size
        self.tp_port = tp_port
        self.task = task
        self._dist = None
        self.engine: Optional[LLMEngine] = None
        self.stt = None  # SpeechToText model when task == "transcribe"
        self.embedder = None  # BertEncoder when task == "embed"
        self.tokenizer = None
        self._submit: "queue.Queue" = queue.Queue()
        self._events: dict[str, tuple[asyncio.AbstractEventLoop, asyncio.Queue]] = {}
        self._aborts: "queue.Queue" = queue.Queue()
        # embed jobs run ON the stepping thread (they share the
        # BlockManager/KV tensors with in-flight generations — ADVICE r1)
        self._embed_queue: "queue.Queue" = queue.Queue()
        self._ready = threading.Event()
        self._stop = threading.Event()
        self._lora_adapters: dict[str, str] = {}
        self._lock = threading.Lock()
        # step-loop health (a dead stepping thread must not keep
        # reporting healthy while every request hangs)
        self._consec_step_errors = 0
        self._fatal_error: Optional[str] = None
        self._thread = threading.Thread(target=self._run, daemon=True)

    # ------------------------------------------------------------- lifecycle
    def start(self) -> None:
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()

    def _run(self) -> None:
        try:
            self._run_inner()
        except Exception:
            import traceback

            traceback.print_exc()
            self._fatal_error = traceback.format_exc(limit=8)

    def _fail_inflight(self) -> list:
        """After a step exception: abort every in-flight request and emit
        finished-abort outputs so no client waits forever. The next loop
        iteration's schedule() applies the abort marks and frees KV."""
        from .engine import RequestOutput

        eng = self.engine
        outs = []
        for r in list(eng.scheduler.running) + list(eng.scheduler.waiting):
            eng.abort_request(r.request_id)
            outs.append(
                RequestOutput(
                    request_id=r.request_id,
                    new_token_ids=[],
                    finished=True,
                    finish_reason="abort",
                    num_prompt_tokens=r.num_prompt_tokens,
                    num_cached_tokens=0,
                    output_token_ids=r.output_token_ids,
                )
            )
        return outs

    def _run_inner(self) -> None:
        # heavy init inside the thread so /health can answer "starting"
        if self.task == "transcribe":
            from kubeai_amd.models.whisper import PRESETS as STT_PRESETS
            from kubeai_amd.models.whisper import (
                SpeechConfig,
                SpeechToText,
                load_weights_whisper,
            )

            ckpt_dir = None
            if os.path.isdir(self.cfg.model) and os.path.exists(
                os.path.join(self.cfg.model, "config.json")
            ):
                import json as _json

                with open(os.path.join(self.cfg.model, "config.json")) as f:
                    hc = _json.load(f)
                scfg = SpeechConfig(
                    n_mels=hc.get("num_mel_bins", 80),
                    n_audio_ctx=hc.get("max_source_positions", 1500),
                    n_text_ctx=hc.get("max_target_positions", 448),
                    n_state=hc.get("d_model", 128),
                    n_head=hc.get("encoder_attention_heads", 4),
                    n_audio_layer=hc.get("encoder_layers", 2),
                    n_text_layer=hc.get("decoder_layers", 2),
                    vocab_size=hc.get("vocab_size", 2048),
                    sot_token=hc.get("decoder_start_token_id", 1),
                    eot_token=hc.get("eos_token_id", 2),
                )
                ckpt_dir = self.cfg.model
            else:
                scfg = STT_PRESETS.get(self.cfg.model, STT_PRESETS["whisper-tiny"])
            self.stt = SpeechToText(
                scfg, device=self.cfg.resolve_device(), seed=self.cfg.seed
            )
            if ckpt_dir is not None:
                load_weights_whisper(self.stt, ckpt_dir)
            self.tokenizer = load_tokenizer(
                self.cfg.model, scfg.vocab_size, scfg.sot_token, scfg.eot_token
            )
            self._ready.set()
            self._stop.wait()  # transcription is request-driven, no stepping
            return
        if self.task == "embed":
            from kubeai_amd.models import bert as bert_mod

            if os.path.isdir(self.cfg.model) and os.path.exists(
                os.path.join(self.cfg.model, "config.json")
            ):
                bcfg = bert_mod.config_from_hf(self.cfg.model)
                self.embedder = bert_mod.BertEncoder(
                    bcfg, device=self.cfg.resolve_device(), seed=self.cfg.seed
                )
                bert_mod.load_weights_bert(self.embedder, self.cfg.model)
            else:
                bcfg = bert_mod.PRESETS.get(
                    self.cfg.model, bert_mod.PRESETS["bert-tiny"]
                )
                self.embedder = bert_mod.BertEncoder(
                    bcfg, device=self.cfg.resolve_device(), seed=self.cfg.seed
                )
            self.tokenizer = load_tokenizer(self.cfg.model, bcfg.vocab_size, 1, 2)
            self._ready.set()
            self._stop.wait()  # encoding is request-driven, no stepping
            return
        tp_group = None
        if self.tp_size > 1 or os.environ.get("KUBEAI_FORCE_TP"):
            import torch.distributed as dist

            from kubeai_amd.parallel.tp import TPGroup

            dist.init_process_group(
                "nccl" if self.cfg.resolve_device().startswith("cuda") else "gloo",
                init_method=f"tcp://127.0.0.1:{self.tp_port}",
                rank=0,
                world_size=self.tp_size,
            )
            self._dist = dist
            tp_group = TPGroup()
        self.engine = LLMEngine(self.cfg, tp_group=tp_group)
        arch = self.engine.arch
        self.tokenizer = load_tokenizer(
            self.cfg.model, arch.vocab_size, arch.bos_token_id, arch.eos_token_id
        )
        if arch.vision is not None and hasattr(self.tokenizer, "image_token_id"):
            self.tokenizer.image_token_id = arch.image_token_id
        # JSON-mode constrained decoding needs the tokenizer at sampling
        self.engine.runner.tokenizer = self.tokenizer
        self._ready.set()
        label = self.served_model_name
        while not self._stop.is_set():
            worked = False
            new_reqs, abort_ids = [], []
            while True:
                try:
                    item = self._submit.get_nowait()
                except queue.Empty:
                    break
                new_reqs.append(item)
            while True:
                try:
                    abort_ids.append(self._aborts.get_nowait())
                except queue.Empty:
                    break
            embed_jobs = []
            while True:
                try:
                    embed_jobs.append(self._embed_queue.get_nowait())
                except queue.Empty:
                    break
            if self._dist is not None:
                # keep TP workers in lockstep: ship this iteration's intents
                self._dist.broadcast_object_list(
                    [{"new": new_reqs, "aborts": abort_ids,
                      "embeds": [j[0] for j in embed_jobs],
                      "shutdown": False}],
                    src=0,
                )
            for rid, toks, params, lora_id, images in new_reqs:
                try:
                    self.engine.add_request(
                        toks, params, request_id=rid, lora_id=lora_id,
                        images=images or None,
                    )
                except RequestError as e:
                    # surface a bad multimodal request as a failed output
                    ent = self._events.get(rid)
                    if ent is not None:
                        loop, q = ent
                        loop.call_soon_threadsafe(q.put_nowait, e)
                    continue
                M_PROMPT.labels(label).inc(len(toks))
                worked = True
            for rid in abort_ids:
                self.engine.abort_request(rid)
            for tok_lists, loop, fut in embed_jobs:
                # embed shares the BlockManager/KV with generations, so it
                # must run here on the stepping thread
                try:
                    vecs = self.engine.embed(tok_lists)
                    loop.call_soon_threadsafe(
                        lambda f=fut, v=vecs: f.done() or f.set_result(v)
                    )
                except Exception as e:  # noqa: BLE001
                    loop.call_soon_threadsafe(
                        lambda f=fut, err=e: f.done() or f.set_exception(err)
                    )
                worked = True
            if self.engine.has_work():
                try:
                    outputs = self.engine.step()
                    self._consec_step_errors = 0
                except Exception:
                    import traceback

                    traceback.print_exc()
                    self._consec_step_errors += 1
                    if self._consec_step_errors >= 3:
                        self._fatal_error = traceback.format_exc(limit=8)
                    outputs = self._fail_inflight()
                worked = True
                for o in outputs:
                    M_GEN.labels(label).inc(len(o.new_token_ids))
                    ent = self._events.get(o.request_id)
                    if ent is not None:
                        loop, q = ent
                        loop.call_soon_threadsafe(q.put_nowait, o)
            s = self.engine.stats()
            M_WAITING.labels(label).set(s["num_waiting"])
            M_RUNNING.labels(label).set(s["num_running"])
            M_KV.labels(label).set(s["kv_usage"])
            M_HIT.labels(label).set(s["prefix_cache_hit_rate"])
            if not worked:
                time.sleep(0.002)
        if self._dist is not None:
            self._dist.broadcast_object_list(
                [{"new": [], "aborts": [], "shutdown": True}], src=0
            )
            self._dist.destroy_process_group()

    # ------------------------------------------------------------- requests
    async def generate(self, token_ids: list[int], params: SamplingParams,
                       lora_id: int = 0, images: list | None = None):
        """Async iterator of RequestOutput for one request."""
        rid = f"cmpl-{uuid.uuid4().hex[:16]}"
        loop = asyncio.get_running_loop()
        q: asyncio.Queue = asyncio.Queue()
        self._events[rid] = (loop, q)
        t0 = time.monotonic()
        first = True
        self._submit.put((rid, token_ids, params, lora_id, images or []))
        try:
            while True:
                o = await q.get()
                if isinstance(o, Exception):
                    raise o  # RequestError -> HTTP 400 via the app handler
                if first:
                    M_TTFT.labels(self.served_model_name).observe(time.monotonic() - t0)
                    first = False
                yield o
                if o.finished:
                    return
        finally:
            self._events.pop(rid, None)
            self._aborts.put(rid)

    async def embed(self, tok_lists: list[list[int]]) -> list[list[float]]:
        """Embed via the stepping thread (generate-task models)."""
        loop = asyncio.get_running_loop()
        fut = loop.create_future()
        self._embed_queue.put((tok_lists, loop, fut))
        return await fut

    # ------------------------------------------------------------- LoRA
    # Registry + per-adapter KV-cache salting; adapter weights are applied
    # by the runner's LoRA manager (kubeai_amd/engine/lora.py).
    def load_lora(self, name: str, path: Optional[str]) -> None:
        if self.tp_size > 1:
            # TP layers do not apply LoRA and adapter loads are not
            # broadcast to worker ranks — honest rejection beats silently
            # serving base-model outputs under an adapter name
            raise ValueError(
                "LoRA adapters are not supported with tensor-parallel "
                "serving (tensor_parallel_size > 1)"
            )
        if path and not (os.path.isdir(path) or os.path.isfile(path)):
            raise ValueError(f"lora path {path} does not exist")
        lid = self.lora_id_of(name)
        if self.engine is not None:
            self.engine.load_lora(lid, path)

    def unload_lora(self, name: str) -> None:
        if self.engine is not None:
            self.engine.unload_lora(self.lora_id_of(name))

    def lora_id_of(self, name: str) -> int:
        import zlib

        return 1 + (zlib.crc32(name.encode()) % 1_000_000)


from .engine import RequestError  # noqa: E402  (shared with the engine core)


def _images_from_messages(server, messages: list) -> list:
    """Decode + preprocess image_url content parts, in reading order.

    Returns CLIP-normalized pixel tensors sized for the model's vision
    tower. Decoding problems and images sent to text-only models raise
    RequestError -> HTTP 400 (the reference forwards parts to vLLM and
    relays its 400s; here the engine is in-house).
    """
    from kubeai_amd.utils import imaging

    urls: list[str] = []
    for m in messages:
        content = m.get("content")
        if not isinstance(content, list):
            continue
        for p in content:
            if isinstance(p, dict) and p.get("type") == "image_url":
                iu = p.get("image_url")
                url = iu.get("url", "") if isinstance(iu, dict) else str(iu or "")
                urls.append(url)
    if not urls:
        return []
    arch = getattr(getattr(server, "engine", None), "arch", None)
    if arch is None or arch.vision is None:
        raise RequestError(
            f"model {server.served_model_name!r} does not accept image input"
        )
    size = int(arch.vision.get("image_size", 336))
    out = []
    for url in urls:
        try:
            img, _raw = imaging.image_from_url(url)
            out.append(imaging.preprocess(img, size))
        except imaging.ImageError as e:
            raise RequestError(f"bad image: {e}") from None
    return out


def build_app(server: EngineServer) -> FastAPI:
    app = FastAPI()
    app.state.eng_server = server  # test/introspection handle
    name = server.served_model_name

    @app.exception_handler(RequestError)
    async def _bad_request(request, exc):
        return JSONResponse(
            {"error": {"message": str(exc), "type": "invalid_request_error"}},
            status_code=400,
        )

    def _generate_unsupported():
        task = "transcribe" if server.stt is not None else "embed"
        return JSONResponse(
            {"error": {"message": f"model {name} runs task={task}; text "
                                  "generation endpoints are not supported"}},
            status_code=400,
        )

    def _logit_bias_from(body: dict) -> Optional[dict]:
        raw = body.get("logit_bias")
        if not raw:
            return None
        vocab = server.engine.arch.vocab_size if server.engine else None
        out: dict[int, float] = {}
        for k, v in raw.items():
            try:
                tid, bias = int(k), float(v)
            except (TypeError, ValueError):
                raise RequestError(f"logit_bias key {k!r} is not a token id")
            if tid < 0 or (vocab is not None and tid >= vocab):
                raise RequestError(
                    f"logit_bias token id {tid} out of range [0, {vocab})"
                )
            out[tid] = bias
        return out

    def _json_mode_from(body: dict):
        """-> (json_mode, schema|None)."""
        rf = body.get("response_format")
        if not rf:
            return False, None
        rtype = rf.get("type") if isinstance(rf, dict) else rf
        if rtype == "json_object":
            return True, None
        if rtype == "json_schema":
            # schema-GUIDED decoding: types, required keys, enums,
            # closed objects (jsonmode.SchemaValidator)
            js = rf.get("json_schema") if isinstance(rf, dict) else None
            schema = js.get("schema") if isinstance(js, dict) else None
            if schema is not None and not isinstance(schema, dict):
                raise RequestError("response_format json_schema.schema "
                                   "must be an object")
            return True, schema
        if rtype == "text":
            return False, None
        raise RequestError(f"unsupported response_format type {rtype!r}")

    def _params_from(body: dict) -> SamplingParams:
        mt = body.get("max_tokens") or body.get("max_completion_tokens") or 128
        temp = body.get("temperature")
        temp = 1.0 if temp is None else float(temp)
        seed = body.get("seed")
        jm, schema = _json_mode_from(body)
        return SamplingParams(
            json_mode=jm,
            json_schema=schema,
            max_tokens=int(mt),
            temperature=float(temp),
            top_p=float(body.get("top_p") or 1.0),
            top_k=int(body.get("top_k") or 0),
            seed=int(seed) if seed is not None else None,
            ignore_eos=bool(body.get("ignore_eos") or False),
            presence_penalty=float(body.get("presence_penalty") or 0.0),
            frequency_penalty=float(body.get("frequency_penalty") or 0.0),
            priority=int(body.get("priority") or 0),
            logit_bias=_logit_bias_from(body),
        )

    def _stop_strings(body: dict) -> list[str]:
        stop = body.get("stop")
        if stop is None:
            return []
        return [stop] if isinstance(stop, str) else [s for s in stop if s]

    def _usage(prompt_toks: int, completion_toks: int) -> dict:
        return {
            "prompt_tokens": prompt_toks,
            "completion_tokens": completion_toks,
            "total_tokens": prompt_toks + completion_toks,
        }

    @app.get("/health")
    async def health():
        if server._fatal_error is not None:
            return JSONResponse(
                {"status": "error", "detail": server._fatal_error.splitlines()[-1]},
                status_code=500,
            )
        if not server._ready.is_set():
            return JSONResponse({"status": "starting"}, status_code=503)
        if not server._thread.is_alive() and not server._stop.is_set():
            return JSONResponse(
                {"status": "error", "detail": "engine step thread died"},
                status_code=500,
            )
        return {"status": "ok"}

    @app.get("/version")
    async def version():
        import kubeai_amd

        return {"version": kubeai_amd.__version__}

    @app.post("/tokenize")
    async def tokenize(request: Request):
        """vLLM-compat tokenizer endpoint (clients pointed straight at an
        engine replica use it for budget accounting)."""
        body = await request.json()
        text = body.get("prompt")
        if text is None and body.get("messages"):
            toks = apply_chat_template(server.tokenizer, body["messages"])
            return {"tokens": toks, "count": len(toks),
                    "max_model_len": server.cfg.max_model_len}
        if not isinstance(text, str):
            raise RequestError("tokenize needs a string `prompt` or `messages`")
        add_special = bool(body.get("add_special_tokens", True))
        toks = server.tokenizer.encode(text, add_bos=add_special)
        return {"tokens": toks, "count": len(toks),
                "max_model_len": server.cfg.max_model_len}

    @app.post("/detokenize")
    async def detokenize(request: Request):
        body = await request.json()
        toks = body.get("tokens")
        if not isinstance(toks, list):
            raise RequestError("detokenize needs a `tokens` array")
        return {"prompt": server.tokenizer.decode([int(t) for t in toks])}

    @app.get("/metrics")
    async def metrics():
        return PlainTextResponse(
            prom.generate_latest(REGISTRY).decode(), media_type=prom.CONTENT_TYPE_LATEST
        )

    @app.get("/v1/models")
    async def models():
        data = [{"id": name, "object": "model", "owned_by": "kubeai-amd"}]
        for a in server._lora_adapters:
            data.append({"id": a, "object": "model", "owned_by": "kubeai-amd", "parent": name})
        return {"object": "list", "data": data}

    async def _finish_tokens(gen, stops=None, steps=None):
        """Drain a generation; with stop strings, truncate at the earliest
        match and abort the engine request early (the generator's finally
        clause issues the abort). `steps`, when given, collects every
        per-token RequestOutput (chat logprobs). Stop detection is
        incremental: O(new text) per token, not a full re-decode."""
        final = None
        sd = StreamDecoder(server.tokenizer) if stops else None
        sc = _StopScanner(stops) if stops else None
        async for o in gen:
            final = o
            if steps is not None:
                steps.append(o)
            if stops:
                text = sd.push(o.new_token_ids)
                idx = sc.scan(text)
                if idx >= 0:
                    return final, text[:idx], "stop"
        return final, None, None

    @app.post("/v1/completions")
    async def completions(request: Request):
        if server.stt is not None or server.embedder is not None:
            return _generate_unsupported()
        body = await request.json()
        params = _params_from(body)
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = prompt[0] if prompt else ""
        if isinstance(prompt, str):
            toks = server.tokenizer.encode(prompt, add_bos=True)
        else:
            toks = list(prompt)
        lora_id = _resolve_lora(server, body.get("model"))
        if isinstance(lora_id, JSONResponse):
            return lora_id
        if body.get("suffix"):
            # insertion/FIM is not supported (vLLM answers the same way
            # behind the reference's passthrough)
            raise RequestError("suffix is not supported")
        n = int(body.get("n") or 1)
        stops = _stop_strings(body)
        echo_text = ""
        if body.get("echo"):
            echo_text = (
                prompt if isinstance(prompt, str) else server.tokenizer.decode(toks)
            )
        if body.get("stream"):
            return StreamingResponse(
                _stream_completion(server, toks, params, name, chat=False,
                                   lora_id=lora_id, stops=stops, n=n,
                                   echo_text=echo_text),
                media_type="text/event-stream",
            )
        want_logprobs = body.get("logprobs") not in (None, False, 0)
        if want_logprobs:
            # completions `logprobs: N` = N top alternatives per token
            params.logprobs = int(body.get("logprobs") or 0)
        choices = []
        pt = ct = 0
        for i in range(n):
            lps: list = []
            toks_out: list = []
            tops: list = []
            final = None
            cut_text = cut_reason = None
            sd = StreamDecoder(server.tokenizer) if stops else None
            sc = _StopScanner(stops) if stops else None
            async for o in server.generate(toks, params, lora_id):
                final = o
                if o.logprob is not None:
                    lps.append(o.logprob)
                    toks_out.append(o.new_token_ids[-1])
                    tops.append(o.top_logprobs)
                if stops:
                    text_probe = sd.push(o.new_token_ids)
                    idx = sc.scan(text_probe)
                    if idx >= 0:
                        cut_text, cut_reason = text_probe[:idx], "stop"
                        break
            text = (
                cut_text
                if cut_text is not None
                else server.tokenizer.decode(
                    _strip_stop(final, params, server.engine.arch.eos_token_id)
                )
            )
            lp_obj = None
            if want_logprobs:
                dec = server.tokenizer.decode
                lp_obj = {
                    "tokens": [dec([t]) for t in toks_out],
                    "token_logprobs": lps,
                    "top_logprobs": [
                        {dec([tid]): lp for tid, lp in (top or [])} for top in tops
                    ],
                }
            choices.append(
                {
                    "index": i,
                    "text": echo_text + text,
                    "finish_reason": cut_reason or final.finish_reason or "stop",
                    "logprobs": lp_obj,
                }
            )
            pt = final.num_prompt_tokens
            ct += len(final.output_token_ids)
        return {
            "id": f"cmpl-{uuid.uuid4().hex[:12]}",
            "object": "text_completion",
            "created": int(time.time()),
            "model": name,
            "choices": choices,
            "usage": _usage(pt, ct),
        }

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        if server.stt is not None or server.embedder is not None:
            return _generate_unsupported()
        body = await request.json()
        params = _params_from(body)
        messages = body.get("messages", [])
        tools = body.get("tools") or []
        tool_choice = body.get("tool_choice")
        use_tools = bool(tools) and tool_choice != "none"
        if use_tools:
            # template-level function calling: tool schemas go into the
            # prompt; a forced tool_choice constrains the output to JSON
            # (the reference gets native tool support via vLLM
            # passthrough — chat_completions.go:350-515)
            messages = _with_tool_instructions(messages, tools, tool_choice)
            if tool_choice == "required" or isinstance(tool_choice, dict):
                params.json_mode = True
                # forced single function: guide decoding with the tool
                # call shape {name: <fn>, arguments: <parameters schema>}
                if isinstance(tool_choice, dict):
                    fn = (tool_choice.get("function") or {}).get("name")
                    tool = next(
                        (t for t in tools
                         if (t.get("function", t) or {}).get("name") == fn),
                        None,
                    )
                    if tool is not None:
                        fdef = tool.get("function", tool) or {}
                        params.json_schema = {
                            "type": "object",
                            "properties": {
                                "name": {"const": fn},
                                "arguments": fdef.get("parameters")
                                or {"type": "object"},
                            },
                            "required": ["name", "arguments"],
                            "additionalProperties": False,
                        }
        toks = apply_chat_template(server.tokenizer, messages)
        images = _images_from_messages(server, messages)
        lora_id = _resolve_lora(server, body.get("model"))
        if isinstance(lora_id, JSONResponse):
            return lora_id
        n = int(body.get("n") or 1)
        stops = _stop_strings(body)
        if body.get("stream"):
            return StreamingResponse(
                _stream_completion(server, toks, params, name, chat=True,
                                   lora_id=lora_id, stops=stops, n=n,
                                   images=images),
                media_type="text/event-stream",
            )
        want_logprobs = bool(body.get("logprobs"))
        if want_logprobs:
            # chat `logprobs: true` + optional `top_logprobs: N`
            params.logprobs = int(body.get("top_logprobs") or 0)
        choices = []
        pt = ct = 0
        for i in range(n):
            steps: list = [] if want_logprobs else None
            final, cut_text, cut_reason = await _finish_tokens(
                server.generate(toks, params, lora_id, images=images), stops,
                steps
            )
            text = (
                cut_text
                if cut_text is not None
                else server.tokenizer.decode(
                    _strip_stop(final, params, server.engine.arch.eos_token_id)
                )
            )
            tc = _parse_tool_call(text, tools) if use_tools else None
            if tc is not None:
                fname, fargs = tc
                choice = {
                    "index": i,
                    "message": {
                        "role": "assistant",
                        "content": None,
                        "tool_calls": [
                            {
                                "id": f"call_{uuid.uuid4().hex[:12]}",
                                "type": "function",
                                "function": {
                                    "name": fname,
                                    "arguments": json.dumps(fargs),
                                },
                            }
                        ],
                    },
                    "finish_reason": "tool_calls",
                }
            else:
                choice = {
                    "index": i,
                    "message": {"role": "assistant", "content": text},
                    "finish_reason": cut_reason or final.finish_reason or "stop",
                }
            if want_logprobs:
                dec = server.tokenizer.decode
                choice["logprobs"] = {
                    "content": [
                        {
                            "token": dec([o.new_token_ids[-1]]),
                            "logprob": o.logprob,
                            "top_logprobs": [
                                {"token": dec([tid]), "logprob": lp}
                                for tid, lp in (o.top_logprobs or [])
                            ],
                        }
                        for o in steps
                        if o.logprob is not None
                    ]
                }
            choices.append(choice)
            pt = final.num_prompt_tokens
            ct += len(final.output_token_ids)
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:12]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": name,
            "choices": choices,
            "usage": _usage(pt, ct),
        }

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        if server.stt is not None:
            return _generate_unsupported()
        body = await request.json()
        inputs = body.get("input", [])
        if isinstance(inputs, str):
            inputs = [inputs]
        tok_lists = [server.tokenizer.encode(t, add_bos=True) for t in inputs]
        if server.embedder is not None:
            # BERT-architecture embedding model (task=embed)
            enc = await asyncio.get_running_loop().run_in_executor(
                None, server.embedder.encode, tok_lists
            )
            vecs = enc.cpu().tolist()
        else:
            vecs = await server.embed(tok_lists)
        data = [
            {"object": "embedding", "index": i, "embedding": v}
            for i, v in enumerate(vecs)
        ]
        return {
            "object": "list",
            "data": data,
            "model": name,
            "usage": _usage(sum(len(t) for t in tok_lists), 0),
        }

    @app.post("/v1/rerank")
    async def rerank(request: Request):
        """Score documents against a query (Reranking feature; reference
        analog: the Infinity engine's /rerank)."""
        if server.stt is not None:
            return _generate_unsupported()
        body = await request.json()
        query = body.get("query", "")
        docs = body.get("documents", []) or []
        top_n = body.get("top_n") or len(docs)
        tok_lists = [server.tokenizer.encode(t, add_bos=True) for t in [query] + docs]
        if server.embedder is not None:
            scores = await asyncio.get_running_loop().run_in_executor(
                None, server.embedder.score_pairs, tok_lists[0], tok_lists[1:]
            )
        else:
            vecs = await server.embed(tok_lists)
            qv = vecs[0]
            scores = [
                sum(a * b for a, b in zip(qv, dv)) for dv in vecs[1:]
            ]  # unit-norm vectors -> cosine
        order = sorted(range(len(docs)), key=lambda i: -scores[i])[: int(top_n)]
        return {
            "model": name,
            "results": [
                {"index": i, "relevance_score": scores[i],
                 "document": {"text": docs[i]}}
                for i in order
            ],
            "usage": _usage(sum(len(t) for t in tok_lists), 0),
        }

    @app.post("/v1/audio/transcriptions")
    async def transcriptions(request: Request):
        # SpeechToText feature (reference routes it to the FasterWhisper
        # engine; here the in-house whisper-architecture model serves it,
        # kubeai_amd/models/whisper.py). Text-generation models answer 501.
        if server.stt is None:
            return JSONResponse(
                {"error": {"message": f"model {name} does not support audio "
                                      "transcription (task=generate)"}},
                status_code=501,
            )
        ctype = request.headers.get("content-type", "")
        raw = await request.body()
        if not ctype.startswith("multipart/form-data"):
            return JSONResponse(
                {"error": "expected multipart/form-data with a `file` field"},
                status_code=400,
            )
        fields = _parse_multipart(raw, ctype)
        wav = fields.get("file")
        if wav is None:
            return JSONResponse({"error": "missing `file` field"}, status_code=400)
        try:
            audio, sr = _decode_wav(wav)
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"error": f"cannot decode audio: {e}"},
                                status_code=400)
        loop = asyncio.get_running_loop()
        toks = await loop.run_in_executor(
            None, server.stt.transcribe_tokens, audio, sr
        )
        # strip sot/eot specials before decoding
        cfg_ = server.stt.cfg
        text = server.tokenizer.decode(
            [t for t in toks if t not in (cfg_.sot_token, cfg_.eot_token)]
        )
        fmt = (fields.get("response_format") or b"json").decode(errors="replace")
        duration = len(audio) / sr if sr else 0.0
        if fmt == "text":
            return PlainTextResponse(text)
        out = {"text": text}
        if fmt == "verbose_json":
            out.update({"task": "transcribe", "duration": duration,
                        "language": fields.get("language", b"en").decode()})
        return out

    @app.post("/v1/load_lora_adapter")
    async def load_lora(request: Request):
        body = await request.json()
        lname, path = body.get("lora_name"), body.get("lora_path")
        with server._lock:
            if lname in server._lora_adapters:
                return JSONResponse(
                    {"error": f"adapter {lname} was already loaded"}, status_code=400
                )
            try:
                server.load_lora(lname, path)
            except Exception as e:  # noqa: BLE001
                return JSONResponse({"error": str(e)}, status_code=400)
            server._lora_adapters[lname] = path
        return PlainTextResponse("OK")

    @app.post("/v1/unload_lora_adapter")
    async def unload_lora(request: Request):
        body = await request.json()
        lname = body.get("lora_name")
        with server._lock:
            if lname not in server._lora_adapters:
                return JSONResponse(
                    {"error": f"adapter {lname} cannot be found"}, status_code=404
                )
            server._lora_adapters.pop(lname)
            server.unload_lora(lname)
        return PlainTextResponse("OK")

    return app


def _parse_multipart(raw: bytes, content_type: str) -> dict[str, bytes]:
    """Minimal multipart/form-data parser (no python-multipart in this
    image): returns {field-name: body-bytes}; good for one file + a few
    small text fields, which is the transcriptions contract."""
    m = re.search(r'boundary="?([^";,]+)"?', content_type)
    if not m:
        return {}
    boundary = b"--" + m.group(1).encode()
    fields: dict[str, bytes] = {}
    for part in raw.split(boundary):
        part = part.strip(b"\r\n")
        if not part or part == b"--":
            continue
        head, _, body = part.partition(b"\r\n\r\n")
        nm = re.search(rb'name="([^"]+)"', head)
        if nm:
            fields[nm.group(1).decode(errors="replace")] = body
    return fields


def _decode_wav(data: bytes):
    """WAV bytes -> (float32 mono ndarray in [-1,1], sample_rate)."""
    import io

    import numpy as np
    from scipy.io import wavfile

    sr, audio = wavfile.read(io.BytesIO(data))
    if audio.dtype.kind == "i":
        audio = audio.astype(np.float32) / float(np.iinfo(audio.dtype).max)
    elif audio.dtype.kind == "u":  # u8 wav
        audio = (audio.astype(np.float32) - 128.0) / 128.0
    else:
        audio = audio.astype(np.float32)
    return audio, sr


def _strip_stop(final, params: SamplingParams, eos_id=None) -> list[int]:
    # the engine appends the model EOS to its own copy of the params, so
    # the server-side tuple alone is not enough (JSON mode force-stops
    # with a bare EOS)
    toks = final.output_token_ids
    stops = set(params.stop_token_ids)
    if eos_id is not None:
        stops.add(eos_id)
    if final.finish_reason == "stop" and toks and toks[-1] in stops:
        return toks[:-1]
    return toks


def _with_tool_instructions(messages: list, tools: list, tool_choice) -> list:
    """Prepend a system message describing the available functions and the
    JSON call convention (template-level tool support)."""
    specs = []
    for t in tools:
        fn = t.get("function", t)
        specs.append({
            "name": fn.get("name"),
            "description": fn.get("description", ""),
            "parameters": fn.get("parameters", {}),
        })
    want = ""
    if isinstance(tool_choice, dict):
        forced = (tool_choice.get("function") or {}).get("name")
        if forced:
            want = f" You must call the function {forced!r}."
    elif tool_choice == "required":
        want = " You must call one of the functions."
    content = (
        "You can call functions. Available functions (JSON schemas): "
        + json.dumps(specs)
        + '. To call a function, respond ONLY with a JSON object '
          '{"name": <function name>, "arguments": <arguments object>}.'
        + want
    )
    return [{"role": "system", "content": content}] + list(messages)


def _parse_tool_call(text: str, tools: list):
    """-> (name, arguments) when the output is a function-call JSON."""
    try:
        obj = json.loads(text)
    except Exception:  # noqa: BLE001
        return None
    if not isinstance(obj, dict) or not isinstance(obj.get("name"), str):
        return None
    names = {
        (t.get("function", t) or {}).get("name") for t in tools
    }
    if obj["name"] not in names:
        return None
    args = obj.get("arguments")
    return obj["name"], (args if isinstance(args, dict) else {})


def _resolve_lora(server: EngineServer, model_field: Optional[str]):
    """Model field == adapter name selects the adapter (vLLM semantics)."""
    if not model_field or model_field == server.served_model_name:
        return 0
    with server._lock:
        if model_field in server._lora_adapters:
            return server.lora_id_of(model_field)
    return JSONResponse(
        {"error": {"message": f"model {model_field} cannot be found"}}, status_code=404
    )


class _ChoiceStream:
    """Per-choice incremental state for streaming responses."""

    def __init__(self, server, params, stops, echo_text: str = ""):
        self.sd = StreamDecoder(server.tokenizer)
        self.sc = _StopScanner(stops) if stops else None
        self.stops = stops
        self.params = params
        self.server = server
        self.emitted = 0
        self.n_out = 0
        self.prompt_tokens = 0
        self.echo_text = echo_text  # flushed with the first piece

    def step(self, o):
        """-> (piece, finished, finish_reason)."""
        new_ids = o.new_token_ids
        if (
            o.finished
            and o.finish_reason == "stop"
            and new_ids
            and new_ids[-1]
            in set(self.params.stop_token_ids)
            | {self.server.engine.arch.eos_token_id}
        ):
            new_ids = new_ids[:-1]
        text = self.sd.push(new_ids)
        finished, reason = o.finished, o.finish_reason
        if self.stops:
            hit = self.sc.scan(text)
            if hit >= 0:
                text = text[:hit]
                finished, reason = True, "stop"
        if finished:
            stable = text
        else:
            # hold back any tail that could still complete a stop string
            # (never leak a stop's leading chars across chunk boundaries)
            stable = (
                text[: len(text) - _holdback_len(text, self.stops)]
                if self.stops
                else text
            )
        piece = stable[self.emitted:] if len(stable) > self.emitted else ""
        self.emitted = max(self.emitted, len(stable))
        if self.echo_text:
            piece = self.echo_text + piece
            self.echo_text = ""
        self.n_out = len(o.output_token_ids)
        self.prompt_tokens = o.num_prompt_tokens
        return piece, finished, reason


async def _stream_completion(server, toks, params, name, chat: bool,
                             lora_id: int = 0, stops=None, n: int = 1,
                             echo_text: str = "", images=None):
    """SSE stream; n>1 runs n engine requests concurrently, interleaving
    chunks with their choice index (the reference gets this via vLLM
    passthrough; r1 rejected stream+n>1)."""
    rid = f"{'chatcmpl' if chat else 'cmpl'}-{uuid.uuid4().hex[:12]}"
    created = int(time.time())
    agg: asyncio.Queue = asyncio.Queue()

    async def pump(i: int):
        async for o in server.generate(toks, params, lora_id, images=images):
            await agg.put((i, o))

    tasks = [asyncio.create_task(pump(i)) for i in range(n)]
    states = [_ChoiceStream(server, params, stops, echo_text) for _ in range(n)]
    live = set(range(n))
    try:
        while live:
            i, o = await agg.get()
            if i not in live:
                continue
            st = states[i]
            piece, finished, reason = st.step(o)
            if finished:
                live.discard(i)
                tasks[i].cancel()
            if chat:
                choice = {
                    "index": i,
                    "delta": {"content": piece} if piece else {},
                    "finish_reason": reason if finished else None,
                }
            else:
                choice = {
                    "index": i,
                    "text": piece,
                    "finish_reason": reason if finished else None,
                }
            chunk = {
                "id": rid,
                "object": "chat.completion.chunk" if chat else "text_completion",
                "created": created,
                "model": name,
                "choices": [choice],
            }
            if not live:  # final chunk carries aggregate usage
                pt = states[0].prompt_tokens
                ct = sum(s.n_out for s in states)
                chunk["usage"] = {
                    "prompt_tokens": pt,
                    "completion_tokens": ct,
                    "total_tokens": pt + ct,
                }
            yield f"data: {json.dumps(chunk)}\n\n"
        yield "data: [DONE]\n\n"
    finally:
        for t in tasks:
            t.cancel()


# --------------------------------------------------------------------------
def _tp_worker_main(rank: int, world: int, port: int, cfg: EngineConfig) -> None:
    """TP worker rank (no HTTP): lockstep scheduler driven by rank 0."""
    import torch
    import torch.distributed as dist

    from kubeai_amd.parallel.tp import TPGroup

    use_cuda = cfg.resolve_device().startswith("cuda")
    if use_cuda:
        torch.cuda.set_device(rank)
        cfg = type(cfg)(**{**cfg.__dict__, "device": f"cuda:{rank}"})
    dist.init_process_group(
        "nccl" if use_cuda else "gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world,
    )
    engine = LLMEngine(cfg, tp_group=TPGroup())
    # identical tokenizer => identical JSON-mode sampling on every rank
    engine.runner.tokenizer = load_tokenizer(
        cfg.model, engine.arch.vocab_size, engine.arch.bos_token_id,
        engine.arch.eos_token_id,
    )
    if engine.arch.vision is not None and hasattr(
        engine.runner.tokenizer, "image_token_id"
    ):
        engine.runner.tokenizer.image_token_id = engine.arch.image_token_id
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0)
        msg = box[0]
        if msg["shutdown"]:
            break
        for item in msg["new"]:
            rid, toks, params, lora_id, images = item
            try:
                engine.add_request(
                    toks, params, request_id=rid, lora_id=lora_id,
                    images=images or None,
                )
            except RequestError:
                continue  # rank 0 already rejected it identically
        for rid in msg["aborts"]:
            engine.abort_request(rid)
        for tok_lists in msg.get("embeds", []):
            engine.embed(tok_lists)  # lockstep with rank 0 (all-reduces)
        if engine.has_work():
            engine.step()
    dist.destroy_process_group()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", required=True, help="preset name or model dir")
    p.add_argument("--served-model-name", default=None)
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--device", default="auto")
    p.add_argument("--max-model-len", type=int, default=8192)
    p.add_argument("--max-num-seqs", type=int, default=256)
    p.add_argument("--gpu-memory-utilization", type=float, default=0.90)
    p.add_argument("--num-gpu-blocks", type=int, default=None)
    p.add_argument("--kv-cache-dtype", choices=["auto", "fp8_e5m2"],
                   default="auto")
    p.add_argument("--quantization", choices=["fp8"], default=None,
                   help="fp8 = W8A8 dynamic with fused activation quant")
    p.add_argument("--enable-lora", action="store_true")
    p.add_argument("--tensor-parallel-size", type=int, default=1)
    p.add_argument("--task", choices=["generate", "transcribe", "embed"],
                   default=None,
                   help="auto: whisper-* -> transcribe, bert/bge/e5 -> embed")
    args = p.parse_args()
    base = os.path.basename(args.model.rstrip("/")).lower()
    if args.task:
        task = args.task
    elif base.startswith("whisper"):
        task = "transcribe"
    elif base.startswith(("bert", "bge", "e5")):
        task = "embed"
    else:
        task = "generate"

    cfg = EngineConfig(
        model=args.model,
        device=args.device,
        max_model_len=args.max_model_len,
        max_num_seqs=args.max_num_seqs,
        gpu_memory_utilization=args.gpu_memory_utilization,
        num_gpu_blocks=args.num_gpu_blocks,
        kv_cache_dtype=args.kv_cache_dtype,
        quantization=args.quantization,
    )
    served = args.served_model_name or os.path.basename(args.model.rstrip("/"))
    tp = args.tensor_parallel_size
    tp_port = 0
    workers = []
    if tp > 1:
        import socket

        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            tp_port = s.getsockname()[1]
        import torch.multiprocessing as mp

        ctx = mp.get_context("spawn")
        for r in range(1, tp):
            p = ctx.Process(
                target=_tp_worker_main, args=(r, tp, tp_port, cfg), daemon=True
            )
            p.start()
            workers.append(p)
        if cfg.device in ("auto", "cuda"):
            cfg = EngineConfig(**{**cfg.__dict__, "device": "cuda:0"})
    server = EngineServer(cfg, served, tp_size=tp, tp_port=tp_port, task=task)
    server.start()
    app = build_app(server)
    import uvicorn

    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
