"""LLMEngine — the per-GPU serving engine facade.

add_request() / step() / abort(); one step = schedule -> forward -> commit.
This is the in-house replacement for the vLLM container the reference
launches per Model pod.
"""
from __future__ import annotations

import dataclasses
import time
from typing import Optional

import torch

from kubeai_amd.models.config import ModelArchConfig, PRESETS

from .kvcache import BlockManager
from .runner import ModelRunner
from .scheduler import Request, SamplingParams, Scheduler


class RequestError(ValueError):
    """Client error in a request body -> HTTP 400."""


@dataclasses.dataclass
class EngineConfig:
    model: str = "llama-tiny"  # preset name or HF dir
    device: str = "auto"  # auto -> cuda if available else cpu
    dtype: str = "bfloat16"
    block_size: int = 16
    num_gpu_blocks: Optional[int] = None  # None -> profile from free HBM
    gpu_memory_utilization: float = 0.90
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 8192
    max_model_len: int = 8192
    enable_prefix_caching: bool = True
    quantization: Optional[str] = None  # "fp8" => W8A8 dynamic (quant.py)
    # "auto" = model dtype; "fp8_e5m2" halves KV bytes + doubles capacity
    kv_cache_dtype: str = "auto"
    enable_graphs: bool = True  # hipGraph capture for pure-decode steps
    prefill_interval: int = 1  # >1: batch prefills onto every Nth step
    seed: int = 0

    def resolve_arch(self) -> ModelArchConfig:
        if self.model in PRESETS:
            return PRESETS[self.model]
        return ModelArchConfig.from_hf_config(self.model)

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        return "cuda" if torch.cuda.is_available() else "cpu"


@dataclasses.dataclass
class RequestOutput:
    request_id: str
    new_token_ids: list[int]
    finished: bool
    finish_reason: Optional[str]  # "stop" | "length" | "abort"
    num_prompt_tokens: int
    num_cached_tokens: int
    output_token_ids: list[int]
    logprob: Optional[float] = None  # logprob of new_token_ids[-1]
    # top alternatives for new_token_ids[-1]: [(token_id, logprob), ...]
    top_logprobs: Optional[list] = None


class LLMEngine:
    def __init__(self, cfg: EngineConfig, tp_group=None):
        self.cfg = cfg
        arch = cfg.resolve_arch()
        self.arch = arch
        device = cfg.resolve_device()
        self.runner = ModelRunner(
            arch,
            device=device,
            dtype=getattr(torch, cfg.dtype),
            block_size=cfg.block_size,
            num_gpu_blocks=cfg.num_gpu_blocks,
            gpu_memory_utilization=cfg.gpu_memory_utilization,
            seed=cfg.seed,
            tp_group=tp_group,
            max_model_len=cfg.max_model_len,
            enable_graphs=cfg.enable_graphs,
            quantization=cfg.quantization,
            model_path=cfg.model,
            kv_cache_dtype=cfg.kv_cache_dtype,
        )
        self.block_manager = BlockManager(self.runner.num_blocks, cfg.block_size)
        self.scheduler = Scheduler(
            self.block_manager,
            max_num_seqs=cfg.max_num_seqs,
            max_num_batched_tokens=cfg.max_num_batched_tokens,
            max_model_len=cfg.max_model_len,
            enable_prefix_caching=cfg.enable_prefix_caching,
            prefill_interval=cfg.prefill_interval,
        )
        self.step_count = 0

    # ------------------------------------------------------------------
    def add_request(
        self,
        prompt_token_ids: list[int],
        params: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
        lora_id: int = 0,
        images: Optional[list] = None,
    ) -> Request:
        if params is None:
            params = SamplingParams()
        if not params.ignore_eos and self.arch.eos_token_id not in params.stop_token_ids:
            params = dataclasses.replace(
                params,
                stop_token_ids=tuple(params.stop_token_ids)
                + (self.arch.eos_token_id,),
            )
        mm_spans: list[tuple[int, int, int]] = []
        if images:
            # expand each image-placeholder token to n_patches positions
            # (vLLM-style); the runner replaces those embedding rows with
            # the vision-tower output
            if self.arch.vision is None:
                raise RequestError(
                    f"model {self.cfg.model!r} does not accept image input"
                )
            img_id = self.arch.image_token_id
            n_patch = self.runner.vision.n_patches
            n_slots = sum(1 for t in prompt_token_ids if t == img_id)
            if n_slots != len(images):
                raise RequestError(
                    f"prompt has {n_slots} image slot(s) but request "
                    f"carries {len(images)} image(s)"
                )
            expanded: list[int] = []
            k = 0
            for t in prompt_token_ids:
                if t == img_id:
                    mm_spans.append((len(expanded), n_patch, k * n_patch))
                    expanded.extend([img_id] * n_patch)
                    k += 1
                else:
                    expanded.append(t)
            prompt_token_ids = expanded
            if len(prompt_token_ids) >= self.cfg.max_model_len:
                raise RequestError(
                    "multimodal prompt exceeds max_model_len "
                    "(image spans cannot be truncated)"
                )
        if len(prompt_token_ids) >= self.cfg.max_model_len:
            prompt_token_ids = prompt_token_ids[-(self.cfg.max_model_len - 1) :]
        req = Request(prompt_token_ids, params, request_id=request_id, lora_id=lora_id)
        if images:
            req.images = list(images)
            req.mm_spans = mm_spans
            # salt the prefix cache with image content: same token ids,
            # different pixels -> different KV
            req.cache_salt = hash(
                (lora_id,)
                + tuple(hash(im.numpy().tobytes()) for im in req.images)
            )
        self.scheduler.add_request(req)
        return req

    def abort_request(self, request_id: str) -> None:
        self.scheduler.abort(request_id)
        self.runner.release_request(request_id)

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    # ------------------------------------------------------------------
    def step(self) -> list[RequestOutput]:
        out = self.scheduler.schedule()
        results: list[RequestOutput] = []
        for req in out.rejected:
            results.append(
                RequestOutput(
                    request_id=req.request_id,
                    new_token_ids=[],
                    finished=True,
                    finish_reason="abort",
                    num_prompt_tokens=req.num_prompt_tokens,
                    num_cached_tokens=0,
                    output_token_ids=[],
                )
            )
        if out.is_empty:
            return results
        sampled = self.runner.execute(out, self.step_count)
        self.step_count += 1
        now = time.monotonic()
        finished = self.scheduler.finish_step(out, sampled, now)
        finished_ids = {r.request_id for r in finished}
        for rid in finished_ids:
            self.runner.release_request(rid)
        for ss in out.all_seqs:
            req = ss.req
            if not ss.samples:
                continue
            tok = sampled[req.request_id]
            results.append(
                RequestOutput(
                    request_id=req.request_id,
                    new_token_ids=[tok],
                    finished=req.request_id in finished_ids,
                    finish_reason=req.status.value if req.status.finished else None,
                    num_prompt_tokens=req.num_prompt_tokens,
                    num_cached_tokens=req.num_cached_prompt_tokens,
                    output_token_ids=req.output_token_ids,
                    logprob=getattr(self.runner, "last_logprobs", {}).get(
                        req.request_id
                    ),
                    top_logprobs=getattr(self.runner, "last_top_logprobs", {}).get(
                        req.request_id
                    ),
                )
            )
        return results

    # ------------------------------------------------------------------
    def embed(self, tok_lists: list[list[int]]) -> list[list[float]]:
        """Mean-pooled, L2-normalised final hidden states (TextEmbedding
        feature; reference analog: the Infinity engine).

        Sequences are packed many-per-forward (token-budget bounded), the
        same packed-prefill shape the generation path uses — batched
        embedding throughput instead of one forward per input.

        MUST run on the stepping thread (the server routes embed jobs
        through its submit loop): it shares the BlockManager and KV
        tensors with in-flight generations.
        """
        from .scheduler import ScheduledSeq, SchedulerOutput

        out: list[Optional[list[float]]] = [None] * len(tok_lists)
        budget = self.cfg.max_num_batched_tokens
        i = 0
        while i < len(tok_lists):
            batch: list[tuple[int, Request, list[int]]] = []
            total = 0
            while i < len(tok_lists):
                toks = list(tok_lists[i][: self.cfg.max_model_len - 1])
                if batch and total + len(toks) > budget:
                    break
                table, _ = self.block_manager.allocate(toks, salt=-1, max_cached=0)
                req = Request(toks, SamplingParams(max_tokens=1))
                req.block_table = table
                batch.append((i, req, toks))
                total += len(toks)
                i += 1
            so = SchedulerOutput(
                decode=[],
                prefill=[ScheduledSeq(r, 0, len(t)) for _, r, t in batch],
                preempted=[],
            )
            fb = self.runner.build_batch(so)
            hidden = self.runner.model(fb)
            ofs = 0
            for idx, req, toks in batch:
                # rows are packed prefill-order; bucket padding sits past
                # the real tokens
                vec = hidden[ofs : ofs + len(toks)].float().mean(dim=0)
                vec = vec / vec.norm().clamp_min(1e-12)
                out[idx] = vec.cpu().tolist()
                ofs += len(toks)
                self.block_manager.free(req.block_table)
        return out

    # ------------------------------------------------------------------
    # LoRA adapter lifecycle (weights managed by the runner's LoRA manager)
    def load_lora(self, lora_id: int, path: Optional[str]) -> None:
        self.runner.load_lora(lora_id, path)

    def unload_lora(self, lora_id: int) -> None:
        self.runner.unload_lora(lora_id)

    # ------------------------------------------------------------------
    # metrics for /metrics + the autoscaler
    def stats(self) -> dict:
        return {
            "num_running": self.scheduler.num_running,
            "num_waiting": self.scheduler.num_waiting,
            "kv_usage": self.block_manager.usage(),
            "prefix_cache_hit_rate": self.block_manager.hit_rate(),
            "step_count": self.step_count,
        }
