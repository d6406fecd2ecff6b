"""Continuous-batching scheduler (chunked prefill, prefix cache, preemption).

One engine step = one call to schedule() -> forward -> finish_step(). Decode
sequences and prefill chunks share a single token-packed forward (decode
first), bounded by max_num_batched_tokens, vLLM-v1 style (re-designed; the
reference has no scheduler — it delegates to vLLM).
"""
from __future__ import annotations

import dataclasses
import enum
import time
from collections import deque
from typing import Optional

from .kvcache import BlockManager, NoFreeBlocks


@dataclasses.dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0  # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0
    seed: Optional[int] = None
    stop_token_ids: tuple[int, ...] = ()
    ignore_eos: bool = False
    # OpenAI penalties over generated tokens (vLLM semantics)
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    # OpenAI logit_bias: {token_id: additive bias} applied before sampling
    logit_bias: Optional[dict] = None
    # number of top-alternative logprobs to return per sampled token
    # (OpenAI `logprobs`/`top_logprobs`); 0 = none
    logprobs: int = 0
    # admission priority: LOWER value is served first (vLLM semantics);
    # FIFO within a priority class. Maps the Model CRD's priorityClassName
    # analog onto per-request scheduling.
    priority: int = 0
    # OpenAI response_format json_object: the sampler enforces that the
    # output is a valid JSON object (engine/jsonmode.py)
    json_mode: bool = False
    # OpenAI response_format json_schema: schema-guided decoding (types,
    # required keys, enums, closed objects — jsonmode.SchemaValidator)
    json_schema: Optional[dict] = None


class RequestStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"
    FINISHED_STOPPED = "stop"
    FINISHED_LENGTH = "length"
    FINISHED_ABORTED = "abort"

    @property
    def finished(self) -> bool:
        return self in (
            RequestStatus.FINISHED_STOPPED,
            RequestStatus.FINISHED_LENGTH,
            RequestStatus.FINISHED_ABORTED,
        )


class Request:
    _counter = 0

    def __init__(
        self,
        prompt_token_ids: list[int],
        params: SamplingParams,
        request_id: Optional[str] = None,
        lora_id: int = 0,
        arrival_time: Optional[float] = None,
    ):
        if request_id is None:
            Request._counter += 1
            request_id = f"req-{Request._counter}"
        self.request_id = request_id
        self.prompt_token_ids = list(prompt_token_ids)
        self.params = params
        self.lora_id = lora_id  # kv-cache salt + LoRA adapter selector
        # multimodal: preprocessed pixel tensors + placeholder spans
        # (start, length, embed_row_base) in prompt coordinates; embeds
        # are computed lazily by the runner at first prefill
        self.images: list = []
        self.mm_spans: list[tuple[int, int, int]] = []
        self.mm_embeds = None
        # prefix-cache salt: lora_id plus (for multimodal) the image
        # content hash — identical placeholder ids with different images
        # must never share KV blocks
        self.cache_salt: int = lora_id
        self.arrival_time = arrival_time if arrival_time is not None else time.monotonic()
        self.first_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None

        self.status = RequestStatus.WAITING
        self.tokens: list[int] = list(prompt_token_ids)  # prompt + generated
        self.num_prompt_tokens = len(prompt_token_ids)
        self.num_computed = 0  # tokens whose KV sits in the cache
        self.block_table: list[int] = []
        self.block_hashes: list[int] = []  # sealed-block hash chain
        self.num_cached_prompt_tokens = 0  # prefix-cache hits at admission

    # ----------------------------------------------------------------
    @property
    def num_generated(self) -> int:
        return len(self.tokens) - self.num_prompt_tokens

    @property
    def in_prefill(self) -> bool:
        return self.num_computed < self.num_prompt_tokens

    @property
    def output_token_ids(self) -> list[int]:
        return self.tokens[self.num_prompt_tokens :]


@dataclasses.dataclass
class ScheduledSeq:
    req: Request
    chunk_start: int  # first token index computed this step
    chunk_len: int    # tokens computed this step

    @property
    def samples(self) -> bool:
        """Does this step produce a sampled token for the request?"""
        return self.chunk_start + self.chunk_len >= self.req.num_prompt_tokens


@dataclasses.dataclass
class SchedulerOutput:
    decode: list[ScheduledSeq]
    prefill: list[ScheduledSeq]
    preempted: list[Request]
    # requests that can never fit the KV pool (rejected at admission)
    rejected: list[Request] = dataclasses.field(default_factory=list)

    @property
    def all_seqs(self) -> list[ScheduledSeq]:
        return self.decode + self.prefill

    @property
    def total_tokens(self) -> int:
        return sum(s.chunk_len for s in self.all_seqs)

    @property
    def is_empty(self) -> bool:
        return not self.decode and not self.prefill


class Scheduler:
    def __init__(
        self,
        block_manager: BlockManager,
        max_num_seqs: int = 256,
        max_num_batched_tokens: int = 8192,
        max_model_len: int = 8192,
        enable_prefix_caching: bool = True,
        prefill_interval: int = 1,
    ):
        self.bm = block_manager
        self.max_num_seqs = max_num_seqs
        self.max_num_batched_tokens = max_num_batched_tokens
        self.max_model_len = max_model_len
        self.enable_prefix_caching = enable_prefix_caching
        # prefill_interval > 1: batch prefill work onto every Nth step so
        # the steps in between are pure decode (hipGraph-replayable). Waiting
        # prefills still run immediately when nothing is decoding.
        self.prefill_interval = max(1, prefill_interval)
        self._step_idx = 0
        self.waiting: deque[Request] = deque()
        self.running: list[Request] = []
        self._aborted: set[str] = set()

    # ----------------------------------------------------------------
    def add_request(self, req: Request) -> None:
        if len(req.prompt_token_ids) + req.params.max_tokens > self.max_model_len:
            # trim generation budget instead of rejecting (caller validates)
            req.params.max_tokens = max(
                1, self.max_model_len - len(req.prompt_token_ids)
            )
        self._insert_waiting(req, retry=False)

    def _insert_waiting(self, req: Request, retry: bool) -> None:
        """Priority-ordered insert: before the first request of a strictly
        worse class (FIFO within a class); a preempted retry goes to the
        FRONT of its own class (recompute-first policy)."""
        p = req.params.priority
        for i, other in enumerate(self.waiting):
            op = other.params.priority
            if op > p or (retry and op == p):
                self.waiting.insert(i, req)
                return
        self.waiting.append(req)

    def abort(self, request_id: str) -> None:
        self._aborted.add(request_id)

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    # ----------------------------------------------------------------
    def schedule(self) -> SchedulerOutput:
        self._apply_aborts()
        self._step_idx += 1
        budget = self.max_num_batched_tokens
        decode: list[ScheduledSeq] = []
        prefill: list[ScheduledSeq] = []
        preempted: list[Request] = []
        rejected: list[Request] = []
        # liveness: a request larger than the ENTIRE pool would block the
        # queue head forever — reject it outright
        bs = self.bm.block_size
        while self.waiting:
            head = self.waiting[0]
            need = (len(head.prompt_token_ids) + head.params.max_tokens + bs - 1) // bs
            if need > self.bm.num_blocks:
                head.status = RequestStatus.FINISHED_ABORTED
                head.finish_time = time.monotonic()
                rejected.append(self.waiting.popleft())
            else:
                break
        have_decodes = any(
            not r.in_prefill and r.status == RequestStatus.RUNNING
            for r in self.running
        )
        defer_prefill = (
            self.prefill_interval > 1
            and have_decodes
            and self._step_idx % self.prefill_interval != 0
        )

        # 1) running sequences: decodes first (latency-critical; each needs 1
        #    token), then mid-prefill chunks. Iterate copies: preemption
        #    mutates self.running. A preemption victim is only ever picked
        #    among not-yet-scheduled requests (see _pick_victim).
        self._scheduled_ids: set[str] = set()
        for req in [r for r in list(self.running) if not r.in_prefill]:
            if budget <= 0 or req.status != RequestStatus.RUNNING:
                continue
            if not self._ensure_blocks(req, req.num_computed + 1, preempted):
                continue
            decode.append(ScheduledSeq(req, req.num_computed, 1))
            self._scheduled_ids.add(req.request_id)
            budget -= 1
        for req in [r for r in list(self.running) if r.in_prefill]:
            if budget <= 0 or req.status != RequestStatus.RUNNING or defer_prefill:
                continue
            chunk = min(req.num_prompt_tokens - req.num_computed, budget)
            if not self._ensure_blocks(req, req.num_computed + chunk, preempted):
                continue  # req itself got preempted
            prefill.append(ScheduledSeq(req, req.num_computed, chunk))
            self._scheduled_ids.add(req.request_id)
            budget -= chunk

        # 2) admit waiting requests
        while (
            not defer_prefill
            and self.waiting
            and budget > 0
            and len(self.running) < self.max_num_seqs
        ):
            req = self.waiting[0]
            n_prompt = req.num_prompt_tokens
            try:
                table, n_cached = self.bm.allocate(
                    req.tokens[:n_prompt],
                    salt=req.cache_salt,
                    max_cached=(n_prompt - 1)
                    if self.enable_prefix_caching
                    else 0,
                )
            except NoFreeBlocks:
                break
            req.block_table = table
            req.num_computed = n_cached
            req.num_cached_prompt_tokens = n_cached
            # rebuild the sealed-hash chain for the cached prefix
            req.block_hashes = []
            parent = None
            bs = self.bm.block_size
            from .kvcache import hash_block

            for i in range(n_cached // bs):
                parent = hash_block(
                    parent, tuple(req.tokens[i * bs : (i + 1) * bs]), req.lora_id
                )
                req.block_hashes.append(parent)
            self.waiting.popleft()
            req.status = RequestStatus.RUNNING
            self.running.append(req)
            chunk = min(n_prompt - n_cached, budget)
            prefill.append(ScheduledSeq(req, n_cached, chunk))
            budget -= chunk

        return SchedulerOutput(
            decode=decode, prefill=prefill, preempted=preempted, rejected=rejected
        )

    def _ensure_blocks(
        self, req: Request, needed_tokens: int, preempted: list[Request]
    ) -> bool:
        """Grow req's block table to cover needed_tokens; preempt on pressure.

        Returns False if req itself had to be preempted.
        """
        bs = self.bm.block_size
        while len(req.block_table) * bs < needed_tokens:
            try:
                self.bm.append_block(req.block_table)
            except NoFreeBlocks:
                victim = self._pick_victim(req)
                if victim is None:
                    self._preempt(req, preempted)
                    return False
                self._preempt(victim, preempted)
        return True

    def _pick_victim(self, requester: Request) -> Optional[Request]:
        # preempt the worst-priority, youngest running request that is not
        # the requester and has not already been scheduled in this step (its
        # ScheduledSeq would otherwise reference freed blocks)
        victim = None
        for pos, req in enumerate(self.running):
            if req is not requester and req.request_id not in self._scheduled_ids:
                key = (req.params.priority, pos)
                if victim is None or key > victim[0]:
                    victim = (key, req)
        return victim[1] if victim else None

    def _preempt(self, req: Request, preempted: list[Request]) -> None:
        self.bm.free(req.block_table)
        req.block_table = []
        req.block_hashes = []
        req.num_computed = 0
        req.status = RequestStatus.WAITING
        self.running.remove(req)
        self._insert_waiting(req, retry=True)
        preempted.append(req)

    # ----------------------------------------------------------------
    def finish_step(
        self, output: SchedulerOutput, sampled: dict[str, int], now: Optional[float] = None
    ) -> list[Request]:
        """Commit a step: advance counters, append sampled tokens, seal
        full blocks into the prefix cache, finish/free requests.

        `sampled` maps request_id -> token for every seq with samples=True.
        Returns requests that finished this step.
        """
        if now is None:
            now = time.monotonic()
        finished: list[Request] = []
        bs = self.bm.block_size
        for ss in output.all_seqs:
            req = ss.req
            req.num_computed += ss.chunk_len
            if ss.samples:
                tok = sampled[req.request_id]
                if req.first_token_time is None:
                    req.first_token_time = now
                req.tokens.append(tok)
                self._maybe_finish(req, tok, now)
            # seal blocks fully computed (prefix cache)
            if self.enable_prefix_caching:
                while (len(req.block_hashes) + 1) * bs <= req.num_computed:
                    i = len(req.block_hashes)
                    parent = req.block_hashes[-1] if req.block_hashes else None
                    h = self.bm.seal_block(
                        req.block_table,
                        i,
                        tuple(req.tokens[i * bs : (i + 1) * bs]),
                        parent,
                        req.cache_salt,
                    )
                    req.block_hashes.append(h)
            if req.status.finished:
                finished.append(req)
        for req in finished:
            self.bm.free(req.block_table)
            req.block_table = []
            self.running.remove(req)
        return finished

    def _maybe_finish(self, req: Request, tok: int, now: float) -> None:
        p = req.params
        if not p.ignore_eos and tok in p.stop_token_ids:
            req.status = RequestStatus.FINISHED_STOPPED
        elif req.num_generated >= p.max_tokens:
            req.status = RequestStatus.FINISHED_LENGTH
        elif len(req.tokens) >= self.max_model_len:
            req.status = RequestStatus.FINISHED_LENGTH
        if req.status.finished:
            req.finish_time = now

    def _apply_aborts(self) -> None:
        if not self._aborted:
            return
        for req in list(self.running):
            if req.request_id in self._aborted:
                req.status = RequestStatus.FINISHED_ABORTED
                self.bm.free(req.block_table)
                req.block_table = []
                self.running.remove(req)
        self.waiting = deque(
            r for r in self.waiting if r.request_id not in self._aborted
        )
        self._aborted.clear()
