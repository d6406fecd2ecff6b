"""ModelRunner — turns a SchedulerOutput into one packed GPU forward + sample.

Owns the model weights and the paged KV-cache tensors (sized against the
MI355X's 288 GB HBM via gpu_memory_utilization), builds the ForwardBatch
tensors, invokes the model, gathers sampling logits, and runs the sampling
kernels.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from kubeai_amd import ops
from kubeai_amd.models.config import ModelArchConfig
from kubeai_amd.models.llama import LlamaForCausalLM

from .batch import ForwardBatch
from .scheduler import SchedulerOutput


class _DecodeGraph:
    """hipGraph-captured pure-decode step for one padded batch size.

    Replays the whole token-packed forward (norms, GEMMs, rope, cache
    write, flash-decoding attention, logits GEMM) with zero launch
    overhead; only the small input buffers are refreshed per step.
    Sampling stays outside (its `step` scalar changes every call).
    """

    def __init__(self, runner: "ModelRunner", b_pad: int, bt_max: int):
        dev = runner.device
        self.b_pad = b_pad
        self.bt_max = bt_max
        self.input_ids = torch.zeros(b_pad, dtype=torch.int32, device=dev)
        self.positions = torch.zeros(b_pad, dtype=torch.int32, device=dev)
        self.slots = torch.full((b_pad,), -1, dtype=torch.int64, device=dev)
        self.block_tables = torch.zeros(
            (b_pad, bt_max), dtype=torch.int32, device=dev
        )
        self.seq_lens = torch.ones(b_pad, dtype=torch.int32, device=dev)
        logits_idx = torch.arange(b_pad, dtype=torch.int64, device=dev)
        self.fb = ForwardBatch(
            input_ids=self.input_ids,
            positions=self.positions,
            slot_mapping=self.slots,
            n_decode=b_pad,
            decode_block_tables=self.block_tables,
            decode_seq_lens=self.seq_lens,
            n_prefill=0,
            prefill_query_start_loc=None,
            prefill_seq_lens=None,
            prefill_block_tables=None,
            logits_indices=logits_idx,
        )
        # warmup outside capture (allocator, hipBLASLt workspace)
        torch.cuda.synchronize()
        for _ in range(2):
            h = runner.model(self.fb)
            runner.model.compute_logits(h[logits_idx])
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            h = runner.model(self.fb)
            self.logits = runner.model.compute_logits(h[logits_idx])

    def run(self, input_ids, positions, slots, tables, seq_lens) -> torch.Tensor:
        b = len(input_ids)
        self.input_ids[:b].copy_(input_ids, non_blocking=True)
        self.positions[:b].copy_(positions, non_blocking=True)
        self.slots[:b].copy_(slots, non_blocking=True)
        if b < self.b_pad:
            self.slots[b:].fill_(-1)
            self.seq_lens[b:].fill_(1)
            self.block_tables[b:].fill_(0)
        self.block_tables[:b, : tables.shape[1]].copy_(tables, non_blocking=True)
        if tables.shape[1] < self.bt_max:
            self.block_tables[:b, tables.shape[1] :].fill_(0)
        self.seq_lens[:b].copy_(seq_lens, non_blocking=True)
        self.graph.replay()
        return self.logits[:b]


class ModelRunner:
    def __init__(
        self,
        arch: ModelArchConfig,
        device: str = "cuda",
        dtype: torch.dtype = torch.bfloat16,
        block_size: int = 16,
        num_gpu_blocks: Optional[int] = None,
        gpu_memory_utilization: float = 0.90,
        seed: int = 0,
        tp_group=None,
        max_model_len: int = 8192,
        enable_graphs: bool = True,
        quantization: Optional[str] = None,
        model_path: Optional[str] = None,
        kv_cache_dtype: str = "auto",
    ):
        self.arch = arch
        self.device = torch.device(device)
        self.dtype = dtype
        self.block_size = block_size
        self.max_model_len = max_model_len
        self.tp_group = tp_group
        # TP decode graphs capture the in-graph all-reduce (one-shot xGMI
        # kernel or RCCL); capture failures are detected on device and
        # agreed across ranks (_execute_graph), falling back to eager —
        # so the gate is just the env switch.
        self.enable_graphs = (
            enable_graphs
            and self.device.type == "cuda"
            and os.environ.get("KUBEAI_GRAPHS", "1") == "1"
            and (tp_group is None or os.environ.get("KUBEAI_TP_GRAPHS", "1") == "1")
        )
        self._graphs: dict[int, _DecodeGraph] = {}
        # JSON-mode constrained decoding: per-request prefix automata +
        # a tokenizer the server/worker injects after startup
        self.tokenizer = None
        self._json_validators: dict = {}
        self._sampling_cache: dict = {}
        torch.manual_seed(seed)
        if tp_group is not None:
            from kubeai_amd.parallel.tp import TPLlamaForCausalLM

            if arch.post_norms:
                raise NotImplementedError(
                    "tensor parallelism is not implemented for the gemma2 "
                    "layer structure (post-norms); run gemma2 replicated"
                )
            self.model = TPLlamaForCausalLM(
                arch, tp_group, device=self.device, dtype=dtype, seed=seed
            )
            self.n_kv_local = self.model.n_kv_local
        else:
            self.model = LlamaForCausalLM(arch, device=self.device, dtype=dtype)
            self.n_kv_local = arch.num_key_value_heads
        self.model.eval()
        # multimodal: the vision tower is replicated on every rank (it
        # runs once per image at prefill; TP ranks compute identical
        # embeddings deterministically instead of broadcasting them)
        self.vision = None
        if arch.vision is not None:
            from kubeai_amd.models.vision import VisionTower

            # independent seed: the TP model's construction leaves the
            # global RNG in a world-size-dependent state; the tower must
            # init identically on every rank AND across TP degrees
            torch.manual_seed(seed + 7919)
            self.vision = VisionTower(arch, device=self.device, dtype=dtype)
            self.vision.eval()
            torch.manual_seed(seed)
        if model_path is not None and os.path.isdir(model_path):
            if tp_group is not None:
                from kubeai_amd.models.loader import load_weights_tp

                load_weights_tp(self.model, model_path, vision=self.vision)
            else:
                from kubeai_amd.models.loader import load_weights

                load_weights(self.model, model_path, vision=self.vision)
        self.quantization = quantization
        if quantization == "fp8":
            # on CPU the Fp8Linear dequant fallback keeps the same
            # semantics (tests); the fast _scaled_mm path needs the GPU
            from .quant import convert_to_fp8

            n = convert_to_fp8(self.model)
            assert n > 0, "no linear layers converted to fp8"

        if kv_cache_dtype == "fp8_e5m2":
            # halves decode KV reads and doubles block capacity; attention
            # kernels dequantize in-register while staging (e5m2 range
            # covers K/V without scale bookkeeping — vLLM's fp8_e5m2 analog)
            self.kv_dtype = torch.float8_e5m2
        elif kv_cache_dtype in ("auto", None):
            self.kv_dtype = dtype
        else:
            raise ValueError(f"unknown kv_cache_dtype {kv_cache_dtype!r}")
        if num_gpu_blocks is None:
            num_gpu_blocks = self._profile_num_blocks(gpu_memory_utilization)
        self.num_blocks = num_gpu_blocks
        kv_shape = (
            num_gpu_blocks,
            self.n_kv_local,
            block_size,
            arch.head_dim,
        )
        self.kv_caches = [
            (
                torch.zeros(kv_shape, dtype=self.kv_dtype, device=self.device),
                torch.zeros(kv_shape, dtype=self.kv_dtype, device=self.device),
            )
            for _ in range(arch.num_hidden_layers)
        ]
        self.model.bind_kv_caches(self.kv_caches)

    def _profile_num_blocks(self, gpu_memory_utilization: float) -> int:
        per_block_bytes = (
            2  # k + v
            * self.n_kv_local
            * self.block_size
            * self.arch.head_dim
            * self.kv_dtype.itemsize
            * self.arch.num_hidden_layers
        )
        if self.device.type == "cuda":
            free, total = torch.cuda.mem_get_info(self.device)
            budget = total * gpu_memory_utilization - (total - free)
            # reserve activation headroom: 4% of total
            budget -= 0.04 * total
            n = int(budget // per_block_bytes)
            if n < 16:
                raise RuntimeError(
                    f"not enough GPU memory for KV cache: budget={budget/1e9:.1f}GB"
                )
            return n
        return 512  # CPU tests

    # ------------------------------------------------------------------
    # LoRA (full SGMV path in kubeai_amd/engine/lora.py; ids are also the
    # KV-cache salt so adapter outputs never share prefix blocks)
    def load_lora(self, lora_id: int, path) -> None:
        from .lora import LoRAManager

        if not hasattr(self, "lora_manager"):
            self.lora_manager = LoRAManager(self.model, self.device, self.dtype)
        self.lora_manager.load(lora_id, path)

    def unload_lora(self, lora_id: int) -> None:
        if hasattr(self, "lora_manager"):
            self.lora_manager.unload(lora_id)

    # ------------------------------------------------------------------
    def build_batch(self, out: SchedulerOutput) -> ForwardBatch:
        bs = self.block_size
        # vision embeddings: computed once per request, at its first
        # scheduled prefill chunk
        if self.vision is not None:
            for ss in out.prefill:
                req = ss.req
                if req.images and req.mm_embeds is None:
                    pixels = torch.stack([im.float() for im in req.images])
                    req.mm_embeds = self.vision.encode(pixels)
        input_ids: list[int] = []
        positions: list[int] = []
        slots: list[int] = []

        def emit(req, start: int, length: int) -> None:
            if length > 64:
                # vectorize long prefill chunks (a pure-Python per-token loop
                # costs ~1 ms per 8k-token chunk)
                import numpy as np

                idx = np.arange(start, start + length)
                input_ids.extend(req.tokens[start : start + length])
                positions.extend(idx.tolist())
                table = np.asarray(req.block_table, dtype=np.int64)
                slots.extend((table[idx // bs] * bs + idx % bs).tolist())
                return
            for i in range(start, start + length):
                input_ids.append(req.tokens[i])
                positions.append(i)
                table_idx = i // bs
                slots.append(req.block_table[table_idx] * bs + i % bs)

        max_bt = 1
        for ss in out.all_seqs:
            max_bt = max(max_bt, len(ss.req.block_table))

        dec_tables: list[list[int]] = []
        dec_lens: list[int] = []
        for ss in out.decode:
            emit(ss.req, ss.chunk_start, 1)
            dec_tables.append(self._pad(ss.req.block_table, max_bt))
            dec_lens.append(ss.chunk_start + 1)

        pre_tables: list[list[int]] = []
        pre_lens: list[int] = []
        mm_idx: list[int] = []
        mm_rows: list[torch.Tensor] = []
        nd_tok = len(input_ids)  # decode tokens precede prefill tokens
        qsl = [0]
        for ss in out.prefill:
            emit(ss.req, ss.chunk_start, ss.chunk_len)
            pre_tables.append(self._pad(ss.req.block_table, max_bt))
            pre_lens.append(ss.chunk_start + ss.chunk_len)
            # image-placeholder positions inside this chunk -> splice rows
            if ss.req.mm_embeds is not None:
                c0, c1 = ss.chunk_start, ss.chunk_start + ss.chunk_len
                flat_base = nd_tok + qsl[-1]
                for s0, n, row0 in ss.req.mm_spans:
                    lo, hi = max(s0, c0), min(s0 + n, c1)
                    if lo < hi:
                        mm_idx.extend(
                            range(flat_base + lo - c0, flat_base + hi - c0)
                        )
                        mm_rows.append(
                            ss.req.mm_embeds[row0 + lo - s0 : row0 + hi - s0]
                        )
            qsl.append(qsl[-1] + ss.chunk_len)

        # logits rows to sample: decode seq i -> packed index i; prefill seq j
        # samples its last chunk token if the chunk completes the prompt
        logit_idx: list[int] = []
        nd = len(out.decode)
        for i, ss in enumerate(out.decode):
            logit_idx.append(i)
        for j, ss in enumerate(out.prefill):
            if ss.samples:
                logit_idx.append(nd + qsl[j + 1] - 1)

        lora_ids = None
        if getattr(self, "lora_manager", None) is not None and self.lora_manager.active:
            ids: list[int] = []
            for ss in out.all_seqs:
                ids.extend([ss.req.lora_id] * ss.chunk_len)
            lora_ids = torch.tensor(ids, dtype=torch.int32, device=self.device)

        # ---- pad the token stream to a bucket size --------------------
        # Keeps GEMM M-shapes on a small fixed set (TunableOp algo table +
        # future hipGraph capture). Pad tokens form a throwaway prefill
        # sequence over cache block 0: slot -1 (no KV write), row-local
        # compute only (GEMMs/norms are row-independent), never sampled.
        T = len(input_ids)
        T_pad = _bucket(T)
        if T_pad > T:
            pad = T_pad - T
            input_ids.extend([0] * pad)
            slots.extend([-1] * pad)
            # split padding into <=block_size mini-seqs so the throwaway
            # causal attention stays O(pad * bs), not O(pad^2)
            left = pad
            while left > 0:
                c = min(left, bs)
                positions.extend(range(c))
                pre_tables.append([0] * max_bt)
                pre_lens.append(c)
                qsl.append(qsl[-1] + c)
                left -= c
            if lora_ids is not None:
                lora_ids = torch.cat(
                    [lora_ids, torch.zeros(pad, dtype=torch.int32, device=self.device)]
                )

        dev = self.device
        t32 = lambda x: torch.tensor(x, dtype=torch.int32, device=dev)
        n_prefill = len(pre_lens)
        return ForwardBatch(
            mm_idx=(
                torch.tensor(mm_idx, dtype=torch.int64, device=dev)
                if mm_idx
                else None
            ),
            mm_embeds=torch.cat(mm_rows).to(dev) if mm_rows else None,
            lora_ids=lora_ids,
            input_ids=t32(input_ids),
            positions=t32(positions),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            n_decode=nd,
            decode_block_tables=t32(dec_tables) if dec_tables else None,
            decode_seq_lens=t32(dec_lens) if dec_lens else None,
            n_prefill=n_prefill,
            prefill_query_start_loc=t32(qsl) if n_prefill else None,
            prefill_seq_lens=t32(pre_lens) if pre_lens else None,
            prefill_block_tables=t32(pre_tables) if pre_tables else None,
            logits_indices=torch.tensor(logit_idx, dtype=torch.int64, device=dev),
        )

    @staticmethod
    def _pad(table: list[int], n: int) -> list[int]:
        return table + [0] * (n - len(table))

    # ------------------------------------------------------------------
    def _execute_graph(self, out: SchedulerOutput):
        """Pure-decode step through a captured hipGraph (bucketed batch)."""
        b = len(out.decode)
        b_pad = _bucket(b)
        bt_max = (self.max_model_len + self.block_size - 1) // self.block_size
        g = self._graphs.get(b_pad)
        if g is None:
            ok = True
            try:
                g = _DecodeGraph(self, b_pad, bt_max)
            except Exception:
                import traceback

                traceback.print_exc()
                ok = False
            if self.tp_group is not None and self.tp_group.world > 1:
                # ranks run in lockstep, so every rank attempts capture on
                # the same step; all must agree or none may use the graph
                # (a lone eager rank would mismatch in-graph collectives)
                import torch.distributed as dist

                flag = torch.tensor(
                    [1 if ok else 0], dtype=torch.int32, device=self.device
                )
                dist.all_reduce(
                    flag, op=dist.ReduceOp.MIN, group=self.tp_group.group
                )
                ok = ok and int(flag.item()) == 1
            if not ok:
                self.enable_graphs = False
                return None
            self._graphs[b_pad] = g
        bs = self.block_size
        ids, pos, slots, lens = [], [], [], []
        import numpy as np

        tables = np.zeros((b, g.bt_max), dtype=np.int32)
        for i, ss in enumerate(out.decode):
            req = ss.req
            t = ss.chunk_start
            ids.append(req.tokens[t])
            pos.append(t)
            slots.append(req.block_table[t // bs] * bs + t % bs)
            lens.append(t + 1)
            bt = req.block_table
            tables[i, : len(bt)] = bt
        return g.run(
            torch.tensor(ids, dtype=torch.int32),
            torch.tensor(pos, dtype=torch.int32),
            torch.tensor(slots, dtype=torch.int64),
            torch.from_numpy(tables),
            torch.tensor(lens, dtype=torch.int32),
        )

    @torch.inference_mode()
    def execute(self, out: SchedulerOutput, step: int) -> dict[str, int]:
        """Run one forward + sampling; returns request_id -> sampled token."""
        logits = None
        if (
            self.enable_graphs
            and not out.prefill
            and out.decode
            and not (
                getattr(self, "lora_manager", None) is not None
                and self.lora_manager.active
            )
        ):
            logits = self._execute_graph(out)
        if logits is None:
            fb = self.build_batch(out)
            hidden = self.model(fb)  # [T, H]
            if fb.logits_indices.numel() == 0:
                return {}
            logits = self.model.compute_logits(hidden[fb.logits_indices])  # [S, V]

        # assemble sampling params in the same order as logits rows
        sample_reqs = [ss.req for ss in out.decode] + [
            ss.req for ss in out.prefill if ss.samples
        ]
        if logits.shape[0] != len(sample_reqs):
            raise RuntimeError(
                f"logits rows {logits.shape[0]} != sample reqs {len(sample_reqs)}"
            )
        # OpenAI presence/frequency penalties over generated tokens:
        # persistent per-request GPU count tensors (updated incrementally
        # after sampling, freed on finish) + ONE batched penalty kernel —
        # no per-row Python loop or per-step re-upload of output ids
        V = logits.shape[1]
        dev_l = logits.device
        pen_rows = [
            (i, r)
            for i, r in enumerate(sample_reqs)
            if (r.params.presence_penalty or r.params.frequency_penalty)
            and r.num_generated > 0
        ]
        if pen_rows:
            counts = torch.stack([self._pen_counts_for(r, V) for _, r in pen_rows])
            idx = torch.tensor([i for i, _ in pen_rows], device=dev_l)
            fp_t = torch.tensor(
                [r.params.frequency_penalty for _, r in pen_rows],
                dtype=torch.float32, device=dev_l,
            ).unsqueeze(1)
            pp_t = torch.tensor(
                [r.params.presence_penalty for _, r in pen_rows],
                dtype=torch.float32, device=dev_l,
            ).unsqueeze(1)
            logits[idx] -= (fp_t * counts + pp_t * (counts > 0)).to(logits.dtype)
        # logit_bias: flat-indexed single scatter over all rows (ids
        # validated at parse time; filtered again so a bad id can never
        # kill the step loop)
        rows, cols, vals = [], [], []
        for i, r in enumerate(sample_reqs):
            if r.params.logit_bias:
                for t, b in r.params.logit_bias.items():
                    if 0 <= t < V:
                        rows.append(i)
                        cols.append(t)
                        vals.append(b)
        if rows:
            flat = torch.tensor(
                [ri * V + ci for ri, ci in zip(rows, cols)],
                dtype=torch.int64, device=dev_l,
            )
            logits.view(-1).scatter_add_(
                0, flat, torch.tensor(vals, dtype=logits.dtype, device=dev_l)
            )
        temps = [r.params.temperature for r in sample_reqs]
        if all(t <= 0.0 for t in temps):
            tokens = ops.greedy_sample(logits.contiguous())
        else:
            needs_topk = any(
                (r.params.top_p < 1.0 or r.params.top_k > 0)
                and r.params.temperature > 0
                for r in sample_reqs
            )
            dev = self.device
            t_t = torch.tensor(temps, dtype=torch.float32, device=dev)
            seeds = torch.tensor(
                [
                    r.params.seed if r.params.seed is not None else _seed_of(r.request_id)
                    for r in sample_reqs
                ],
                dtype=torch.int64,
                device=dev,
            )
            if needs_topk:
                # rejection nucleus sampling — no full-vocab sort; the
                # per-row param tensors are cached across steps (the
                # batch's sampling params rarely change step to step)
                tokens = _sample_topk_topp(
                    logits,
                    [r.params.top_p for r in sample_reqs],
                    [r.params.top_k for r in sample_reqs],
                    [r.params.temperature for r in sample_reqs],
                    t_t, seeds, step,
                    cache=self._sampling_cache,
                ).to(dev)
            else:
                tokens = ops.gumbel_sample(logits.contiguous(), t_t, seeds, step)
        tokens = tokens.cpu().tolist()
        if any(r.params.json_mode for r in sample_reqs):
            tokens = self._constrain_json(sample_reqs, tokens, logits)
        tokens_t = torch.tensor(tokens, dtype=torch.int64, device=logits.device)
        # logprob of the chosen token: logit - logsumexp(row)
        lse = torch.logsumexp(logits, dim=-1)
        chosen = logits.gather(1, tokens_t.view(-1, 1)).squeeze(1)
        logprobs = (chosen - lse).cpu().tolist()
        self.last_logprobs = {
            r.request_id: float(lp) for r, lp in zip(sample_reqs, logprobs)
        }
        # top-alternative logprobs, only when some request asked for them
        self.last_top_logprobs = {}
        want = max((r.params.logprobs for r in sample_reqs), default=0)
        if want > 0:
            k = min(want, logits.shape[-1])
            tv, ti = (logits - lse.unsqueeze(1)).topk(k, dim=-1)
            tv, ti = tv.cpu().tolist(), ti.cpu().tolist()
            for r, vs, ids in zip(sample_reqs, tv, ti):
                if r.params.logprobs > 0:
                    n = r.params.logprobs
                    self.last_top_logprobs[r.request_id] = [
                        (int(i), float(v)) for i, v in zip(ids[:n], vs[:n])
                    ]
        # keep persistent penalty counts current with this step's samples
        upd = [
            (r, t) for r, t in zip(sample_reqs, tokens)
            if (r.params.presence_penalty or r.params.frequency_penalty)
        ]
        if upd:
            for r, t in upd:
                c = self._pen_counts_for(r, logits.shape[1])
                c[int(t)] += 1.0
        return {r.request_id: int(t) for r, t in zip(sample_reqs, tokens)}

    # ------------------------------------------------------------------
    def _constrain_json(self, sample_reqs, tokens: list, logits) -> list:
        """OpenAI response_format json_object: vet each sampled token's
        decoded text against a streaming JSON-prefix automaton; mask and
        resample (argmax over survivors) on violation, hold EOS until the
        top-level object closes, force-stop once it has. Runs identically
        on every TP rank (the worker injects the same tokenizer)."""
        tok = self.tokenizer
        if tok is None:
            return tokens  # no tokenizer injected (bare-engine use)
        from .jsonmode import JsonPrefixValidator, SchemaValidator

        eos = self.arch.eos_token_id
        for i, r in enumerate(sample_reqs):
            if not r.params.json_mode:
                continue
            v = self._json_validators.get(r.request_id)
            if v is None:
                v = self._json_validators[r.request_id] = (
                    SchemaValidator(r.params.json_schema)
                    if r.params.json_schema
                    else JsonPrefixValidator()
                )
            if v.complete:
                tokens[i] = eos  # the object closed on a previous step
                continue
            tail = r.output_token_ids[-8:]
            prev = tok.decode(tail)
            stop_ids = set(r.params.stop_token_ids) | {eos}
            snap = v.snapshot()

            def ok(t: int) -> bool:
                if t in stop_ids:
                    return v.complete
                piece = tok.decode(tail + [t])[len(prev):]
                if v.feed(piece):
                    return True
                v.restore(snap)
                return False

            if ok(tokens[i]):
                continue
            # sampled token breaks JSON: walk candidates in logit order
            # (valid tokens can be a tiny vocab fraction, e.g. only '{'
            # opens a document — plain resampling would almost never hit)
            V = logits.shape[1]
            cand = torch.topk(logits[i], min(V, 4096)).indices.cpu().tolist()
            chosen = None
            for t in cand:
                if ok(t):
                    chosen = t
                    break
            if chosen is None:
                # last resort: tokens whose text starts with an allowed
                # char (first-char buckets built once per tokenizer)
                for t in self._json_char_candidates(v, snap, tok):
                    if ok(t):
                        chosen = t
                        break
            if chosen is not None:
                tokens[i] = int(chosen)
            else:
                # nothing representable continues this JSON (schema dead
                # end) — end the request rather than emit garbage; the
                # client sees the longest valid prefix
                tokens[i] = eos
        return tokens

    def _json_char_candidates(self, v, snap, tok):
        buckets = getattr(self, "_json_first_char", None)
        if buckets is None:
            buckets = {}
            for t in range(self.arch.vocab_size):
                s = tok.decode([t])
                if s:
                    buckets.setdefault(s[0], []).append(t)
            self._json_first_char = buckets
        out = []
        for ch, ids in buckets.items():
            if v.feed(ch):
                out.extend(ids)
            v.restore(snap)
        return out

    # ------------------------------------------------------------------
    def _pen_counts_for(self, r, vocab: int) -> torch.Tensor:
        """Per-request generated-token counts on device (penalty state)."""
        counts = getattr(self, "_pen_counts", None)
        if counts is None:
            counts = self._pen_counts = {}
        c = counts.get(r.request_id)
        if c is None:
            c = torch.zeros(vocab, dtype=torch.float32, device=self.device)
            if r.output_token_ids:
                out_t = torch.tensor(
                    r.output_token_ids, dtype=torch.int64, device=self.device
                )
                c.scatter_add_(
                    0, out_t, torch.ones_like(out_t, dtype=torch.float32)
                )
            counts[r.request_id] = c
        return c

    def release_request(self, request_id: str) -> None:
        """Drop per-request runner state (called when a request ends)."""
        if getattr(self, "_pen_counts", None):
            self._pen_counts.pop(request_id, None)
        self._json_validators.pop(request_id, None)


def _sample_topk_topp(logits, top_ps, top_ks, temps, t_t, seeds, step,
                      cache=None):
    """Nucleus/top-k sampling by rejection — no full-vocab sort.

    Sampling from the renormalized nucleus == sampling the FULL
    distribution and accepting iff the drawn token lies in the nucleus.
    Membership is one reduction, not a sort: token t is kept iff the
    probability mass STRICTLY above it (`mass_above`) is < top_p, and
    (for top-k) fewer than k logits exceed it. Expected attempts = 1 /
    P(nucleus) <= 1/top_p regardless of how flat the distribution is —
    the case where the cumsum-prefix trick degenerates (a near-uniform
    128k-vocab nucleus at p=0.9 spans ~30k tokens). Rows that keep
    rejecting (tiny top_k on a flat distribution) fall back to the exact
    sort-mask path after a few rounds. Greedy rows (temp<=0) accept
    immediately (argmax is always in the nucleus).
    """
    S, V = logits.shape
    dev = logits.device
    key = (tuple(top_ps), tuple(top_ks), tuple(temps))
    hit = cache.get("params") if cache is not None else None
    if hit is not None and hit[0] == key:
        tp, tk, tt = hit[1]
    else:
        tp = torch.tensor(top_ps, device=dev, dtype=torch.float32)
        tk = torch.tensor(top_ks, device=dev, dtype=torch.int32)
        tt = torch.tensor(temps, device=dev, dtype=torch.float32)
        if cache is not None:
            cache["params"] = (key, (tp, tk, tt))
    logits_c = logits.float().contiguous()
    # per-row softmax stats once; each draw is then ONE fused pass
    m, z = ops.nucleus_stats(logits_c, tt)

    def draw(attempt: int):
        # fresh deterministic noise per attempt
        cand = ops.gumbel_sample(
            logits_c, t_t, seeds, step + (attempt + 1) * 1_000_003
        ).to(dev)
        ok = ops.nucleus_accept(
            logits_c, cand, m, z, tt, tp, tk
        ).to(torch.bool)
        return cand, ok

    # three sync-free draws; P(all miss) <= (1-p)^3 per row, so the
    # host-synced fallback check fires rarely
    c1, ok1 = draw(0)
    c2, ok2 = draw(1)
    c3, ok3 = draw(2)
    tokens = torch.where(ok1, c1, torch.where(ok2, c2, c3))
    pending = ~(ok1 | ok2 | ok3)
    if cache is not None:
        cache["calls"] = cache.get("calls", 0) + 1
    if bool(pending.any()):
        # a few more cheap draws before the expensive exact path — a
        # fallback round costs ~3 fused passes vs ~40 for the sort-mask
        for attempt in range(3, 6):
            c, ok = draw(attempt)
            tokens = torch.where(pending & ok, c, tokens)
            pending &= ~ok
        if bool(pending.any()):
            if cache is not None:
                cache["fallbacks"] = cache.get("fallbacks", 0) + 1
            # stragglers: exact sort-mask path (tiny top_k on a flat
            # distribution can reject indefinitely)
            masked = _apply_topk_topp(logits, top_ps, top_ks, temps)
            rest = ops.gumbel_sample(masked.contiguous(), t_t, seeds, step).to(dev)
            tokens = torch.where(pending, rest, tokens)
    return tokens


def _apply_topk_topp(logits, top_ps, top_ks, temps):
    """Mask logits outside the per-row top-k / top-p (nucleus) sets.

    Fully batched on-device, and the common case avoids the full-vocab
    sort: torch.topk over the leading K=1024 candidates plus a full-row
    logsumexp gives EXACT nucleus masks whenever the nucleus fits in K
    (checked per row; the rare uncovered row falls back to a full sort).
    Sampling from the masked logits via Gumbel-argmax is exactly nucleus
    sampling: probabilities renormalize implicitly. Greedy rows
    (temp<=0) pass through unmasked.
    """
    S, V = logits.shape
    dev = logits.device
    tp = torch.tensor(top_ps, device=dev, dtype=torch.float32).unsqueeze(1)
    tk = torch.tensor(top_ks, device=dev, dtype=torch.int64).unsqueeze(1)
    tt = torch.tensor(temps, device=dev, dtype=torch.float32).unsqueeze(1)
    max_k = max((k for k in top_ks if k and k > 0), default=0)
    K = min(V, max(1024, max_k))
    if K >= V:
        vals, idx = logits.sort(dim=-1, descending=True)
    else:
        vals, idx = logits.topk(K, dim=-1)
    pos = torch.arange(K if K < V else V, device=dev).unsqueeze(0)
    keep = (tk <= 0) | (pos < tk)                    # top-k
    # exact full-distribution probabilities for the leading K entries
    scaled = logits.float() / tt.clamp_min(1e-6)
    lse = torch.logsumexp(scaled, dim=-1, keepdim=True)
    probs = torch.exp(vals.float() / tt.clamp_min(1e-6) - lse)
    csum = probs.cumsum(-1)
    keep &= (csum - probs < tp) | (pos == 0)         # top-p, rank-0 safe
    keep |= tt <= 0                                  # greedy rows untouched
    if K < V:
        # nucleus spilling past K (top-K mass < top_p with no tighter
        # top-k)? those rows get the exact full-sort treatment
        uncovered = (
            (csum[:, -1:] < tp) & ((tk <= 0) | (tk > K)) & (tt > 0)
        ).squeeze(1)
        if bool(uncovered.any()):
            sub_lg = logits[uncovered]
            sv, si = sub_lg.sort(dim=-1, descending=True)
            sub_tt = tt[uncovered]
            sub_tp = tp[uncovered]
            sub_tk = tk[uncovered]
            posV = torch.arange(V, device=dev).unsqueeze(0)
            skeep = (sub_tk <= 0) | (posV < sub_tk)
            sprobs = torch.softmax(sv.float() / sub_tt.clamp_min(1e-6), -1)
            scs = sprobs.cumsum(-1)
            skeep &= (scs - sprobs < sub_tp) | (posV == 0)
            skeep |= sub_tt <= 0
            smask = torch.zeros_like(skeep)
            smask.scatter_(1, si, skeep)
            out = logits.clone()
            mask = torch.zeros_like(logits, dtype=torch.bool)
            mask.scatter_(1, idx, keep)
            mask[uncovered] = smask
            mask |= tt <= 0  # greedy rows keep the whole vocab
            return out.masked_fill_(~mask, float("-inf"))
    mask = torch.zeros_like(logits, dtype=torch.bool)
    mask.scatter_(1, idx, keep)
    mask |= tt <= 0  # greedy rows keep the whole vocab
    return logits.masked_fill(~mask, float("-inf"))


_BUCKETS = [8, 16, 24, 32, 40, 48, 64, 80, 96, 128, 160, 192, 256, 320,
            384, 512, 640, 768, 1024, 1280, 1536, 2048, 3072, 4096, 6144,
            8192, 12288, 16384]


def _bucket(n: int) -> int:
    for b in _BUCKETS:
        if n <= b:
            return b
    return n


def _seed_of(request_id: str) -> int:
    import zlib

    return zlib.crc32(request_id.encode())
