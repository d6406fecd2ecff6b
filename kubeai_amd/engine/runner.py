"""ModelRunner — turns a SchedulerOutput into one packed GPU forward + sample.

Owns the model weights and the paged KV-cache tensors (sized against the
MI355X's 288 GB HBM via gpu_memory_utilization), builds the ForwardBatch
tensors, invokes the model, gathers sampling logits, and runs the sampling
kernels.
"""
from __future__ import annotations

from typing import Optional

import torch

from kubeai_amd import ops
from kubeai_amd.models.config import ModelArchConfig
from kubeai_amd.models.llama import LlamaForCausalLM

from .batch import ForwardBatch
from .kvcache import BlockManager
from .scheduler import SchedulerOutput


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
    ):
        self.arch = arch
        self.device = torch.device(device)
        self.dtype = dtype
        self.block_size = block_size
        torch.manual_seed(seed)
        if tp_group is not None:
            from kubeai_amd.parallel.tp import TPLlamaForCausalLM

            self.model = TPLlamaForCausalLM(
                arch, tp_group, device=self.device, dtype=dtype, seed=seed
            )
            self.n_kv_local = self.model.n_kv_local
        else:
            self.model = LlamaForCausalLM(arch, device=self.device, dtype=dtype)
            self.n_kv_local = arch.num_key_value_heads
        self.model.eval()

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
                torch.zeros(kv_shape, dtype=dtype, device=self.device),
                torch.zeros(kv_shape, dtype=dtype, device=self.device),
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
            * self.dtype.itemsize
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
        input_ids: list[int] = []
        positions: list[int] = []
        slots: list[int] = []

        def emit(req, start: int, length: int) -> None:
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
        qsl = [0]
        for ss in out.prefill:
            emit(ss.req, ss.chunk_start, ss.chunk_len)
            pre_tables.append(self._pad(ss.req.block_table, max_bt))
            pre_lens.append(ss.chunk_start + ss.chunk_len)
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
    @torch.inference_mode()
    def execute(self, out: SchedulerOutput, step: int) -> dict[str, int]:
        """Run one forward + sampling; returns request_id -> sampled token."""
        fb = self.build_batch(out)
        hidden = self.model(fb)  # [T, H]
        if fb.logits_indices.numel() == 0:
            return {}
        logits = self.model.compute_logits(hidden[fb.logits_indices])  # [S, V] f32

        # assemble sampling params in the same order as logits rows
        sample_reqs = [ss.req for ss in out.decode] + [
            ss.req for ss in out.prefill if ss.samples
        ]
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
                from kubeai_amd.ops import ref as ops_ref

                top_p = torch.tensor(
                    [r.params.top_p for r in sample_reqs], dtype=torch.float32
                )
                top_k = torch.tensor(
                    [r.params.top_k for r in sample_reqs], dtype=torch.int64
                )
                tokens = ops_ref.topk_topp_sample(
                    logits.cpu(), t_t.cpu(), top_p, top_k, seeds.cpu(), step
                ).to(dev)
            else:
                tokens = ops.gumbel_sample(logits.contiguous(), t_t, seeds, step)
        tokens = tokens.cpu().tolist()
        return {r.request_id: int(t) for r, t in zip(sample_reqs, tokens)}


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
