"""ForwardBatch — the token-packed batch descriptor handed to the model.

One engine step runs ONE forward over a packed token stream containing both
decode tokens (one per running sequence) and prefill-chunk tokens, vLLM-v1
style. Decode sequences come FIRST in the packing so the attention layer can
slice q[:n_decode_tokens] for the decode kernel and hand the rest to the
prefill kernel.
"""
from __future__ import annotations

import dataclasses

import torch


@dataclasses.dataclass
class ForwardBatch:
    # packed over all scheduled tokens [T]
    input_ids: torch.Tensor      # int32 [T]
    positions: torch.Tensor      # int32 [T]
    slot_mapping: torch.Tensor   # int64 [T]

    # decode part: sequences 0..n_decode-1, one token each
    n_decode: int
    decode_block_tables: torch.Tensor | None   # int32 [n_decode, max_blocks]
    decode_seq_lens: torch.Tensor | None       # int32 [n_decode]

    # prefill part: sequences n_decode.., packed after decode tokens
    n_prefill: int
    prefill_query_start_loc: torch.Tensor | None  # int32 [n_prefill+1], rel.
    prefill_seq_lens: torch.Tensor | None         # int32 [n_prefill]
    prefill_block_tables: torch.Tensor | None     # int32 [n_prefill, max_blocks]

    # indices (into the packed stream) of tokens whose logits are sampled
    logits_indices: torch.Tensor   # int64 [n_sample]

    # per-token LoRA adapter ids (None when no adapters are active)
    lora_ids: torch.Tensor | None = None  # int32 [T]

    # multimodal splice: rows of mm_embeds replace the token embeddings at
    # packed positions mm_idx (image-placeholder tokens; models/vision.py)
    mm_idx: torch.Tensor | None = None     # int64 [M]
    mm_embeds: torch.Tensor | None = None  # [M, hidden]

    @property
    def n_tokens(self) -> int:
        return self.input_ids.shape[0]
