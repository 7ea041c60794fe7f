"""Device-side batch metadata handed to the model forward."""

from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class ForwardBatch:
    is_prefill: bool
    input_ids: torch.Tensor  # [total_tokens] int64
    positions: torch.Tensor  # [total_tokens] int64
    slot_mapping: torch.Tensor  # [total_tokens] int64 (KV write slots)
    # prefill only:
    cu_seqlens: torch.Tensor | None = None  # [num_seqs+1] int32 (device)
    seq_lens_list: list[int] | None = None  # host copy for tile building
    # decode only:
    block_tables: torch.Tensor | None = None  # [num_seqs, max_blocks] int32
    seq_lens: torch.Tensor | None = None  # [num_seqs] int32 (device)
    # prefill-extend only: precomputed work-sorted (tiles64, tiles256) for
    # the paged extend kernels, shared by every layer of the forward
    ext_tiles: tuple | None = None
    # rows of the hidden states from which logits are needed (last token of
    # each sequence for prefill; everything for decode)
    logits_indices: torch.Tensor | None = None

    @property
    def num_tokens(self) -> int:
        return self.input_ids.shape[0]
