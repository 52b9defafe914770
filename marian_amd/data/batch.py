"""Batches (ref: src/data/batch.h, corpus_base.h SubBatch/CorpusBatch).

A SubBatch holds the padded word indices and mask for one stream (source or
target). The reference lays words out time-major ``[T, B]``; on MI355X the
natural layout for batched GEMM/attention is batch-major ``[B, T]`` —
we keep ``[B, T]`` tensors and provide the same statistics
(batch_width/batch_size/words).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch


class SubBatch:
    def __init__(self, indices: torch.Tensor, mask: torch.Tensor,
                 n_tokens: Optional[int] = None):
        # indices: [B, T] long; mask: [B, T] float (1 for real token incl. EOS)
        assert indices.shape == mask.shape
        self.indices = indices
        self.mask = mask
        # token count, computed host-side while the mask is still on CPU so
        # the training step never has to sync the device for it
        if n_tokens is None and not mask.is_cuda:
            n_tokens = int(mask.sum().item())
        self.n_tokens = n_tokens

    @classmethod
    def from_sentences(cls, sents: Sequence[Sequence[int]], pad_id: int = 0) -> "SubBatch":
        B = len(sents)
        lens = [len(s) for s in sents]
        T = max(lens, default=0)
        # one bulk list->tensor conversion (a per-row as_tensor/copy loop
        # measured 2.4x slower and bottlenecked real-corpus training)
        padded = [list(s) + [pad_id] * (T - l) if l != T else list(s)
                  for s, l in zip(sents, lens)]
        idx = (torch.tensor(padded, dtype=torch.long) if B else
               torch.zeros((0, 0), dtype=torch.long))
        lens_t = torch.tensor(lens, dtype=torch.long) if B else \
            torch.zeros(0, dtype=torch.long)
        mask = (torch.arange(T)[None, :] < lens_t[:, None]).float()
        return cls(idx, mask)

    @property
    def batch_size(self) -> int:
        return self.indices.shape[0]

    @property
    def batch_width(self) -> int:
        return self.indices.shape[1]

    def batch_words(self) -> int:
        if self.n_tokens is not None:
            return self.n_tokens
        return int(self.mask.sum().item())

    def to(self, device, non_blocking: bool = True) -> "SubBatch":
        return SubBatch(self.indices.to(device, non_blocking=non_blocking),
                        self.mask.to(device, non_blocking=non_blocking),
                        n_tokens=self.n_tokens)

    def pin_memory(self) -> "SubBatch":
        """Pin the host buffers so .to(cuda, non_blocking=True) is a true
        async DMA (pageable-source copies serialize the stream); called
        from the prefetch thread, off the training hot path."""
        if self.indices.is_cuda or self.indices.is_pinned():
            return self
        return SubBatch(self.indices.pin_memory(), self.mask.pin_memory(),
                        n_tokens=self.n_tokens)

    def narrow(self, start: int, size: int) -> "SubBatch":
        return SubBatch(self.indices[start:start + size].clone(),
                        self.mask[start:start + size].clone())


class CorpusBatch:
    """A tuple of SubBatches; index 0 = source, last = target."""

    def __init__(self, sub_batches: List[SubBatch],
                 guided_alignment: Optional[torch.Tensor] = None,
                 data_weights: Optional[torch.Tensor] = None,
                 sentence_ids: Optional[List[int]] = None):
        self.sub = sub_batches
        self.guided_alignment = guided_alignment
        self.data_weights = data_weights
        self.sentence_ids = sentence_ids or []

    @property
    def source(self) -> SubBatch:
        return self.sub[0]

    @property
    def target(self) -> SubBatch:
        return self.sub[-1]

    @target.setter
    def target(self, sb: SubBatch):
        """Models may rewrite the target stream (BERT masking transform)."""
        if len(self.sub) == 1:
            self.sub.append(sb)
        else:
            self.sub[-1] = sb

    def size(self) -> int:
        return self.sub[0].batch_size

    def words(self, stream: int = -1) -> int:
        return self.sub[stream].batch_words()

    def source_words(self) -> int:
        return self.sub[0].batch_words()

    def target_words(self) -> int:
        return self.sub[-1].batch_words()

    def pin_memory(self) -> "CorpusBatch":
        return CorpusBatch([sb.pin_memory() for sb in self.sub],
                           None if self.guided_alignment is None
                           else self.guided_alignment.pin_memory(),
                           None if self.data_weights is None
                           else self.data_weights.pin_memory(),
                           self.sentence_ids)

    def to(self, device, non_blocking: bool = True) -> "CorpusBatch":
        b = CorpusBatch([sb.to(device, non_blocking) for sb in self.sub],
                        None if self.guided_alignment is None
                        else self.guided_alignment.to(device),
                        None if self.data_weights is None
                        else self.data_weights.to(device),
                        self.sentence_ids)
        return b

    def split(self, n: int) -> List["CorpusBatch"]:
        """Split into up to n batches along the sentence dimension
        (ref: CorpusBatch::split used by SyncGraphGroup)."""
        B = self.size()
        n = min(n, B)
        out: List[CorpusBatch] = []
        base = B // n
        rem = B % n
        start = 0
        for i in range(n):
            size = base + (1 if i < rem else 0)
            if size == 0:
                continue
            subs = [sb.narrow(start, size) for sb in self.sub]
            ids = self.sentence_ids[start:start + size] if self.sentence_ids else []
            dw = None
            if self.data_weights is not None:
                dw = self.data_weights[start:start + size].clone()
            out.append(CorpusBatch(subs, None, dw, ids))
            start += size
        return out
