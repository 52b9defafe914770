"""BatchGenerator (ref: src/data/batch_generator.h).

Streams sentence tuples into padded CorpusBatches:
- maxi-batch pool of ``maxi_batch * mini_batch`` sentences, length-sorted
  (``--maxi-batch-sort src|trg|none``);
- sentence-based (``--mini-batch``) or token-based (``--mini-batch-words``)
  sizing;
- background prefetch thread (the reference fills the pool in a thread);
- deterministic under a fixed seed.
"""
from __future__ import annotations

import queue

import torch
import threading
from typing import Iterable, Iterator, List, Optional

from .batch import CorpusBatch, SubBatch
from .corpus import SentenceTuple


class BatchGenerator:
    def __init__(self, source: Iterable[SentenceTuple],
                 mini_batch: int = 64, mini_batch_words: int = 0,
                 batch_scale_fn=None,
                 maxi_batch: int = 100, maxi_batch_sort: str = "trg",
                 mini_batch_fit_table=None, prefetch: bool = True,
                 pad_ids: Optional[List[int]] = None):
        self.source = source
        self.mini_batch = mini_batch
        self.mini_batch_words = mini_batch_words
        # --mini-batch-warmup: callable -> multiplier in (0, 1] applied to
        # the sentence/token budget (ref: graph_group.cpp ramp-up logic)
        self.batch_scale_fn = batch_scale_fn
        self.maxi_batch = maxi_batch
        self.maxi_batch_sort = maxi_batch_sort
        self.fit_table = mini_batch_fit_table  # BatchStats when --mini-batch-fit
        self.prefetch = prefetch
        self.pad_ids = pad_ids

    # ------------------------------------------------------------------
    def _sort_key(self, tup: SentenceTuple):
        streams = tup[1]
        if self.maxi_batch_sort == "src":
            return (len(streams[0]), len(streams[-1]))
        return (len(streams[-1]), len(streams[0]))

    def _make_batch(self, tuples: List[SentenceTuple]) -> CorpusBatch:
        n_streams = len(tuples[0][1])
        subs = []
        for s in range(n_streams):
            pad = self.pad_ids[s] if self.pad_ids else 0
            subs.append(SubBatch.from_sentences([t[1][s] for t in tuples], pad_id=pad))
        ga = dw = None
        extras = [t[2] if len(t) > 2 else {} for t in tuples]
        if any("align" in e for e in extras):
            import torch
            from .alignment import alignments_to_tensor
            ga = alignments_to_tensor([e.get("align", []) for e in extras],
                                      subs[0].batch_width,
                                      subs[-1].batch_width)
        if any("weight" in e for e in extras):
            import torch
            ws = []
            T = subs[-1].batch_width
            per_word = any(isinstance(e.get("weight"), list) for e in extras)
            for e in extras:
                w = e.get("weight", 1.0)
                if per_word:
                    w = (w if isinstance(w, list) else [w] * T)
                    w = (w + [1.0] * T)[:T]
                ws.append(w)
            dw = torch.tensor(ws, dtype=torch.float32)
        return CorpusBatch(subs, guided_alignment=ga, data_weights=dw,
                           sentence_ids=[t[0] for t in tuples])

    def _emit_from_pool(self, pool: List[SentenceTuple]) -> Iterator[CorpusBatch]:
        if self.maxi_batch_sort != "none":
            pool.sort(key=self._sort_key)
        cur: List[SentenceTuple] = []
        max_trg = 0
        max_src = 0
        for tup in pool:
            t_len = len(tup[1][-1])
            s_len = len(tup[1][0])
            nm_trg = max(max_trg, t_len)
            nm_src = max(max_src, s_len)
            if cur:
                full = False
                scale = self.batch_scale_fn() if self.batch_scale_fn else 1.0
                if self.fit_table is not None:
                    cap = self.fit_table.find(nm_src, nm_trg)
                    full = len(cur) + 1 > max(1, int(cap * scale))
                elif self.mini_batch_words > 0:
                    full = (len(cur) + 1) * nm_trg > \
                        max(1, int(self.mini_batch_words * scale))
                else:
                    full = len(cur) + 1 > max(1, int(self.mini_batch * scale))
                if full:
                    yield self._make_batch(cur)
                    cur = []
                    nm_trg, nm_src = t_len, s_len
            cur.append(tup)
            max_trg, max_src = nm_trg, nm_src
        if cur:
            yield self._make_batch(cur)

    def _batches(self) -> Iterator[CorpusBatch]:
        pool: List[SentenceTuple] = []
        pool_cap = max(1, self.maxi_batch) * max(1, self.mini_batch)
        for tup in self.source:
            pool.append(tup)
            if len(pool) >= pool_cap:
                yield from self._emit_from_pool(pool)
                pool = []
        if pool:
            yield from self._emit_from_pool(pool)

    # ------------------------------------------------------------------
    def __iter__(self) -> Iterator[CorpusBatch]:
        if not self.prefetch:
            yield from self._batches()
            return
        # background prefetch thread, like the reference's pool-filler thread
        q: "queue.Queue" = queue.Queue(maxsize=16)
        SENTINEL = object()
        err: List[BaseException] = []

        pin = torch.cuda.is_available()

        def worker():
            try:
                for b in self._batches():
                    # pin in the prefetch thread: pageable-source H2D
                    # copies serialize the stream; pinned ones are async
                    q.put(b.pin_memory() if pin else b)
            except BaseException as e:  # propagate to consumer
                err.append(e)
            finally:
                q.put(SENTINEL)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is SENTINEL:
                break
            yield item
        t.join()
        if err:
            raise err[0]


class BatchStats:
    """--mini-batch-fit memory table (ref: src/data/batch_stats.h):
    (src_len, trg_len) -> max batch size that fits the workspace."""

    def __init__(self):
        self.map = {}  # (src_len, trg_len) -> batch size

    def add(self, src_len: int, trg_len: int, batch_size: int) -> None:
        self.map[(src_len, trg_len)] = batch_size

    def find(self, src_len: int, trg_len: int) -> int:
        """Batch size of the tightest entry covering the requested lengths
        (the reference's lower_bound lookup on the (length,length) grid)."""
        best_key = None
        best_b = None
        for (s, t), b in self.map.items():
            if s >= src_len and t >= trg_len:
                if best_key is None or (s, t) < best_key:
                    best_key, best_b = (s, t), b
        if best_b is None and self.map:
            best_b = min(self.map.values())  # longer than any entry: smallest
        return best_b or 1
