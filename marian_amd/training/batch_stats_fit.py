"""--mini-batch-fit: measured (length x batch) -> memory fit table
(ref: src/training/graph_group.cpp collectStats + data/batch_stats.h).

The reference runs fake batches over a (length, size) grid against its
workspace arena high-water mark. Here the equivalent is
torch.cuda.max_memory_allocated() against the --workspace budget (or the
free HBM of the device; MI355X has 288 GB so the table typically saturates
at very large batches).
"""
from __future__ import annotations

import torch

from ..common.logging_utils import log_info
from ..data.batch_generator import BatchStats
from ..data.synthetic import synthetic_batch


def collect_stats(options, graph_group) -> BatchStats:
    device = graph_group.device
    if device.type != "cuda":
        raise RuntimeError("mini-batch-fit measurement needs a GPU")
    ws_mb = int(options.get("workspace", -1) or -1)
    if ws_mb > 0:
        budget = ws_mb * 1024 * 1024
    else:
        free, total = torch.cuda.mem_get_info(device)
        budget = int(free * 0.8)

    max_len = int(options.get("max-length", 50))
    step = int(options.get("mini-batch-fit-step", 10))
    vocab = int(options.list("dim-vocabs")[-1])
    stats = BatchStats()
    model = graph_group.model
    was_training = model.training
    model.train()

    lengths = list(range(step, max_len + 1, step)) or [max_len]
    start_batch = 64
    for L in lengths:
        lo, hi = 1, None
        B = start_batch
        # grow until OOM/budget, then bisect
        while True:
            if _fits(graph_group, B, L, vocab, budget):
                lo = B
                if hi is not None and hi - lo <= max(1, lo // 8):
                    break
                B = B * 2 if hi is None else (lo + hi) // 2
                if B > 2 ** 16:
                    lo = B
                    break
            else:
                hi = B
                if hi - lo <= max(1, lo // 8):
                    break
                B = (lo + hi) // 2
        stats.add(L, L, lo)
        start_batch = max(1, lo // 2)
        log_info("mini-batch-fit: length %d -> batch %d", L, lo)
    if not was_training:
        model.eval()
    graph_group.store.zero_grad()
    return stats


def _fits(graph_group, B: int, L: int, vocab: int, budget: int) -> bool:
    torch.cuda.reset_peak_memory_stats(graph_group.device)
    try:
        batch = synthetic_batch(B, L, L, vocab, device=graph_group.device)
        logits = graph_group.model(batch)
        lv = graph_group.criterion(logits, batch)
        lv.ce_sum.backward()
        graph_group.store.zero_grad()
        peak = torch.cuda.max_memory_allocated(graph_group.device)
        return peak <= budget
    except torch.cuda.OutOfMemoryError:
        graph_group.store.zero_grad()
        torch.cuda.empty_cache()
        return False
