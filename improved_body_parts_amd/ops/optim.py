"""Host-side chunking for the fused multi-tensor SGD kernel."""
from __future__ import annotations

import torch

from ._backend import hip_extension

CHUNK = 1 << 16
_table_cache = {}


def _build_table(batch, device):
    rows = []
    for p, g, m, master in batch:
        numel = p.numel()
        off = 0
        pp, gp, mp = p.data_ptr(), g.data_ptr(), m.data_ptr()
        wp = master.data_ptr() if master is not None else 0
        # per-chunk dtype tag in the high half of the count word (a bf16
        # model still has fp32 BN affine parameters)
        dtag = 0 if p.dtype == torch.bfloat16 else 1
        while off < numel:
            cnt = min(CHUNK, numel - off)
            rows.append((pp, gp, mp, wp, off, cnt | (dtag << 32)))
            off += cnt
    t = torch.tensor(rows, dtype=torch.int64)
    if device.type == "cuda":
        # pinned staging: a pageable async H2D is a no-op under hipGraph
        # capture (the kernel then reads a dangling table pointer)
        t = t.pin_memory()
    return t.to(device, non_blocking=True)


def fused_sgd_step(batch, lr, momentum, weight_decay):
    """batch: list of (param, grad, momentum_fp32, master_fp32_or_None)."""
    if not batch:
        return
    ext = hip_extension()
    device = batch[0][0].device
    key = tuple(id(p) for p, _, _, _ in batch)
    entry = _table_cache.get(key)
    # freshness covers EVERY pointer the table embeds — params, grads,
    # momentum and master buffers (set_to_none grads move between steps;
    # a stale row is a dangling device pointer)
    ptrs = tuple(t.data_ptr() if t is not None else 0
                 for row in batch for t in row)
    if entry is None or entry[0] != ptrs:
        table = _build_table(batch, device)
        _table_cache[key] = (ptrs, table)
    else:
        table = entry[1]
    dtype_tag = 0 if batch[0][0].dtype == torch.bfloat16 else 1
    ext.fused_sgd(table, lr, momentum, weight_decay, dtype_tag)
    # the kernel writes parameters in place without going through ATen, so
    # autograd version counters never tick — bump them or version-keyed
    # caches (the conv weight packs) serve stale data forever
    for p, _, _, _ in batch:
        torch.autograd.graph.increment_version(p)
