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
        while off < numel:
            cnt = min(CHUNK, numel - off)
            rows.append((pp, gp, mp, wp, off, cnt))
            off += cnt
    t = torch.tensor(rows, dtype=torch.int64)
    return t.to(device, non_blocking=True)


def fused_sgd_step(batch, lr, momentum, weight_decay):
    """batch: list of (param, grad, momentum_fp32, master_fp32_or_None)."""
    if not batch:
        return
    ext = hip_extension()
    device = batch[0][0].device
    key = tuple(id(p) for p, _, _, _ in batch) + \
        tuple(b[1].data_ptr() for b in batch[:1])
    entry = _table_cache.get(key)
    ptrs = tuple(b[0].data_ptr() for b in batch) + tuple(b[1].data_ptr() for b in batch)
    if entry is None or entry[0] != ptrs:
        table = _build_table(batch, device)
        _table_cache[key] = (ptrs, table)
    else:
        table = entry[1]
    dtype_tag = 0 if batch[0][0].dtype == torch.bfloat16 else 1
    ext.fused_sgd(table, lr, momentum, weight_decay, dtype_tag)
