"""Sorted-merge join on device (u64 keys, u64 payloads).

Both inputs must be sorted by key (use radix.sort_pairs). Inner join:
returns (keys, a_payloads, b_payloads) of all matching pairs.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import load


def merge_join_sorted(a_keys: torch.Tensor, a_vals: Optional[torch.Tensor],
                      b_keys: torch.Tensor, b_vals: torch.Tensor
                      ) -> Tuple[torch.Tensor, Optional[torch.Tensor], torch.Tensor]:
    m = load()
    na, nb = a_keys.numel(), b_keys.numel()
    dev = a_keys.device
    s = torch.cuda.current_stream().cuda_stream
    if na == 0 or nb == 0:
        empty = torch.empty(0, dtype=torch.int64, device=dev)
        return empty, (empty if a_vals is not None else None), empty.clone()
    counts = torch.empty(na, dtype=torch.int32, device=dev)
    lo_idx = torch.empty(na, dtype=torch.int32, device=dev)
    m.join_count(a_keys.data_ptr(), na, b_keys.data_ptr(), nb,
                 counts.data_ptr(), lo_idx.data_ptr(), s)
    counts64 = counts.to(torch.int64)
    offsets = torch.cumsum(counts64, 0) - counts64
    total = int(offsets[-1].item() + counts64[-1].item())
    out_key = torch.empty(total, dtype=torch.int64, device=dev)
    out_a = (torch.empty(total, dtype=torch.int64, device=dev)
             if a_vals is not None else None)
    out_b = torch.empty(total, dtype=torch.int64, device=dev)
    m.join_emit(a_keys.data_ptr(),
                a_vals.data_ptr() if a_vals is not None else 0, na,
                b_vals.data_ptr(), counts.data_ptr(), lo_idx.data_ptr(),
                offsets.data_ptr(), out_key.data_ptr(),
                out_a.data_ptr() if out_a is not None else 0,
                out_b.data_ptr(), s)
    return out_key, out_a, out_b
