"""Small tensor helpers shared by the workloads."""

from __future__ import annotations


def as_device_i64(chunk, device="cuda"):
    """View a fetched chunk as an int64 device tensor.

    Fast path: already a device uint8 tensor (HBM fetch). Slow path: host
    bytes (spilled blocks served from shm) are uploaded — the consumer of
    a mixed HBM/host shuffle sees one uniform type.
    """
    import torch
    if isinstance(chunk, torch.Tensor):
        return chunk.view(torch.int64)
    return torch.frombuffer(bytearray(chunk), dtype=torch.int64).to(device)
