"""distmlip_amd — MI355X-native graph-parallel MLIP inference engine.

A from-scratch rebuild of the hot path of AegisIK/DistMLIP (the CHGNet
energy+force distributed forward: neighbor list + slab partitioning, edge
gather, radial-basis/edge embedding, per-edge gated-MLP message compute,
segmented scatter-add aggregation, and the inter-partition ghost-atom halo
exchange) for AMD MI355X (gfx950): PyTorch-ROCm host, a C-ABI HIP extension
for the hot kernels, RCCL send/recv over xGMI for the halo.

Global dtype defaults mirror the reference package surface
(/root/reference/DistMLIP/__init__.py:7-33): fp32 features, int32 indices,
switchable via set_default_dtype.
"""
from __future__ import annotations

import numpy as np
import torch

float_np = np.float32
float_th = torch.float
int_np = np.int32
int_th = torch.int32


def set_default_dtype(type_: str = "float", size: int = 32):
    """Set default dtype size (16/32/64) for int or float.

    Mirrors DistMLIP.set_default_dtype (reference __init__.py:15-33).
    """
    if size in (16, 32, 64):
        globals()[f"{type_}_th"] = getattr(torch, f"{type_}{size}")
        globals()[f"{type_}_np"] = getattr(np, f"{type_}{size}")
        torch.set_default_dtype(getattr(torch, f"float{size}"))
    else:
        raise ValueError("Invalid dtype size")
    if type_ == "float" and size == 16 and not torch.cuda.is_available():
        raise Exception("torch.float16 requires a GPU device")
