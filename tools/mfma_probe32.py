#!/usr/bin/env python3
"""Empirically identify the v_mfma_f32_32x32x16_bf16 operand/result
layouts (ROUND2 design 2a): tries A/B contiguous-vs-split k mappings and
two D mappings against a host reference with asymmetric operands."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from rbg_amd import ops

dev = torch.device("cuda:0")
torch.manual_seed(0)
a = (torch.arange(32 * 16, dtype=torch.float32, device=dev).reshape(32, 16)
     % 13 - 6).bfloat16()
b = (torch.arange(16 * 32, dtype=torch.float32, device=dev).reshape(16, 32)
     % 7 - 3).bfloat16()
ref = a.float() @ b.float()
hip = ops._require_hip()
for asp in (0, 1):
    for bsp in (0, 1):
        for dm in (0, 1):
            d = hip.mfma_probe32(a, b, asp, bsp, dm)
            err = (d - ref).abs().max().item()
            tag = "MATCH" if err < 1e-3 else ""
            print(f"a_split={asp} b_split={bsp} dmap={dm}: "
                  f"maxerr {err:.4f} {tag}", flush=True)
