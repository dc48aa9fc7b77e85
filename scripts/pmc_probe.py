#!/usr/bin/env python3
"""Tiny PMC workload: a few launches of the ByteGrad hot kernels."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bagua_amd.ops import native

lib = native.require()
n = 1 << 24  # 16Mi
chunks = 8
x = torch.randn(n, device="cuda", dtype=torch.bfloat16)
stride = lib.compressed_chunk_stride(n // chunks)
wire = torch.empty(stride * chunks, dtype=torch.uint8, device="cuda")
for _ in range(3):
    lib.compress_chunked(x, wire, chunks, -1)
    lib.dequant_reduce(wire, x, chunks, 0, True)
    lib.decompress_chunked(wire, x, chunks, -1)
torch.cuda.synchronize()
print("pmc workload done")
