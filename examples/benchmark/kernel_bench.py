#!/usr/bin/env python3
"""Micro-benchmark the CDNA4 kernel pack vs the HBM3E roofline.

Every kernel is memory-bound; the table prints achieved GB/s next to the
theoretical bytes moved. MI355X HBM3E peak is 8 TB/s (≈6.3 achievable).
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

from bagua_amd.ops import native


def bench(fn, bytes_moved, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return bytes_moved / dt / 1e9, dt * 1e6


def main():
    lib = native.require()
    n = 1 << 26  # 64 Mi elements
    results = []

    for dtype, name in [(torch.float32, "f32"), (torch.float16, "f16"),
                        (torch.bfloat16, "bf16")]:
        es = torch.tensor([], dtype=dtype).element_size()
        x = torch.randn(n, device="cuda").to(dtype)
        y = torch.randn(n, device="cuda").to(dtype)

        gbs, us = bench(lambda: lib.average_inplace(x, y), 3 * n * es)
        results.append(("average_inplace", name, gbs, us))

        gbs, us = bench(lambda: lib.addmul_inplace(x, y, 0.3), 3 * n * es)
        results.append(("addmul_inplace", name, gbs, us))

        chunks = 8
        gbs, us = bench(
            lambda: lib.reduce_chunk_inplace(x, chunks, 0, True),
            n * es + (n // chunks) * es)
        results.append(("reduce_chunk(8)", name, gbs, us))

        chunk = n // chunks
        stride = lib.compressed_chunk_stride(chunk)
        wire = torch.empty(stride * chunks, dtype=torch.uint8,
                           device="cuda")
        # compress = minmax read (n*es) + quantize read (n*es) + write n
        gbs, us = bench(lambda: lib.compress_chunked(x, wire, chunks, -1),
                        2 * n * es + n)
        results.append(("compress_chunked", name, gbs, us))

        gbs, us = bench(
            lambda: lib.decompress_chunked(wire, x, chunks, -1),
            n + n * es)
        results.append(("decompress_chunked", name, gbs, us))

        # fused dequantize+reduce (ByteGrad hot path) vs the unfused
        # chain it replaces: decompress (n + n*es) + reduce
        # (n*es + chunk*es) -> fused reads n u8 + writes chunk*es only
        gbs, us = bench(
            lambda: lib.dequant_reduce(wire, x, chunks, 0, True),
            n + chunk * es)
        results.append(("dequant_reduce(8)", name, gbs, us))

        def unfused():
            lib.decompress_chunked(wire, x, chunks, -1)
            lib.reduce_chunk_inplace(x, chunks, 0, True)

        gbs, us = bench(unfused, n + chunk * es)  # same useful bytes
        results.append(("  unfused chain", name, gbs, us))

    m = torch.zeros(n, device="cuda")
    g = torch.randn(n, device="cuda")
    p = torch.randn(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    gbs, us = bench(lambda: lib.fused_sgd_step(
        p, g, m, 0.01, 0.9, 0.0, 1e-4, False, True), 5 * n * 4)
    results.append(("fused_sgd_step", "f32", gbs, us))
    gbs, us = bench(lambda: lib.fused_adam_step(
        p, g, m, v, 10, 1e-3, 0.9, 0.999, 1e-8, 0.0, False), 7 * n * 4)
    results.append(("fused_adam_step", "f32", gbs, us))

    # pure-bf16 flagship path: bf16 params/grads + fp32 master & momentum
    pb = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    gb = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    master = pb.float()
    mm = torch.zeros(n, device="cuda")
    # bytes: read g (2B) + master (4) + m (4); write p (2) + master (4) + m (4)
    gbs, us = bench(lambda: lib.fused_sgd_mixed_step(
        pb, gb, master, mm, 0.01, 0.9, 0.0, 1e-4, False, True), 20 * n)
    results.append(("fused_sgd_mixed", "bf16", gbs, us))

    print("%-20s %-5s %10s %10s" % ("kernel", "dtype", "GB/s", "us"))
    for name, dt, gbs, us in results:
        print("%-20s %-5s %10.0f %10.1f" % (name, dt, gbs, us))


if __name__ == "__main__":
    main()
