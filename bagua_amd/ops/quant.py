"""MinMaxUInt8 quantization — pure-torch reference implementation.

Semantics match the reference CUDA kernels exactly
(reference: rust/bagua-core/bagua-core-internal/kernels/bagua_kernels.cu:403-501
and the golden Python model in tests/internal/compressor.py:4-33):

    scale       = 255 / (max - min + 1e-7)
    upper_bound = rint(max * scale)
    lower_bound = upper_bound - 255
    q           = clamp(rint(x * scale), max=upper_bound) - lower_bound   (uint8)
    x'          = (q + lower_bound) / scale

This module is the CPU executor's compressor and the numerics oracle for
the hand-written CDNA4 HIP kernels (tests compare bitwise on the uint8
payload).
"""

import torch

EPS = 1e-7
LEVELS = 255.0


def compress(tensor: torch.Tensor):
    """Compress a 1-D tensor. Returns (minmax[2] float32, payload uint8)."""
    t = tensor.float()
    _min, _max = t.min(), t.max()
    scale = LEVELS / (_max - _min + EPS)
    upper = torch.round(_max * scale)
    lower = upper - LEVELS
    level = torch.clamp(torch.round(t * scale), max=upper)
    minmax = torch.stack([_min, _max]).to(torch.float32)
    return minmax, (level - lower).to(torch.uint8)


def decompress(minmax: torch.Tensor, payload: torch.Tensor,
               dtype=torch.float32) -> torch.Tensor:
    _min, _max = minmax[0].float(), minmax[1].float()
    scale = LEVELS / (_max - _min + EPS)
    upper = torch.round(_max * scale)
    lower = upper - LEVELS
    return ((payload.float() + lower) / scale).to(dtype)


# ---------------------------------------------------------------------------
# Chunked wire format (matches the native kernels): per chunk, a 32-byte
# header holding min,max ALWAYS as float32 (bytes 8..31 zero), then the
# uint8 payload. Deliberate deviation from the reference's layout, which
# stores min/max in the SOURCE dtype (datatypes/mod.rs:700-777): a fixed
# f32 header keeps one kernel per dtype and loses no precision for
# f16/bf16 inputs. Wire-incompatible with the reference by design.
# ---------------------------------------------------------------------------

HEADER_BYTES = 32


def compressed_chunk_bytes(chunk_numel: int) -> int:
    # payload aligned to 32 bytes like the reference
    payload = (chunk_numel + HEADER_BYTES - 1) // HEADER_BYTES * HEADER_BYTES
    return HEADER_BYTES + payload


def compress_chunked(tensor: torch.Tensor, num_chunks: int,
                     target_chunk: int = -1,
                     out: torch.Tensor = None) -> torch.Tensor:
    """Compress ``tensor`` (numel divisible by num_chunks) into the wire
    buffer. ``target_chunk=-1`` compresses all chunks; otherwise only that
    chunk's region is written."""
    flat = tensor.reshape(-1)
    assert flat.numel() % num_chunks == 0
    chunk = flat.numel() // num_chunks
    stride = compressed_chunk_bytes(chunk)
    if out is None:
        out = torch.zeros(stride * num_chunks, dtype=torch.uint8,
                          device=tensor.device)
    for c in range(num_chunks):
        if target_chunk >= 0 and c != target_chunk:
            continue
        minmax, payload = compress(flat.narrow(0, c * chunk, chunk))
        hdr = out.narrow(0, c * stride, HEADER_BYTES)
        hdr[:8].view(torch.float32).copy_(minmax)
        out.narrow(0, c * stride + HEADER_BYTES, chunk).copy_(payload)
    return out


def decompress_chunked(buf: torch.Tensor, num_chunks: int, chunk: int,
                       dtype=torch.float32,
                       target_chunk: int = -1) -> torch.Tensor:
    stride = compressed_chunk_bytes(chunk)
    res = torch.zeros(num_chunks * chunk, dtype=dtype, device=buf.device)
    for c in range(num_chunks):
        if target_chunk >= 0 and c != target_chunk:
            continue
        minmax = buf.narrow(0, c * stride, 8).view(torch.float32)
        payload = buf.narrow(0, c * stride + HEADER_BYTES, chunk)
        res.narrow(0, c * chunk, chunk).copy_(
            decompress(minmax, payload, dtype))
    return res
