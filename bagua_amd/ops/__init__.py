"""Device-math ops with native-HIP dispatch.

Every function here has two paths:

* **GPU**: hand-written CDNA4 HIP kernels from ``bagua_amd._C`` (built for
  gfx950; mandatory on GPU boxes — see ops/native.py);
* **CPU**: torch reference implementations (tests, gloo path).

Kernel inventory mirrors the reference
(rust/bagua-core/bagua-core-internal/kernels/bagua_kernels.cu:196-501) plus
bf16 support and fused optimizer steps, re-tiled for 64-wide wavefronts.
"""

import torch

from . import quant
from . import native


def _use_native(*tensors) -> bool:
    return all(t.is_cuda for t in tensors) and native.available()


def average_inplace(x: torch.Tensor, y: torch.Tensor):
    """x = (x + y) / 2"""
    if _use_native(x, y):
        native.lib().average_inplace(x, y)
    else:
        x.add_(y).mul_(0.5)


def add_inplace(x: torch.Tensor, y: torch.Tensor):
    if _use_native(x, y):
        native.lib().add_inplace(x, y)
    else:
        x.add_(y)


def substract_inplace(x: torch.Tensor, y: torch.Tensor):
    if _use_native(x, y):
        native.lib().substract_inplace(x, y)
    else:
        x.sub_(y)


def addmul_inplace(x: torch.Tensor, y: torch.Tensor, factor: float):
    """x += y * factor"""
    if _use_native(x, y):
        native.lib().addmul_inplace(x, y, float(factor))
    else:
        x.add_(y, alpha=factor)


def divide_inplace(x: torch.Tensor, divisor: float):
    if _use_native(x):
        native.lib().divide_inplace(x, float(divisor))
    else:
        x.div_(divisor)


def async_model_average(x: torch.Tensor, reduced: torch.Tensor,
                        x_copy: torch.Tensor, nranks: int):
    """x += reduced / nranks - x_copy
    (reference: bagua_kernels.cu:257-267)"""
    if _use_native(x, reduced, x_copy):
        native.lib().async_model_average(x, reduced, x_copy, float(nranks))
    else:
        x.add_(reduced / nranks - x_copy)


def reduce_chunk_inplace(flat: torch.Tensor, num_chunks: int,
                         target_chunk: int, average: bool):
    """Reduce ``num_chunks`` equal chunks of ``flat`` into chunk
    ``target_chunk`` (reference: bagua_kernels.cu:374-401)."""
    if _use_native(flat):
        native.lib().reduce_chunk_inplace(flat, num_chunks, target_chunk,
                                          average)
        return
    v = flat.view(num_chunks, -1)
    if average:
        red = v.float().mean(0).to(flat.dtype)
    else:
        red = v.float().sum(0).to(flat.dtype)
    v[target_chunk].copy_(red)


def compressed_buffer_numel(numel: int, num_chunks: int) -> int:
    chunk = numel // num_chunks
    return quant.compressed_chunk_bytes(chunk) * num_chunks


def compress_chunked(flat: torch.Tensor, num_chunks: int,
                     target_chunk: int = -1,
                     out: torch.Tensor = None) -> torch.Tensor:
    """MinMaxUInt8-compress into the chunked wire buffer."""
    if _use_native(flat):
        if out is None:
            out = torch.empty(
                compressed_buffer_numel(flat.numel(), num_chunks),
                dtype=torch.uint8, device=flat.device)
        native.lib().compress_chunked(flat, out, num_chunks, target_chunk)
        return out
    return quant.compress_chunked(flat, num_chunks, target_chunk, out)


def dequant_reduce(buf: torch.Tensor, flat: torch.Tensor, num_chunks: int,
                   target_chunk: int, average: bool):
    """Fused: flat[target_chunk] = reduce over the dequantized wire
    chunks — skips materializing the non-target chunks (they are only
    reduction inputs on the ByteGrad path). Bitwise-identical to
    decompress_chunked_into + reduce_chunk_inplace: values round through
    flat.dtype between dequantize and f32 accumulation."""
    if _use_native(buf, flat) and num_chunks <= 64:
        native.lib().dequant_reduce(buf, flat, num_chunks, target_chunk,
                                    average)
        return
    if _use_native(buf, flat):  # beyond the kernel's LDS param table
        decompress_chunked_into(buf, flat, num_chunks)
        reduce_chunk_inplace(flat, num_chunks, target_chunk, average)
        return
    chunk = flat.numel() // num_chunks
    dec = quant.decompress_chunked(buf, num_chunks, chunk, flat.dtype)
    v = dec.view(num_chunks, -1).float()
    red = v.mean(0) if average else v.sum(0)
    flat.view(num_chunks, -1)[target_chunk].copy_(red.to(flat.dtype))


def decompress_chunked_into(buf: torch.Tensor, flat: torch.Tensor,
                            num_chunks: int, target_chunk: int = -1):
    """Decompress the wire buffer into ``flat`` (all chunks or one)."""
    if _use_native(buf, flat):
        native.lib().decompress_chunked(buf, flat, num_chunks, target_chunk)
        return
    chunk = flat.numel() // num_chunks
    res = quant.decompress_chunked(buf, num_chunks, chunk, flat.dtype,
                                   target_chunk)
    if target_chunk >= 0:
        flat.view(num_chunks, -1)[target_chunk].copy_(
            res.view(num_chunks, -1)[target_chunk])
    else:
        flat.view(-1).copy_(res)
