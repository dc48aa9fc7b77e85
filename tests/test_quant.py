"""MinMaxUInt8 quantizer numerics vs the golden formula
(reference semantics: tests/internal/compressor.py:4-33)."""

import torch

from bagua_amd.ops import quant


def _golden_compress(tensor):
    eps, levels = 1e-7, 255.0
    _min, _max = torch.min(tensor), torch.max(tensor)
    scale = levels / (_max - _min + eps)
    upper = torch.round(_max * scale)
    lower = upper - levels
    level = torch.clamp(torch.round(tensor * scale), max=upper)
    return _min, _max, (level - lower).to(torch.uint8)


def test_roundtrip_error_bound():
    torch.manual_seed(0)
    x = torch.rand(1000)
    minmax, payload = quant.compress(x)
    y = quant.decompress(minmax, payload)
    # quantization step = range/255; roundtrip error <= one step
    step = (x.max() - x.min()) / 255.0
    assert (x - y).abs().max() <= step + 1e-6


def test_matches_golden():
    torch.manual_seed(1)
    x = torch.randn(4096)
    _min, _max, gold = _golden_compress(x)
    minmax, payload = quant.compress(x)
    assert torch.equal(payload, gold)
    assert torch.allclose(minmax[0], _min) and torch.allclose(minmax[1], _max)


def test_chunked_wire_roundtrip():
    torch.manual_seed(2)
    n_chunks = 4
    x = torch.randn(n_chunks * 256)
    buf = quant.compress_chunked(x, n_chunks)
    y = quant.decompress_chunked(buf, n_chunks, 256)
    step = (x.view(n_chunks, -1).max(1).values
            - x.view(n_chunks, -1).min(1).values).max() / 255.0
    assert (x - y).abs().max() <= step + 1e-6


def test_chunked_target_chunk():
    torch.manual_seed(3)
    n_chunks = 4
    x = torch.randn(n_chunks * 64)
    full = quant.compress_chunked(x, n_chunks)
    only2 = quant.compress_chunked(x, n_chunks, target_chunk=2)
    stride = quant.compressed_chunk_bytes(64)
    assert torch.equal(full[2 * stride:3 * stride],
                       only2[2 * stride:3 * stride])
    assert only2[:stride].sum() == 0  # untouched chunks stay zero
