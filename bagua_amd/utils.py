"""Misc utilities (reference: bagua/torch_api/utils.py:127-244 and the
Rust show_version, lib.rs:103-123)."""

import sys
import time
from typing import List, Tuple

import torch


def show_version():
    """Print build / runtime versions (reference: lib.rs:103-123)."""
    import bagua_amd

    lines = [
        "bagua_amd %s" % bagua_amd.__version__,
        "python    %s" % sys.version.split()[0],
        "torch     %s" % torch.__version__,
        "ROCm/HIP  %s" % getattr(torch.version, "hip", None),
        "GPU       %s" % (torch.cuda.get_device_name(0)
                          if torch.cuda.is_available() else "none"),
    ]
    try:
        from bagua_amd.ops import native

        lines.append("native    %s" % ("loaded" if native.available()
                                       else "unavailable"))
    except Exception:  # noqa: BLE001
        lines.append("native    error")
    print("\n".join(lines))
    return lines


def flatten(tensors: List[torch.Tensor]) -> torch.Tensor:
    return torch.cat([t.reshape(-1) for t in tensors])


def unflatten(flat: torch.Tensor, tensors: List[torch.Tensor]
              ) -> List[torch.Tensor]:
    out, offset = [], 0
    for t in tensors:
        out.append(flat.narrow(0, offset, t.numel()).view_as(t))
        offset += t.numel()
    return out


def to_bagua_datatype(dtype: torch.dtype) -> str:
    return {
        torch.float32: "f32",
        torch.float16: "f16",
        torch.bfloat16: "bf16",
        torch.uint8: "u8",
        torch.int64: "i64",
    }[dtype]


class StatisticalAverage:
    """Exponential-window throughput average
    (reference: utils.py:127-244): ``record(value)`` time-stamped samples;
    ``get(window_s)`` returns the mean of samples within the window."""

    def __init__(self, max_records: int = 128):
        self.records: List[Tuple[float, float]] = []
        self.max_records = max_records

    def record(self, value: float, now: float = None):
        now = time.time() if now is None else now
        self.records.append((now, float(value)))
        del self.records[:-self.max_records]

    def get(self, window_s: float = 60.0, now: float = None) -> float:
        now = time.time() if now is None else now
        vals = [v for t, v in self.records if now - t <= window_s]
        if not vals:
            return 0.0
        return sum(vals) / len(vals)

    def total_recording_time(self) -> float:
        if len(self.records) < 2:
            return 0.0
        return self.records[-1][0] - self.records[0][0]


def check_contiguous(tensors) -> bool:
    """True iff the tensors lie back-to-back in one storage
    (reference: utils.py:51-58)."""
    data_ptr = None
    for t in tensors:
        if data_ptr is not None and t.data_ptr() != data_ptr:
            return False
        data_ptr = t.data_ptr() + t.numel() * t.element_size()
    return True


def apply_flattened_call(tensors, call):
    """Flatten ``tensors`` (same dtype) into one buffer, run ``call`` on
    it, copy the result back (reference: utils.py:16-29)."""
    flat = flatten(tensors)
    call(flat)
    offset = 0
    for t in tensors:
        t.copy_(flat.narrow(0, offset, t.numel()).view_as(t))
        offset += t.numel()


def apply_flattened_call_all(tensors, call):
    """Group by dtype, then apply_flattened_call per group
    (reference: utils.py:31-49)."""
    groups = {}
    for t in tensors:
        groups.setdefault(t.dtype, []).append(t)
    for group in groups.values():
        apply_flattened_call(group, call)


def average_by_removing_extreme_values(raw_scores):
    """Robust mean of speed samples: drop the warmup third, then
    iteratively remove >1-sigma outliers (reference: utils.py:94-126).
    Returns (mean, std, kept_samples)."""
    import numpy as np

    scores = np.asarray(raw_scores, dtype=float)
    scores = scores[len(scores) // 3:]

    def weed(x):
        mean, std = np.mean(x), np.std(x)
        kept = x[np.abs(x - mean) < std]
        return kept if len(kept) else x

    scores = weed(scores)
    for _ in range(10):
        if np.std(scores) < np.mean(scores):
            break
        scores = weed(scores)
    return float(np.mean(scores)), float(np.std(scores)), scores.tolist()
