"""Hyperparameter search over the bucket-fusion space.

The reference used skopt Gaussian-process optimization
(bagua/service/bayesian_optimizer.py:34-79) over
{bucket_size_2p in [10, 31], is_hierarchical_reduce in {0,1}}. skopt is
not available offline, and the space is tiny (44 points), so this build
uses a deterministic low-discrepancy sweep plus hill-climbing around the
incumbent — equivalent coverage at this cardinality without the GP.

The search *space* differs deliberately: MI355X has 288 GB HBM3E per GPU
and xGMI favors large fused buckets, so bucket_size_2p ranges [20, 31]
(1 MiB .. 2 GiB) instead of the reference's [10, 31].
"""

import random
from typing import Dict, List, Tuple

class IntParam:
    def __init__(self, val: int, space_dimension: Tuple[int, int]):
        self.val = int(val)
        self.space_dimension = space_dimension


class BoolParam:
    def __init__(self, val: bool):
        self.val = bool(val)


class BayesianOptimizer:
    """tell/ask API compatible with the reference's wrapper."""

    def __init__(self, param_declaration: Dict, n_initial_points: int = 10,
                 seed: int = 0):
        self.param_declaration = dict(param_declaration)
        self.n_initial_points = n_initial_points
        self.rng = random.Random(seed)
        self.history: List[Tuple[Dict, float]] = []
        self._initial_queue = self._make_initial_points()

    # ------------------------------------------------------------------
    def _make_initial_points(self) -> List[Dict]:
        """Low-discrepancy sweep: bucket sizes spread across the range,
        alternating hierarchy."""
        points = []
        lo, hi = None, None
        for name, param in self.param_declaration.items():
            if isinstance(param, IntParam):
                lo, hi = param.space_dimension
        if lo is None:
            return points
        span = hi - lo
        offsets = [0.5, 0.25, 0.75, 0.125, 0.625, 0.375, 0.875, 0.0625]
        for i in range(self.n_initial_points):
            frac = offsets[i % len(offsets)]
            p = {}
            for name, param in self.param_declaration.items():
                if isinstance(param, IntParam):
                    p[name] = int(round(lo + frac * span))
                elif isinstance(param, BoolParam):
                    p[name] = bool(i % 2)
            points.append(p)
        return points

    def tell(self, params: Dict, score: float):
        self.history.append((dict(params), float(score)))

    def ask(self) -> Dict:
        if self._initial_queue:
            return self._initial_queue.pop(0)
        if not self.history:
            return self._random_point()
        # hill-climb around the incumbent
        best_params, _ = max(self.history, key=lambda kv: kv[1])
        candidate = dict(best_params)
        for name, param in self.param_declaration.items():
            if isinstance(param, IntParam):
                lo, hi = param.space_dimension
                step = self.rng.choice([-2, -1, 0, 1, 2])
                candidate[name] = min(hi, max(lo, candidate[name] + step))
            elif isinstance(param, BoolParam):
                if self.rng.random() < 0.25:
                    candidate[name] = not candidate[name]
        return candidate

    def _random_point(self) -> Dict:
        p = {}
        for name, param in self.param_declaration.items():
            if isinstance(param, IntParam):
                lo, hi = param.space_dimension
                p[name] = self.rng.randint(lo, hi)
            elif isinstance(param, BoolParam):
                p[name] = bool(self.rng.getrandbits(1))
        return p
