"""Generic fused optimizer — kernel fusion by contiguity.

Re-design of the reference's contrib fused optimizer
(bagua/torch_api/contrib/fuse/optimizer.py:14-574): works with ANY torch
optimizer by (1) flattening each param group's weights/grads/state into
contiguous storage, (2) at ``fuse_step`` time finding runs of parameters
whose weight, grad and every tensor state are *mutually* contiguous,
(3) building flat views over those runs and stepping a shadow optimizer
instance on the fused views, (4) writing per-param state back as views.

On MI355X the fused views mean one kernel launch per group instead of one
per parameter — the launch-bound regime the 256-CU chip hates most. (The
dedicated HIP multi-tensor SGD/Adam kernels in ops/csrc cover the bucket
hot path; this generic version covers arbitrary optimizers.)
"""

import copy
import logging
from collections import defaultdict
from typing import Any, Dict, List, Optional, Tuple

import torch

logger = logging.getLogger(__name__)


# ---------------------------------------------------------------------------
# contiguity machinery
# ---------------------------------------------------------------------------


def _is_adjacent(a: torch.Tensor, b: torch.Tensor) -> bool:
    size_a = a.numel() * a.element_size()
    size_b = b.numel() * b.element_size()
    return (a.data_ptr() == b.data_ptr() + size_b
            or b.data_ptr() == a.data_ptr() + size_a)


def check_contiguous(tensors: List[torch.Tensor]) -> bool:
    ptr = None
    for t in tensors:
        if ptr is not None and t.data_ptr() != ptr:
            return False
        ptr = t.data_ptr() + t.numel() * t.element_size()
    return True


def _find_contiguous_runs(tensors: List[torch.Tensor]) -> List[List[int]]:
    order = sorted(range(len(tensors)), key=lambda i: tensors[i].data_ptr())
    runs, cur = [], []
    for i in order:
        if cur and not _is_adjacent(tensors[i], tensors[cur[-1]]):
            if len(cur) > 1:
                runs.append(cur)
            cur = []
        cur.append(i)
    if len(cur) > 1:
        runs.append(cur)
    return runs


def _mutual_runs(tensors_list: List[List[torch.Tensor]]) -> List[List[int]]:
    """Runs contiguous in EVERY tensor list (weights, grads, each state)."""
    if not tensors_list:
        return []
    runs = _find_contiguous_runs(tensors_list[0])
    for tensors in tensors_list[1:]:
        other = _find_contiguous_runs(tensors)
        runs = [r for r in runs if r in other]
    return runs


def _flat_view(tensors: List[torch.Tensor], indices: List[int]
               ) -> torch.Tensor:
    group = [tensors[i] for i in indices]
    assert check_contiguous(group), "fused run must be contiguous"
    total = sum(t.numel() for t in group)
    t0 = group[0]
    out = torch.empty(0, dtype=t0.dtype, device=t0.device)
    out.set_(t0.untyped_storage(),
             t0.storage_offset(), (total,))
    return out


def _same_view(existing: Optional[torch.Tensor],
               candidate: torch.Tensor) -> bool:
    return (existing is not None
            and existing.data_ptr() == candidate.data_ptr()
            and existing.numel() == candidate.numel()
            and existing.dtype == candidate.dtype)


# ---------------------------------------------------------------------------
# flattening
# ---------------------------------------------------------------------------


def _flatten_inplace(tensors: List[torch.Tensor],
                     set_fns: List) -> None:
    """Copy tensors into one contiguous buffer and re-point each through
    its setter."""
    if not tensors:
        return
    total = sum(t.numel() for t in tensors)
    flat = torch.zeros(total, dtype=tensors[0].dtype,
                       device=tensors[0].device)
    offset = 0
    for t, set_fn in zip(tensors, set_fns):
        view = flat.narrow(0, offset, t.numel()).view_as(t)
        view.copy_(t)
        set_fn(view)
        offset += t.numel()


def flatten_params_and_states(optimizer: torch.optim.Optimizer):
    """Flatten weights, grads and tensor states per group
    (reference: fuse/optimizer.py:14-81)."""
    for group in optimizer.param_groups:
        params = group["params"]
        if not params:
            continue

        weights = [p.data for p in params]

        def make_wset(p):
            def s(view):
                p.data = view
            return s

        _flatten_inplace(weights, [make_wset(p) for p in params])

        grads, gsets = [], []
        for p in params:
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            grads.append(p.grad.data)

            def make_gset(p):
                def s(view):
                    p.grad = view
                    p._bagua_grad_view = view
                return s

            gsets.append(make_gset(p))
        _flatten_inplace(grads, gsets)

        # tensor states shared by all params of the group
        names = set()
        for p in params:
            names |= {k for k, v in optimizer.state[p].items()
                      if isinstance(v, torch.Tensor)}
        for name in names:
            if not all(name in optimizer.state[p] for p in params):
                continue
            tensors = [optimizer.state[p][name] for p in params]

            def make_sset(p, name):
                def s(view):
                    optimizer.state[p][name] = view
                return s

            _flatten_inplace(tensors,
                             [make_sset(p, name) for p in params])


# ---------------------------------------------------------------------------
# public API
# ---------------------------------------------------------------------------


def is_fused_optimizer(optimizer: torch.optim.Optimizer) -> bool:
    return hasattr(optimizer, "_bagua_fused_optimizer")


def _make_shadow(optimizer: torch.optim.Optimizer):
    shadow = copy.copy(optimizer)
    shadow.param_groups = []
    for group in optimizer.param_groups:
        new_group = {k: v for k, v in group.items() if k != "params"}
        new_group["params"] = list(group["params"])
        shadow.add_param_group(new_group)
    shadow.state = defaultdict(dict)
    return shadow


def fuse_optimizer(optimizer: torch.optim.Optimizer, do_flatten: bool = True,
                   check_flatten: bool = True):
    """Convert any torch optimizer into a fused optimizer. Adds a
    ``fuse_step()`` method; ``step()`` keeps its original behavior."""
    if is_fused_optimizer(optimizer):
        raise RuntimeError("trying to fuse an optimizer twice!")
    optimizer._bagua_check_flatten = do_flatten and check_flatten
    optimizer._bagua_fused_count = 0
    optimizer._bagua_fused_optimizer = _make_shadow(optimizer)
    if do_flatten:
        flatten_params_and_states(optimizer)
    if not hasattr(type(optimizer), "fuse_step"):
        type(optimizer).fuse_step = fuse_step
    return optimizer


def _collect_states(optimizer, params
                    ) -> Tuple[Optional[Dict[str, List[torch.Tensor]]],
                               Optional[Dict[str, Any]]]:
    """(tensor states by name, scalar states by name) across params; None
    when parameters disagree (reference: fuse/optimizer.py:522-574)."""
    state_tensors: Dict[str, List[torch.Tensor]] = {}
    state_scalars: Dict[str, Any] = {}
    tnames = {k for p in params for k, v in optimizer.state[p].items()
              if isinstance(v, torch.Tensor)}
    snames = {k for p in params for k, v in optimizer.state[p].items()
              if not isinstance(v, torch.Tensor)}
    for name in tnames:
        tensors = []
        for p in params:
            if name not in optimizer.state[p]:
                return None, None
            tensors.append(optimizer.state[p][name])
        state_tensors[name] = tensors
    for name in snames:
        scalar = None
        for p in params:
            if name not in optimizer.state[p]:
                return None, None
            v = optimizer.state[p][name]
            if scalar is not None:
                eq = (torch.equal(scalar, v)
                      if isinstance(scalar, torch.Tensor) else scalar == v)
                if not eq:
                    return None, None
            scalar = v
        state_scalars[name] = scalar
    return state_tensors, state_scalars


def fuse_step(optimizer: torch.optim.Optimizer, closure=None):
    """Fused parameter update (reference: fuse/optimizer.py:319-344)."""
    assert is_fused_optimizer(optimizer), (
        "call fuse_optimizer() before fuse_step()")
    _do_fuse(optimizer)
    result = optimizer._bagua_fused_optimizer.step(closure)
    _sync_back(optimizer)
    return result


def _do_fuse(optimizer):
    shadow = optimizer._bagua_fused_optimizer
    for group, fused_group in zip(optimizer.param_groups,
                                  shadow.param_groups):
        # keep hyperparameters in sync
        for k, v in group.items():
            if k != "params":
                fused_group[k] = v

        params = group["params"]
        weights = [p.data for p in params]
        grads = []
        for p in params:
            # re-pin grads into the flat buffer when
            # zero_grad(set_to_none=True) or autograd replaced them
            view = getattr(p, "_bagua_grad_view", None)
            if view is not None:
                if p.grad is None:
                    view.zero_()
                    p.grad = view
                elif p.grad.data_ptr() != view.data_ptr():
                    view.copy_(p.grad.detach())
                    p.grad = view
            elif p.grad is None:
                p.grad = torch.zeros_like(p)
            grads.append(p.grad)
        state_tensors, state_scalars = _collect_states(optimizer, params)
        if state_tensors is None:
            fused_group["params"] = list(params)
            for p in params:
                shadow.state[p] = optimizer.state[p]
            continue

        if optimizer._bagua_check_flatten and not (
                check_contiguous(weights) and check_contiguous(grads)):
            logger.warning(
                "parameter storage changed after flatten; fused update "
                "will fall back to per-run fusion")

        runs = _mutual_runs([weights, grads] + list(state_tensors.values()))
        if runs:
            optimizer._bagua_fused_count += 1

        new_params = []
        fused_ids = set()
        for indices in runs:
            w = _flat_view(weights, indices)
            g = _flat_view(grads, indices)
            fp = torch.nn.Parameter(w, requires_grad=False)
            fp.grad = g
            fp._bagua_fused_param_ids = indices
            shadow.state[fp] = {}
            for name, tensors in state_tensors.items():
                shadow.state[fp][name] = _flat_view(tensors, indices)
            for name, scalar in state_scalars.items():
                shadow.state[fp][name] = copy.deepcopy(scalar) \
                    if isinstance(scalar, torch.Tensor) else scalar
            new_params.append(fp)
            fused_ids.update(indices)

        for idx, p in enumerate(params):
            if idx not in fused_ids:
                new_params.append(p)
                shadow.state[p] = optimizer.state[p]

        fused_group["params"] = new_params


def _sync_back(optimizer):
    """Propagate fused state back onto the original parameters
    (reference: fuse/optimizer.py:476-510)."""
    shadow = optimizer._bagua_fused_optimizer
    for group, fused_group in zip(optimizer.param_groups,
                                  shadow.param_groups):
        params = group["params"]
        for fp in fused_group["params"]:
            ids = getattr(fp, "_bagua_fused_param_ids", None)
            if ids is None:
                continue
            originals = [params[i] for i in ids]
            for name, v in shadow.state[fp].items():
                if isinstance(v, torch.Tensor) and v.dim() == 1 \
                        and v.numel() == sum(p.numel() for p in originals):
                    offset = 0
                    for p in originals:
                        optimizer.state[p][name] = v.narrow(
                            0, offset, p.numel()).view_as(p)
                        offset += p.numel()
                else:
                    for p in originals:
                        optimizer.state[p][name] = (
                            v.clone() if isinstance(v, torch.Tensor) else v)
