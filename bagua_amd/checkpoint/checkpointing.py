"""Megatron-style checkpoint save/load with MoE-aware expert sharding
(reference: bagua/torch_api/checkpoint/checkpointing.py:17-363).

Layout:
    {path}/latest_checkpointed_iteration.txt
    {path}/iter_{iteration:07d}/mp_rank_00_model_states.pt
    {path}/iter_{iteration:07d}/expert_{gid}_mp_rank_00_model_states.pt
    {path}/iter_{iteration:07d}/expert_parallel_rank_{r}_mp_rank_00_optim_states.pt

Dense models: rank 0 saves everything. MoE models: every EP rank saves
its local experts under global-expert-id filenames plus its own optimizer
state; load remaps global -> local ids.
"""

import logging
import os
import re
from collections import defaultdict
from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

_MOE_PREFIX = ".bagua_moe.experts.bagua_experts."


def _unwrap(model: torch.nn.Module) -> torch.nn.Module:
    if hasattr(model, "inner"):  # DistributedDataParallel wrapper
        return model.module
    return model


def _has_moe_layers(model) -> Tuple[bool, int]:
    from ..parallel.moe.layer import MoE

    for _, module in model.named_modules():
        if isinstance(module, MoE):
            return True, module.num_experts
    return False, 0


def _ensure_dir(filename: str):
    os.makedirs(os.path.dirname(filename), exist_ok=True)


def _iter_dir(iteration: int, release: bool = False) -> str:
    return "release" if release else "iter_{:07d}".format(iteration)


def _model_ckpt_name(path, iteration, mp_rank=0, release=False):
    return os.path.join(path, _iter_dir(iteration, release),
                        "mp_rank_{:02d}_model_states.pt".format(mp_rank))


def _expert_ckpt_name(path, expert_id, iteration, mp_rank=0, release=False):
    return os.path.join(
        path, _iter_dir(iteration, release),
        "expert_{}_mp_rank_{:02d}_model_states.pt".format(expert_id,
                                                          mp_rank))


def _optimizer_ckpt_name(path, iteration, ep_rank, mp_rank=0,
                         release=False):
    return os.path.join(
        path, _iter_dir(iteration, release),
        "expert_parallel_rank_{}_mp_rank_{:02d}_optim_states.pt".format(
            ep_rank, mp_rank))


def _tracker_filename(path: str) -> str:
    return os.path.join(path, "latest_checkpointed_iteration.txt")


def _read_metadata(tracker_filename: str) -> Tuple[int, bool]:
    with open(tracker_filename) as f:
        meta = f.read().strip()
    try:
        return int(meta), False
    except ValueError:
        if meta == "release":
            return 0, True
        raise ValueError("invalid tracker file %s" % tracker_filename)


def _rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def _world() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


# ---------------------------------------------------------------------------


def save_checkpoint(iteration: int, checkpoints_path: str, model,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    lr_scheduler=None):
    """Save a checkpoint and update the tracker file."""
    model = _unwrap(model)
    logger.info("saving checkpoint at iteration %7d to %s", iteration,
                checkpoints_path)
    has_moe, num_experts = _has_moe_layers(model)
    if has_moe:
        _save_moe_checkpoint(iteration, checkpoints_path, num_experts,
                             model, optimizer, lr_scheduler)
    else:
        _save_dense_checkpoint(iteration, checkpoints_path, model,
                               optimizer, lr_scheduler)
    if dist.is_initialized():
        dist.barrier()
    if _rank() == 0:
        with open(_tracker_filename(checkpoints_path), "w") as f:
            f.write(str(iteration))
    if dist.is_initialized():
        dist.barrier()


def _save_dense_checkpoint(iteration, path, model, optimizer, lr_scheduler):
    if _rank() != 0:
        return
    state = {"iteration": iteration, "model": model.state_dict()}
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if lr_scheduler is not None:
        state["lr_scheduler"] = lr_scheduler.state_dict()
    name = _model_ckpt_name(path, iteration)
    _ensure_dir(name)
    torch.save(state, name)


def _split_moe_state_dict(full_state: Dict, num_local_experts: int,
                          ep_rank: int):
    """(experts by GLOBAL id, non-expert state)
    (reference: checkpointing.py:230-258)."""
    experts = defaultdict(dict)
    moe_keys = [k for k in full_state
                if "expert" in k and "gate.wg.weight" not in k]
    for key in moe_keys:
        m = re.match(".*{}([0-9]+).*".format(re.escape(_MOE_PREFIX)), key)
        if not m:
            logger.warning("no expert id found in key %s", key)
            continue
        local_id = int(m.group(1))
        global_id = ep_rank * num_local_experts + local_id
        gkey = key.replace("%s%d" % (_MOE_PREFIX, local_id),
                           "%s%d" % (_MOE_PREFIX, global_id))
        experts[str(global_id)][gkey] = full_state.pop(key)
    return experts, full_state


def _save_moe_checkpoint(iteration, path, num_experts, model, optimizer,
                         lr_scheduler):
    ep_rank = _rank()
    num_local = num_experts // _world()
    experts_state, dense_state = _split_moe_state_dict(
        dict(model.state_dict()), num_local, ep_rank)

    for gid, est in experts_state.items():
        name = _expert_ckpt_name(path, gid, iteration)
        _ensure_dir(name)
        torch.save(est, name)

    opt_name = _optimizer_ckpt_name(path, iteration, ep_rank)
    _ensure_dir(opt_name)
    torch.save({"optimizer":
                optimizer.state_dict() if optimizer else None}, opt_name)

    if ep_rank == 0:
        state = {"iteration": iteration, "model": dense_state}
        if lr_scheduler is not None:
            state["lr_scheduler"] = lr_scheduler.state_dict()
        name = _model_ckpt_name(path, iteration)
        _ensure_dir(name)
        torch.save(state, name)


# ---------------------------------------------------------------------------


def load_checkpoint(checkpoints_path: str, model,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    lr_scheduler=None, strict: bool = True) -> int:
    """Load the latest checkpoint; returns its iteration (0 if none)."""
    model = _unwrap(model)
    tracker = _tracker_filename(checkpoints_path)
    if not os.path.isfile(tracker):
        logger.warning("no checkpoint tracker at %s", tracker)
        return 0
    iteration, release = _read_metadata(tracker)
    _load_checkpoint(iteration, checkpoints_path, model, optimizer,
                     lr_scheduler, strict)
    logger.info("loaded checkpoint iteration %d from %s", iteration,
                checkpoints_path)
    return iteration


def _load_checkpoint(iteration, path, model, optimizer, lr_scheduler,
                     strict):
    ep_rank = _rank()
    ckpt = torch.load(_model_ckpt_name(path, iteration),
                      map_location="cpu", weights_only=False)
    has_moe, num_experts = _has_moe_layers(model)
    if has_moe:
        num_local = num_experts // _world()
        _merge_moe_state(path, iteration, num_local, ep_rank,
                         ckpt["model"])
    model.load_state_dict(ckpt["model"], strict=strict)

    if optimizer is not None:
        if has_moe:
            opt_ckpt = torch.load(
                _optimizer_ckpt_name(path, iteration, ep_rank),
                map_location="cpu", weights_only=False)
        else:
            opt_ckpt = ckpt
        if opt_ckpt.get("optimizer") is not None:
            optimizer.load_state_dict(opt_ckpt["optimizer"])
    if lr_scheduler is not None and "lr_scheduler" in ckpt:
        lr_scheduler.load_state_dict(ckpt["lr_scheduler"])


def _merge_moe_state(path, iteration, num_local_experts, ep_rank,
                     state_dict):
    """Load this rank's experts, remapping global -> local ids
    (reference: checkpointing.py:341-363)."""
    for local_id in range(num_local_experts):
        global_id = ep_rank * num_local_experts + local_id
        est = torch.load(
            _expert_ckpt_name(path, str(global_id), iteration),
            map_location="cpu", weights_only=False)
        for key in list(est.keys()):
            lkey = key.replace("%s%d" % (_MOE_PREFIX, global_id),
                               "%s%d" % (_MOE_PREFIX, local_id))
            est[lkey] = est.pop(key)
        state_dict.update(est)
