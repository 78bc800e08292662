"""Checkpoint I/O — byte-format parity with the reference:

1. IMPALA format: a single ``model.tar`` torch.save of
   ``{"model_state_dict", "optimizer_state_dict", "hparam"}``
   (reference: impala_atari.py:496-515).  Unlike the reference, a load path
   exists.
2. Agent format: ``{"actor_state_dict", "actor_target_state_dict",
   "optimizer_state_dict"}`` (reference: dqn_agent.py:210-233).

Both writers are atomic (tmp + rename) and both loaders map to an explicit
device.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch


def _atomic_save(obj: Dict[str, Any], path: str) -> None:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    tmp = path + ".tmp"
    torch.save(obj, tmp)
    os.replace(tmp, path)


def save_checkpoint(path: str, *, model=None, optimizer=None,
                    hparam: Optional[Dict[str, Any]] = None,
                    extra: Optional[Dict[str, Any]] = None) -> None:
    """IMPALA-format checkpoint (keys as impala_atari.py:503-510)."""
    ckpt: Dict[str, Any] = {
        "model_state_dict": model.state_dict() if model is not None else {},
        "optimizer_state_dict": optimizer.state_dict() if optimizer is not None else {},
        "hparam": hparam or {},
    }
    if extra:
        ckpt.update(extra)
    _atomic_save(ckpt, path)


def load_checkpoint(path: str, *, model=None, optimizer=None,
                    map_location="cpu") -> Dict[str, Any]:
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    if model is not None and ckpt.get("model_state_dict"):
        model.load_state_dict(ckpt["model_state_dict"])
    if optimizer is not None and ckpt.get("optimizer_state_dict"):
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
    return ckpt


def save_agent_checkpoint(path: str, *, actor, actor_target=None,
                          optimizer=None, extra: Optional[Dict[str, Any]] = None) -> None:
    """Agent-format checkpoint (keys as dqn_agent.py:210-222)."""
    ckpt: Dict[str, Any] = {"actor_state_dict": actor.state_dict()}
    if actor_target is not None:
        ckpt["actor_target_state_dict"] = actor_target.state_dict()
    if optimizer is not None:
        ckpt["optimizer_state_dict"] = optimizer.state_dict()
    if extra:
        ckpt.update(extra)
    _atomic_save(ckpt, path)


def load_agent_checkpoint(path: str, *, actor=None, actor_target=None,
                          optimizer=None, map_location="cpu") -> Dict[str, Any]:
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    if actor is not None and "actor_state_dict" in ckpt:
        actor.load_state_dict(ckpt["actor_state_dict"])
    if actor_target is not None and "actor_target_state_dict" in ckpt:
        actor_target.load_state_dict(ckpt["actor_target_state_dict"])
    if optimizer is not None and "optimizer_state_dict" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
    return ckpt
