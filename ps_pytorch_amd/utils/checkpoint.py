"""Checkpoint layout compatibility: train_dir/model_step_<k> state_dicts.

Reference parity: sync_replicas_master_nn.py:264-270 / distributed_worker.py:
301-307 torch.save(state_dict) to an NFS dir the evaluator polls
(distributed_evaluator.py:79-88). Same file naming and payload here.
"""
from __future__ import annotations

import os
from pathlib import Path

import torch
import torch.nn as nn


def model_step_path(train_dir: str, step: int) -> str:
    return os.path.join(train_dir, f"model_step_{step}")


def save_model_step(model_or_sd, train_dir: str, step: int) -> str:
    Path(train_dir).mkdir(parents=True, exist_ok=True)
    sd = (model_or_sd.state_dict() if isinstance(model_or_sd, nn.Module)
          else model_or_sd)
    # checkpoints are f32 regardless of compute dtype
    sd = {k: (v.to(torch.float32) if torch.is_floating_point(v) else v)
          for k, v in sd.items()}
    path = model_step_path(train_dir, step)
    tmp = path + ".tmp"
    torch.save(sd, tmp)
    os.replace(tmp, path)   # atomic: the polling evaluator never sees partial files
    return path


def load_model_step(model: nn.Module, train_dir: str, step: int,
                    strict: bool = False) -> nn.Module:
    sd = torch.load(model_step_path(train_dir, step), map_location='cpu',
                    weights_only=True)
    model.load_state_dict(sd, strict=strict)
    return model
