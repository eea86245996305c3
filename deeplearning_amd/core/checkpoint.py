"""Checkpoint save/load/auto-resume.

Reference parity (SURVEY.md §5 Checkpoint/resume): both layouts —
(a) weights-only model_{epoch}.pth + best_model.pth (mnist/train.py:162-165),
(b) full state dict {model, optimizer, lr_scheduler, epoch, scaler, max_accuracy}
(swin utils/torch_utils.py:233-246) with auto-resume scan (:261-271),
`module.` strip for DDP (MAE utils/utils.py translate_state_dict),
partial/pretrained load with key filtering + strict=False (mnist/train.py:110-117).
"""
from __future__ import annotations

import glob
import os
import re
from pathlib import Path

import torch


def strip_module_prefix(state_dict: dict) -> dict:
    return { (k[7:] if k.startswith("module.") else k): v for k, v in state_dict.items() }


def unwrap_model(model):
    return model.module if hasattr(model, "module") else model


def save_weights(model, path: str | Path) -> None:
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    torch.save(unwrap_model(model).state_dict(), path)


def save_checkpoint(path: str | Path, model, optimizer=None, lr_scheduler=None,
                    epoch: int = 0, scaler=None, ema=None, **extra) -> None:
    """Full training state (reference layout (b))."""
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    state = {"model": unwrap_model(model).state_dict(), "epoch": epoch}
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if lr_scheduler is not None:
        state["lr_scheduler"] = lr_scheduler.state_dict()
    if scaler is not None:
        state["scaler"] = scaler.state_dict()
    if ema is not None:
        state["ema"] = ema.state_dict() if hasattr(ema, "state_dict") else ema
    state.update(extra)
    torch.save(state, path)


def load_checkpoint(path: str | Path, model, optimizer=None, lr_scheduler=None,
                    scaler=None, map_location: str = "cpu") -> dict:
    """Restore full training state; returns the raw dict (epoch etc.)."""
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    state = ckpt.get("model", ckpt)
    unwrap_model(model).load_state_dict(strip_module_prefix(state))
    if optimizer is not None and "optimizer" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer"])
    if lr_scheduler is not None and "lr_scheduler" in ckpt:
        lr_scheduler.load_state_dict(ckpt["lr_scheduler"])
    if scaler is not None and "scaler" in ckpt:
        scaler.load_state_dict(ckpt["scaler"])
    return ckpt


def load_pretrained(model, path: str | Path, map_location: str = "cpu",
                    skip_mismatch: bool = True, logger=None) -> list:
    """Partial load: drop keys missing or shape-mismatched (reference
    mnist/train.py:110-117, train_with_DDP:167-169). Returns dropped keys."""
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    state = strip_module_prefix(ckpt.get("model", ckpt))
    state = interpolate_rel_pos_tables(state, model)
    own = unwrap_model(model).state_dict()
    dropped = []
    filtered = {}
    for k, v in state.items():
        if k in own and own[k].shape == v.shape:
            filtered[k] = v
        else:
            dropped.append(k)
    if dropped and not skip_mismatch:
        raise RuntimeError(f"pretrained load mismatch: {dropped}")
    unwrap_model(model).load_state_dict(filtered, strict=False)
    if logger and dropped:
        logger.info(f"load_pretrained: dropped {len(dropped)} keys: {dropped[:8]}...")
    return dropped


def auto_resume_helper(output_dir: str | Path, prefix: str = "ckpt_epoch_") -> str | None:
    """Find latest ckpt_epoch_N.pth in output dir (reference swin
    utils/torch_utils.py:261-271)."""
    files = glob.glob(os.path.join(str(output_dir), f"{prefix}*.pth"))
    if not files:
        return None

    def epoch_of(f):
        m = re.search(rf"{re.escape(prefix)}(\d+)\.pth$", f)
        return int(m.group(1)) if m else -1

    return max(files, key=epoch_of)


def interpolate_rel_pos_tables(state_dict: dict, model) -> dict:
    """Shape-aware load: bicubic-resize Swin relative_position_bias_table
    entries whose window size differs, and drop stale relative_position_index
    buffers (reference swin load_pretrained, utils/torch_utils.py:143-231)."""
    import torch.nn.functional as F

    own = unwrap_model(model).state_dict()
    out = dict(state_dict)
    for k in list(out.keys()):
        if k.endswith("relative_position_index"):
            out.pop(k)  # buffer, recomputed at construction
            continue
        if not k.endswith("relative_position_bias_table") or k not in own:
            continue
        src, dst = out[k], own[k]
        if src.shape == dst.shape:
            continue
        L1, heads = src.shape
        L2 = dst.shape[0]
        s1 = int(L1 ** 0.5)
        s2 = int(L2 ** 0.5)
        table = src.permute(1, 0).reshape(1, heads, s1, s1)
        table = F.interpolate(table, size=(s2, s2), mode="bicubic",
                              align_corners=False)
        out[k] = table.reshape(heads, L2).permute(1, 0)
    return out
