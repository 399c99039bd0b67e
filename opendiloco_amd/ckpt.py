"""Checkpoint save/load/resume/GC.

Mirrors the reference's ckpt_utils.py structure and discovery semantics:
  - checkpoint dirs named ``model_step_{N}`` (CKPT_PREFIX, ckpt_utils.py:13)
  - per-DiLoCo-rank subdirs ``diloco_rank_{r}`` (:196-197)
  - ``global_state_dict.pt`` holding scheduler/outer-optimizer/scaler/loss
    (:92-100)
  - per-rank dataloader state ``__{rank}_0.pt`` (:83-87)
  - resume discovery by highest step (:23-45) and top-k GC (:170-179).

Deviation (documented in DESIGN.md): the model+inner-optimizer shard is a
plain ``torch.save`` file ``model_optim.pt`` instead of a torch-DCP
directory — in the MI355X mapping every DiLoCo worker is a world of exactly
one GPU (BASELINE.json configs[4]), so there is nothing to shard and DCP's
collective save would couple independent workers.
"""

from __future__ import annotations

import os
import shutil

import torch

GLOBAL_STATE_FILE = "global_state_dict.pt"
MODEL_OPTIM_FILE = "model_optim.pt"
CKPT_PREFIX = "model_step"


def get_diloco_rank_dir_name(world_rank_diloco: int) -> str:
    return f"diloco_rank_{world_rank_diloco}"


def filter_ckpt_files(f: str) -> bool:
    if CKPT_PREFIX not in f:
        return False
    try:
        int(f.split("_")[-1])
        return True
    except ValueError:
        return False


def get_resume_info(ckpt_config) -> tuple[bool, str | None]:
    """(should_resume, path) — ckpt_utils.py:23-45 semantics."""
    if ckpt_config.resume is None:
        return False, None
    if isinstance(ckpt_config.resume, bool):
        if not ckpt_config.resume:
            return False, None
        try:
            files = [os.path.join(ckpt_config.path, f) for f in os.listdir(ckpt_config.path)
                     if filter_ckpt_files(f)]
        except FileNotFoundError:
            return False, None
        if not files:
            return False, None
        return True, max(files, key=lambda f: int(f.split("_")[-1]))
    return True, ckpt_config.resume


def save_checkpoint(checkpoint_path: str, model, optimizer, scheduler,
                    outer_optimizer=None, scaler=None, loss: float | None = None,
                    data_loader=None, save_global_state: bool = True,
                    save_model_optim: bool = True, rank: int = 0):
    """ckpt_utils.py:48-100 analogue (see module docstring for the format).

    ``save_model_optim=False`` lets non-zero ranks of a replicated (non-hv)
    world skip the model_optim.pt write: the states are identical across
    ranks after the grad all-reduce, and concurrent writes of the same file
    could interleave/corrupt it.
    """
    os.makedirs(checkpoint_path, exist_ok=True)
    if save_model_optim:
        torch.save({
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict(),
        }, os.path.join(checkpoint_path, MODEL_OPTIM_FILE))
    if data_loader is not None:
        torch.save({"data_loader": data_loader.state_dict()},
                   os.path.join(checkpoint_path, f"__{rank}_0.pt"))
    if not save_global_state:
        return
    global_state = {"scheduler": scheduler.state_dict(), "loss": loss if loss is not None else 0}
    if outer_optimizer is not None:
        global_state["outer_optimizer"] = outer_optimizer.state_dict()
    if scaler is not None:
        global_state["scaler"] = scaler.state_dict()
    torch.save(global_state, os.path.join(checkpoint_path, GLOBAL_STATE_FILE))


def load_checkpoint(checkpoint_path: str, model, optimizer, scheduler=None,
                    outer_optimizer=None, scaler=None, data_loader=None,
                    rank: int = 0) -> float:
    """ckpt_utils.py:103-156 analogue; returns the checkpointed loss."""
    blob = torch.load(os.path.join(checkpoint_path, MODEL_OPTIM_FILE),
                      map_location="cpu", weights_only=False)
    model_sd = blob["model"]
    model.load_state_dict(model_sd)
    if optimizer is not None:
        optimizer.load_state_dict(blob["optimizer"])
    if data_loader is not None:
        rank_blob = torch.load(os.path.join(checkpoint_path, f"__{rank}_0.pt"),
                               map_location="cpu", weights_only=False)
        data_loader.load_state_dict(rank_blob["data_loader"])
    global_state = torch.load(os.path.join(checkpoint_path, GLOBAL_STATE_FILE),
                              map_location="cpu", weights_only=False)
    if scheduler is not None:
        scheduler.load_state_dict(global_state["scheduler"])
        optimizer.param_groups[0]["lr"] = scheduler.get_last_lr()[0]  # ckpt_utils.py:151
    if outer_optimizer is not None:
        outer_optimizer.load_state_dict(global_state["outer_optimizer"])
    if scaler is not None:
        scaler.load_state_dict(global_state["scaler"])
    from opendiloco_amd.model import bump_weights_version

    bump_weights_version()  # master weights changed in place
    return global_state["loss"]


def delete_old_checkpoints(checkpoint_path: str, topk: int) -> list[str]:
    """ckpt_utils.py:170-179."""
    files = [os.path.join(checkpoint_path, f) for f in os.listdir(checkpoint_path)
             if filter_ckpt_files(f)]
    files.sort(key=lambda x: int(x.split("_")[-1]))
    deleted = []
    for f in files[:-topk]:
        shutil.rmtree(f, ignore_errors=True)
        deleted.append(f)
    return deleted


def check_checkpoint_path_access(checkpoint_path: str, rank: int, world_rank_hv: int | None = None):
    """ckpt_utils.py:182-193."""
    if world_rank_hv:
        p = os.path.join(checkpoint_path, get_diloco_rank_dir_name(world_rank_hv), f"dummy_file_{rank}.txt")
    else:
        p = os.path.join(checkpoint_path, f"dummy_file_{rank}.txt")
    os.makedirs(os.path.dirname(p), exist_ok=True)
    with open(p, "w") as f:
        f.write("This is a dummy file for testing access.")
    os.remove(p)
