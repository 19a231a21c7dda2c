"""Checkpoint round-trip — format-compatible with the reference
(SURVEY.md section 3.5): dict {epoch, arch, state_dict (module.-prefixed),
best_acc1, optimizer} saved as checkpoint.pth.tar, best copied to
model_best.pth.tar."""


import torch

from ..utils.utils import save_checkpoint


def save_state(model, optimizer, epoch, arch, best_acc1, is_best, save_path):
    state = {
        "epoch": epoch + 1,
        "arch": arch,
        "state_dict": model.state_dict(),
        "best_acc1": best_acc1,
        "optimizer": optimizer.state_dict(),
    }
    save_checkpoint(state, is_best, save_path)


def load_state(path, model, optimizer=None, map_location="cpu",
               reset_resume=False):
    """Load a reference-format checkpoint.  Returns (start_epoch, best_acc1).

    ``reset_resume`` loads weights only (ref:train.py:355 — used between
    the recipe's two training steps).
    """
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    state_dict = ckpt.get("state_dict", ckpt)
    model_keys = set(model.state_dict().keys())
    ck_keys = set(state_dict.keys())
    if model_keys != ck_keys:
        # tolerate a module.-prefix mismatch in either direction
        if all(k.startswith("module.") for k in ck_keys) and not any(
                k.startswith("module.") for k in model_keys):
            state_dict = {k[len("module."):]: v for k, v in state_dict.items()}
        elif all(k.startswith("module.") for k in model_keys) and not any(
                k.startswith("module.") for k in ck_keys):
            state_dict = {"module." + k: v for k, v in state_dict.items()}
    model.load_state_dict(state_dict)
    if reset_resume:
        return 0, 0.0
    if optimizer is not None and "optimizer" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer"])
    best = ckpt.get("best_acc1", 0.0)
    if isinstance(best, torch.Tensor):
        best = best.item()
    return ckpt.get("epoch", 0), float(best)
