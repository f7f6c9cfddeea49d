"""Per-task checkpoint / resume — a capability the reference lacks entirely
(no torch.save/load anywhere; a crash in task 4 restarts at task 0).

Snapshot at each task boundary: backbone + head widths + heads, optimizer,
scheduler, rehearsal memory contents, task counters, acc1s trajectory, RNG states.
Resume restores the exact task boundary on every rank.
"""

import os
import random

import numpy as np
import torch

from ..distributed.init import is_main_process, barrier


def save_task_checkpoint(path_dir, task_id, model, memory, acc1s, args,
                         optimizer=None, scheduler=None):
    if not path_dir:
        return None
    os.makedirs(path_dir, exist_ok=True)
    path = os.path.join(path_dir, f"task_{task_id}.pth")
    if is_main_process():
        state = {
            "task_id": task_id,
            "known_classes": getattr(args, "known_classes", 0),
            "nb_classes": getattr(args, "nb_classes", 0),
            "class_order": getattr(args, "class_order", None),
            "head_widths": [h.out_features for h in model.fc.heads],
            "model": {k: v.cpu() for k, v in model.state_dict().items()},
            "memory": memory.state_dict() if memory is not None else None,
            "acc1s": list(acc1s),
            "optimizer": optimizer.state_dict() if optimizer is not None else None,
            "scheduler": scheduler.state_dict() if scheduler is not None else None,
            "rng": {
                "torch": torch.get_rng_state(),
                "cuda": (torch.cuda.get_rng_state_all()
                         if torch.cuda.is_available() else None),
                "numpy": np.random.get_state(),
                "python": random.getstate(),
            },
        }
        torch.save(state, path)
    barrier()
    return path


def load_task_checkpoint(path, model, memory, args, restore_rng=True):
    """Rebuild head structure, load weights/memory/counters. Returns the state
    dict (caller reads task_id / acc1s to continue the task loop)."""
    state = torch.load(path, map_location="cpu", weights_only=False)
    # grow the classifier to the checkpointed shape before loading weights
    for width in state["head_widths"]:
        model.prev_model_adaption(width)
    model.load_state_dict(state["model"])
    if memory is not None and state["memory"] is not None:
        memory.load_state_dict(state["memory"])
    args.task_id = state["task_id"]
    args.known_classes = state["known_classes"]
    args.nb_classes = state["nb_classes"]
    if state["class_order"] is not None:
        args.class_order = state["class_order"]
    if restore_rng:
        torch.set_rng_state(state["rng"]["torch"])
        if torch.cuda.is_available() and state["rng"]["cuda"] is not None:
            try:
                torch.cuda.set_rng_state_all(state["rng"]["cuda"])
            except RuntimeError:
                pass  # different GPU count than the saving run
        np.random.set_state(state["rng"]["numpy"])
        random.setstate(state["rng"]["python"])
    return state
