"""Per-round model checkpointing.

Keeps the reference's templated checkpoint naming contract
(utils_run_task.py:327-397): the aggregated global model of round r is
persisted under ``model_update_style`` with ``{task_id}`` and
``{current_round}`` substituted — default
``{task_id}_{current_round}_result_model.safetensors`` (the reference's
default suffix is .mnn, a phone-side format; payload here is
safetensors).  Round r>0 resumes by loading round r-1's artifact.
"""

from __future__ import annotations

import os
from typing import Dict, Optional

import torch

DEFAULT_STYLE = "{task_id}_{current_round}_result_model.safetensors"


def checkpoint_name(task_id: str, current_round: int,
                    model_update_style: str = "") -> str:
    style = model_update_style or DEFAULT_STYLE
    name = style.format(task_id=task_id, current_round=current_round)
    # the rendered name is joined into the checkpoint directory: a
    # template (or task_id) carrying separators or '..' must not be
    # able to address files outside it (validate.py rejects these at
    # submit time; this guards direct engine use)
    if os.sep in name or (os.altsep and os.altsep in name) or ".." in name:
        raise ValueError(f"unsafe checkpoint name {name!r}")
    return name


def save_checkpoint(directory: str, task_id: str, current_round: int,
                    state: Dict[str, torch.Tensor],
                    model_update_style: str = "") -> str:
    from safetensors.torch import save_file
    os.makedirs(directory, exist_ok=True)
    path = os.path.join(directory, checkpoint_name(
        task_id, current_round, model_update_style))
    save_file({k: v.detach().cpu().contiguous() for k, v in state.items()}, path)
    return path


def latest_round(directory: str, task_id: str,
                 model_update_style: str = "") -> int:
    """Highest round with a saved artifact, or -1 (used for crash
    resume: the reference's actors fetch round r-1's model when
    round > 0, utils_run_task.py:327-397)."""
    import re
    style = model_update_style or DEFAULT_STYLE
    best = -1
    if not os.path.isdir(directory):
        return best
    pat = re.escape(style).replace(
        re.escape("{task_id}"), re.escape(task_id)).replace(
        re.escape("{current_round}"), r"(\d+)")
    rx = re.compile("^" + pat + "$")
    for f in os.listdir(directory):
        m = rx.match(f)
        if m:
            best = max(best, int(m.group(1)))
    return best


def load_checkpoint(directory: str, task_id: str, current_round: int,
                    model_update_style: str = "",
                    device: str = "cpu") -> Optional[Dict[str, torch.Tensor]]:
    from safetensors.torch import load_file
    path = os.path.join(directory, checkpoint_name(
        task_id, current_round, model_update_style))
    if not os.path.exists(path):
        return None
    return load_file(path, device=device)
