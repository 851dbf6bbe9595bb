"""Checkpoint IO in the reference's ``model_ChebConv_*`` directory layout.

The reference saves Keras weights to
``model/model_ChebConv_<training_set>_a5_c5_ACO_agent/cp-<epoch:04d>.ckpt``
plus a ``checkpoint`` manifest, and resumes from the latest entry
(``gnn_offloading_agent.py:125-132``, ``AdHoc_train.py:59,204-206``).

We keep the directory and file naming and the float64 tensor layout
(5 layers × {kernel (K, F_in, F_out), bias (F_out)}), stored as an ``.npz``
sidecar ``cp-XXXX.ckpt.npz`` with TF-style variable names, and write the same
``checkpoint`` manifest so the latest-checkpoint protocol is identical.
"""

from __future__ import annotations

import os
import re
from typing import Optional

import numpy as np
import torch


def model_dir(model_root: str, training_set: str) -> str:
    return os.path.join(model_root,
                        f"model_ChebConv_{training_set}_a5_c5_ACO_agent")


def _var_names(i):
    # TF/Keras variable naming for the i-th ChebConv layer
    suffix = "" if i == 0 else f"_{i}"
    return (f"cheb_conv{suffix}/kernel:0", f"cheb_conv{suffix}/bias:0")


def save(model: torch.nn.Module, ckpt_path: str):
    """``ckpt_path`` like ``.../cp-0003.ckpt`` (TF-style, extensionless)."""
    os.makedirs(os.path.dirname(ckpt_path), exist_ok=True)
    arrays = {}
    for i, layer in enumerate(model.layers):
        kname, bname = _var_names(i)
        arrays[kname] = layer.weight.detach().cpu().double().numpy()
        arrays[bname] = layer.bias.detach().cpu().double().numpy()
    np.savez(ckpt_path + ".npz", **arrays)
    base = os.path.basename(ckpt_path)
    with open(os.path.join(os.path.dirname(ckpt_path), "checkpoint"), "w") as f:
        f.write(f'model_checkpoint_path: "{base}"\n')
        f.write(f'all_model_checkpoint_paths: "{base}"\n')


def latest_checkpoint(directory: str) -> Optional[str]:
    manifest = os.path.join(directory, "checkpoint")
    if os.path.isfile(manifest):
        with open(manifest) as f:
            m = re.search(r'model_checkpoint_path:\s*"([^"]+)"', f.read())
        if m:
            cand = os.path.join(directory, m.group(1))
            if os.path.isfile(cand + ".npz"):
                return cand
    if not os.path.isdir(directory):
        return None
    cps = sorted(f[:-len(".npz")] for f in os.listdir(directory)
                 if f.startswith("cp-") and f.endswith(".ckpt.npz"))
    return os.path.join(directory, cps[-1]) if cps else None


def load(model: torch.nn.Module, ckpt_path: str):
    data = np.load(ckpt_path + ".npz")
    with torch.no_grad():
        for i, layer in enumerate(model.layers):
            kname, bname = _var_names(i)
            k = torch.from_numpy(data[kname])
            b = torch.from_numpy(data[bname])
            if k.shape != layer.weight.shape:
                raise ValueError(
                    f"checkpoint kernel {i} shape {tuple(k.shape)} != model "
                    f"{tuple(layer.weight.shape)} (check --K / --num_layer)")
            layer.weight.copy_(k.to(layer.weight.dtype))
            layer.bias.copy_(b.to(layer.bias.dtype))
