"""Checkpoint I/O.

Three jobs (the reference only had the first — SURVEY §5 'Checkpoint'):

1. **Load** the reference whitening-ResNet50 layout:
   ``{'state_dict': {'module.<key>': tensor}}`` with whitening stats under
   ``*.wh.running_{mean,variance}``, shared affines under ``*.{gamma,beta}``
   and BN stats under ``*.running_{mean,var}`` / ``*.{weight,bias}``
   (resnet50_dwt_mec_officehome.py:365-378, 466-479).

2. **Save/resume** full training state (model + optimizer + scheduler + EMA
   buffers + iteration) — a capability the reference lacks entirely.

3. **Export** a trained model back into the reference layout so reference
   tooling can consume our checkpoints.
"""
from __future__ import annotations

import os
from typing import Dict, Optional

import torch

MODULE_PREFIX = "module."


def load_reference_state_dict(path: str, device=None) -> Dict[str, torch.Tensor]:
    blob = torch.load(path, map_location=device if device is not None else "cpu",
                      weights_only=False)
    sd = blob["state_dict"] if isinstance(blob, dict) and "state_dict" in blob else blob
    out = {}
    for k, v in sd.items():
        out[k[len(MODULE_PREFIX):] if k.startswith(MODULE_PREFIX) else k] = v
    return out


def compute_bn_stats(state_dict: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Keep the norm-related entries ('bn' or 'downsample' in the key), like
    the reference's compute_bn_stats (resnet50_dwt_mec_officehome.py:466-479)."""
    return {k: v for k, v in state_dict.items()
            if ("bn" in k) or ("downsample" in k)}


# ---------------------------------------------------------------------------
# Training-state save / resume (new capability)
# ---------------------------------------------------------------------------


def save_training_state(path: str, model, optimizer=None, scheduler=None,
                        iteration: int = 0, extra: Optional[dict] = None) -> None:
    state = {
        "format": "dwt_amd.v1",
        "iteration": iteration,
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "scheduler": scheduler.state_dict() if scheduler is not None else None,
        "extra": extra or {},
    }
    tmp = path + ".tmp"
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    torch.save(state, tmp)
    os.replace(tmp, path)


def load_training_state(path: str, model, optimizer=None, scheduler=None,
                        map_location="cpu") -> int:
    state = torch.load(path, map_location=map_location, weights_only=False)
    assert state.get("format") == "dwt_amd.v1", "not a dwt_amd training checkpoint"
    model.load_state_dict(state["model"])
    if optimizer is not None and state["optimizer"] is not None:
        optimizer.load_state_dict(state["optimizer"])
    if scheduler is not None and state["scheduler"] is not None:
        scheduler.load_state_dict(state["scheduler"])
    return int(state["iteration"])


# ---------------------------------------------------------------------------
# Export to the reference layout
# ---------------------------------------------------------------------------


def export_reference_checkpoint(model, path: Optional[str] = None) -> dict:
    """Serialize a ResNetDWT into the reference §3.4 key layout.

    Norm sites collapse to single-branch keys using the *target* branch
    (`bnt*`) statistics — the branch the reference evaluates with.
    """
    sd = model.state_dict()
    out: Dict[str, torch.Tensor] = {}
    for k, v in sd.items():
        v = v.detach().cpu().clone()
        if ".bns" in k or "_aug" in k or ".downsample_bns" in k:
            continue  # only the target branch is exported
        nk = k
        # 'bnt1.wh.running_mean' -> 'bn1.wh.running_mean'
        nk = nk.replace("downsample_bnt.", "downsample_bn.")
        import re
        nk = re.sub(r"\bbnt(\d)\.", r"bn\1.", nk)
        # shared affines: gammaN/betaN -> bnN.gamma/beta (wh sites) is handled
        # by layer position; keep direct names for BN weight/bias
        out[MODULE_PREFIX + nk] = v

    # shared gamma/beta parameters -> per-site names
    def move(src, dst):
        if MODULE_PREFIX + src in out:
            out[MODULE_PREFIX + dst] = out.pop(MODULE_PREFIX + src)

    def site_names(prefix, idx, is_wh):
        g, b = f"{prefix}gamma{idx}", f"{prefix}beta{idx}"
        if is_wh:
            move(g, f"{prefix}bn{idx}.gamma")
            move(b, f"{prefix}bn{idx}.beta")
        else:
            # BN affines stored flat (C,) in the reference layout
            gk, bk = MODULE_PREFIX + g, MODULE_PREFIX + b
            if gk in out:
                out[MODULE_PREFIX + f"{prefix}bn{idx}.weight"] = out.pop(gk).reshape(-1)
                out[MODULE_PREFIX + f"{prefix}bn{idx}.bias"] = out.pop(bk).reshape(-1)

    site_names("", 1, True)  # stem
    layer_blocks = {1: 3, 2: 4, 3: 6, 4: 3}
    for layer, blocks in layer_blocks.items():
        is_wh = layer == 1
        for i in range(blocks):
            for idx in (1, 2, 3):
                site_names(f"layer{layer}.{i}.", idx, is_wh)
            dg = f"layer{layer}.{i}.downsample_gamma"
            db = f"layer{layer}.{i}.downsample_beta"
            if MODULE_PREFIX + dg in out:
                if is_wh:
                    move(dg, f"layer{layer}.{i}.downsample_bn.gamma")
                    move(db, f"layer{layer}.{i}.downsample_bn.beta")
                else:
                    out[MODULE_PREFIX + f"layer{layer}.{i}.downsample_bn.weight"] = \
                        out.pop(MODULE_PREFIX + dg).reshape(-1)
                    out[MODULE_PREFIX + f"layer{layer}.{i}.downsample_bn.bias"] = \
                        out.pop(MODULE_PREFIX + db).reshape(-1)

    blob = {"state_dict": out}
    if path is not None:
        torch.save(blob, path)
    return blob
