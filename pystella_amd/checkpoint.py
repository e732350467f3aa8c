"""Field-state checkpoint / restore.

The reference has no true checkpointing: its persistence is
HDF5 time-series appends, and ``gather_array``/``scatter_array`` are the
building blocks a checkpoint would use (reference decomp.py:536-722).
This module provides both forms:

* ``mode="gather"`` — rank 0 writes one file containing global interior
  arrays (portable across different proc_shapes on restore);
* ``mode="shard"`` — every rank writes its local pencil (fast path for
  same-topology restarts; one file per rank).

Arrays may be halo-padded; halos are stripped on save and re-shared on
load.
"""

from __future__ import annotations

import os

import torch

__all__ = ["save_checkpoint", "load_checkpoint"]


def _is_padded(t, decomp):
    h = decomp.halo_shape
    if decomp.rank_shape is None:
        return False
    return tuple(t.shape[-3:]) == tuple(
        n + 2 * hh for n, hh in zip(decomp.rank_shape, h))


def save_checkpoint(path, decomp, arrays, attrs=None, mode="gather"):
    """Write a checkpoint of ``arrays`` (dict name → tensor)."""
    meta = {"attrs": dict(attrs or {}), "mode": mode,
            "proc_shape": decomp.proc_shape,
            "grid_shape": decomp.grid_shape,
            "padded": {}}
    if mode == "shard":
        payload = {}
        for name, t in arrays.items():
            padded = _is_padded(t, decomp)
            meta["padded"][name] = padded
            payload[name] = t.cpu()
        torch.save({"meta": meta, "arrays": payload},
                   f"{path}.rank{decomp.rank}.pt")
        decomp.barrier()
        return

    payload = {}
    for name, t in arrays.items():
        padded = _is_padded(t, decomp)
        meta["padded"][name] = padded
        interior = decomp.remove_halos(t) if padded else t
        full = decomp.gather_array(interior.contiguous())
        if decomp.rank == 0:
            payload[name] = full.cpu()
    if decomp.rank == 0:
        tmp = f"{path}.tmp"
        torch.save({"meta": meta, "arrays": payload}, tmp)
        os.replace(tmp, path)
    decomp.barrier()


def load_checkpoint(path, decomp, arrays):
    """Restore ``arrays`` (dict name → preallocated tensor) in place."""
    shard_path = f"{path}.rank{decomp.rank}.pt"
    if os.path.exists(shard_path):
        blob = torch.load(shard_path, weights_only=False)
        for name, t in arrays.items():
            t.copy_(blob["arrays"][name].to(t.device))
            if blob["meta"]["padded"].get(name):
                decomp.share_halos(t)
        return blob["meta"]["attrs"]

    blob = None
    meta = None
    if decomp.rank == 0:
        blob = torch.load(path, weights_only=False)
        meta = blob["meta"]
    meta = decomp.bcast(meta, root=0)
    for name, t in arrays.items():
        full = blob["arrays"][name].to(t.device) if decomp.rank == 0 \
            else None
        template = full if decomp.rank == 0 else t
        piece = decomp.scatter_array(template)
        if meta["padded"].get(name):
            decomp.restore_halos(t, piece.to(t.device))
            decomp.share_halos(t)
        else:
            t.copy_(piece.to(t.device))
    return meta["attrs"]
