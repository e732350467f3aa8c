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

import json
import os

import numpy as np
import torch

__all__ = ["save_checkpoint", "load_checkpoint"]


def _h5_save(path, meta, payload):
    """Write a checkpoint as an HDF5 file (fields under /fields,
    metadata as root attrs) via the self-contained writer — the same
    format family as OutputFile (reference output.py's h5py layout)."""
    from pystella_amd.hdf5 import File
    tmp = f"{path}.tmp"
    with File(tmp) as f:
        f.attrs["mode"] = meta["mode"]
        f.attrs["proc_shape"] = np.asarray(meta["proc_shape"],
                                           dtype=np.int64)
        if meta.get("grid_shape"):
            f.attrs["grid_shape"] = np.asarray(meta["grid_shape"],
                                               dtype=np.int64)
        f.attrs["padded"] = json.dumps(meta["padded"])
        f.attrs["user_attrs"] = json.dumps(meta["attrs"], default=str)
        for name, t in payload.items():
            f.create_dataset(f"fields/{name}", t.numpy())
    os.replace(tmp, path)


def _h5_load(path):
    """Read a checkpoint written by :func:`_h5_save` (h5py if present,
    else the self-contained reader)."""
    try:
        import h5py
        with h5py.File(path, "r") as f:
            meta = {"mode": f.attrs["mode"],
                    "padded": json.loads(f.attrs["padded"]),
                    "attrs": json.loads(f.attrs["user_attrs"])}
            arrays = {k: torch.as_tensor(np.asarray(v))
                      for k, v in f["fields"].items()}
            return meta, arrays
    except ImportError:
        pass
    from pystella_amd.hdf5 import read_file
    tree = read_file(path)
    meta = {"mode": tree["attrs"]["mode"],
            "padded": json.loads(tree["attrs"]["padded"]),
            "attrs": json.loads(tree["attrs"]["user_attrs"])}
    arrays = {k: torch.as_tensor(v["data"].copy())
              for k, v in tree["children"]["fields"]["children"].items()}
    return meta, arrays


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
    h5 = str(path).endswith(".h5")
    if mode == "shard":
        payload = {}
        for name, t in arrays.items():
            padded = _is_padded(t, decomp)
            meta["padded"][name] = padded
            payload[name] = t.cpu()
        if h5:
            _h5_save(f"{path}.rank{decomp.rank}", meta, payload)
        else:
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
        if h5:
            _h5_save(path, meta, payload)
        else:
            tmp = f"{path}.tmp"
            torch.save({"meta": meta, "arrays": payload}, tmp)
            os.replace(tmp, path)
    decomp.barrier()


def load_checkpoint(path, decomp, arrays):
    """Restore ``arrays`` (dict name → preallocated tensor) in place."""
    h5 = str(path).endswith(".h5")
    shard_path = (f"{path}.rank{decomp.rank}" if h5
                  else f"{path}.rank{decomp.rank}.pt")
    if os.path.exists(shard_path):
        if h5:
            smeta, sarrays = _h5_load(shard_path)
        else:
            blob = torch.load(shard_path, weights_only=False)
            smeta, sarrays = blob["meta"], blob["arrays"]
        for name, t in arrays.items():
            t.copy_(sarrays[name].to(t.device))
            if smeta["padded"].get(name):
                decomp.share_halos(t)
        return smeta["attrs"]

    garrays = None
    meta = None
    if decomp.rank == 0:
        if h5:
            meta, garrays = _h5_load(path)
        else:
            blob = torch.load(path, weights_only=False)
            meta, garrays = blob["meta"], blob["arrays"]
    meta = decomp.bcast(meta, root=0)
    for name, t in arrays.items():
        full = garrays[name].to(t.device) if decomp.rank == 0 else None
        template = full if decomp.rank == 0 else t
        piece = decomp.scatter_array(template)
        if meta["padded"].get(name):
            decomp.restore_halos(t, piece.to(t.device))
            decomp.share_halos(t)
        else:
            t.copy_(piece.to(t.device))
    return meta["attrs"]
