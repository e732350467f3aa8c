"""Run output: appendable time-series datasets + provenance capture.

Analogue of reference pystella/output.py:52-181 (``OutputFile`` over
h5py).  The default backend writes a REAL HDF5 file with the
reference's layout — through h5py when importable, otherwise through
the self-contained spec-subset writer (:mod:`pystella_amd.hdf5`; this
image ships no h5py, and the files open with stock h5py/libhdf5
elsewhere):

* ``out.output("energy", t=..., a=..., **components)`` appends one row
  per call to each named dataset under the group (h5py-resizable
  semantics; reference output.py:157-181);
* run provenance (hostname, device, argv, package versions, git
  revisions, the run script's text) is captured as root attributes,
  matching reference output.py:98-155.

``PYSTELLA_OUTPUT=dir`` selects the ``.npy``-directory store instead
(same append semantics, no binary format).
"""

from __future__ import annotations

import json
import os
import socket
import sys
import time

import numpy as np

__all__ = ["OutputFile"]


def get_versions(dependencies):
    versions = {}
    for dep in dependencies:
        try:
            mod = __import__(dep)
            versions[dep] = str(getattr(mod, "__version__", "unknown"))
        except ImportError:
            versions[dep] = "not installed"
    return versions


def _git_rev(path):
    import subprocess
    try:
        return subprocess.check_output(
            ["git", "rev-parse", "HEAD"], cwd=path,
            stderr=subprocess.DEVNULL).decode().strip()
    except Exception:
        return None


class _DirStore:
    """Append-mode dataset store over a directory of .npy stacks."""

    def __init__(self, path):
        self.path = path
        os.makedirs(path, exist_ok=True)
        self._cache = {}

    def set_attrs(self, attrs):
        with open(os.path.join(self.path, "attrs.json"), "w") as f:
            json.dump(attrs, f, indent=1, default=str)

    def append(self, group, name, value):
        key = (group, name)
        arr = np.asarray(value)
        buf = self._cache.setdefault(key, [])
        buf.append(arr)
        gdir = os.path.join(self.path, group)
        os.makedirs(gdir, exist_ok=True)
        np.save(os.path.join(gdir, name + ".npy"),
                np.stack(self._cache[key]))

    def read(self, group, name):
        return np.load(os.path.join(self.path, group, name + ".npy"))

    def close(self):
        pass


class _H5Store:
    def __init__(self, path):
        import h5py
        self.f = h5py.File(path, "a")

    def set_attrs(self, attrs):
        for k, v in attrs.items():
            try:
                self.f.attrs[k] = v
            except TypeError:
                self.f.attrs[k] = str(v)

    def append(self, group, name, value):
        arr = np.asarray(value)
        g = self.f.require_group(group)
        if name not in g:
            g.create_dataset(name, shape=(0,) + arr.shape,
                             maxshape=(None,) + arr.shape,
                             dtype=arr.dtype)
        ds = g[name]
        ds.resize(ds.shape[0] + 1, axis=0)
        ds[-1] = arr

    def read(self, group, name):
        return np.asarray(self.f[group][name])

    def close(self):
        self.f.close()


class _MiniH5Store:
    """HDF5 store over the self-contained writer
    (:mod:`pystella_amd.hdf5`) — same group/dataset/attribute layout as
    the h5py path, no libhdf5 needed.  The file is rewritten on every
    append (datasets are contiguous; time-series outputs are small)."""

    def __init__(self, path):
        from pystella_amd.hdf5 import File
        self.f = File(path)

    def set_attrs(self, attrs):
        for k, v in attrs.items():
            if isinstance(v, (dict, list, tuple)):
                import json as _json
                v = _json.dumps(v, default=str)
            self.f.attrs[k] = v
        self.f.flush()

    def append(self, group, name, value):
        self.f.require_group(group)
        self.f.append(f"{group}/{name}", np.asarray(value))
        self.f.flush()

    def read(self, group, name):
        from pystella_amd.hdf5 import read_file
        tree = read_file(self.f.filename)
        return np.asarray(
            tree["children"][group]["children"][name]["data"])

    def close(self):
        self.f.close()


class OutputFile:
    """Appendable run output with provenance attributes
    (reference output.py:52-181)."""

    def __init__(self, ctx=None, name=None, runfile=None, suffix="",
                 **kwargs):
        if name is None:
            stamp = time.strftime("%Y-%m-%d-%H%M%S")
            name = f"output-{stamp}{suffix}"

        attrs = {
            "hostname": socket.gethostname(),
            "argv": " ".join(sys.argv),
            "created": time.strftime("%Y-%m-%d %H:%M:%S"),
            "versions": get_versions(
                ["numpy", "scipy", "torch", "pystella_amd"]),
        }
        try:
            import torch
            if torch.cuda.is_available():
                attrs["device"] = torch.cuda.get_device_name(0)
        except Exception:
            pass
        rev = _git_rev(os.path.dirname(os.path.abspath(
            runfile or __file__)))
        if rev:
            attrs["git_rev"] = rev
        if runfile is not None:
            try:
                with open(runfile) as f:
                    attrs["runfile_text"] = f.read()
            except OSError:
                pass
        attrs.update({k: str(v) for k, v in kwargs.items()})

        backend = os.environ.get("PYSTELLA_OUTPUT", "h5")
        if backend == "dir":
            self.store = _DirStore(name)
            self.filename = name
        else:
            try:
                import h5py  # noqa: F401
                self.store = _H5Store(name + ".h5")
            except ImportError:
                # self-contained HDF5 writer: same file layout, no
                # libhdf5 in the image (pystella_amd/hdf5.py)
                self.store = _MiniH5Store(name + ".h5")
            self.filename = name + ".h5"
        self.store.set_attrs(attrs)

    def output(self, group, **datasets):
        """Append one row to each named dataset in ``group``."""
        for key, val in datasets.items():
            self.store.append(group, key, val)

    def read(self, group, name):
        return self.store.read(group, name)

    def close(self):
        self.store.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
