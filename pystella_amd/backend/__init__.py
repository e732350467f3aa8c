"""Execution backends for pystella_amd.

* ``torcheval`` — evaluates symbolic statements with torch tensor ops.
  This is the CPU reference path (tests/oracles) and also runs on GPU
  tensors for debugging, but the production GPU path is the HIP one.
* ``codegen`` — emits HIP C++ from symbolic statements, spliced into
  hand-written CDNA4 kernel templates.
* ``hip`` — loads the in-tree ``_C`` extension (AOT kernels + hiprtc JIT
  runtime) and provides launch helpers.
"""
