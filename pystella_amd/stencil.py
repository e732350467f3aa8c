"""Stencil kernels: elementwise maps whose right-hand sides read
shifted (neighbor) values of halo-padded fields.

Analogue of reference pystella/stencil.py:36-143.  In the reference,
``Stencil`` and ``StreamingStencil`` are separate loopy code paths (a
workgroup-tile prefetch kernel vs an x-streaming variant).  Here both
collapse onto one implementation: the CDNA4 elementwise template
already walks x with a (64z × 4y) tile per block (the "streaming"
shape), and shifted reads of padded fields are part of the expression
language — neighbor loads are served by L1/L2 with coalesced z access.
``lsize``/``prefetch_args`` keyword arguments are accepted for API
compatibility and ignored.
"""

from __future__ import annotations

from pystella_amd.elementwise import ElementWiseMap

__all__ = ["Stencil", "StreamingStencil"]


class Stencil(ElementWiseMap):
    def __init__(self, map_dict, tmp_instructions=None, halo_shape=0,
                 rank_shape=None, lsize=None, prefetch_args=None, **kwargs):
        super().__init__(map_dict, tmp_instructions=tmp_instructions,
                         halo_shape=halo_shape, rank_shape=rank_shape,
                         **kwargs)


class StreamingStencil(Stencil):
    pass
