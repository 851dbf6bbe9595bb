"""Loader for the in-tree HIP extension (built by ``setup.py build_ext
--inplace`` — the .so lives inside the package so it travels to GPU boxes
with the repo snapshot)."""

from __future__ import annotations


def load():
    from multihop_offload_amd import _hip_ops  # built in-tree
    return _hip_ops
