"""Loader for the in-tree HIP extension (poseidon_amd.ops._hip).

Fails loudly with build instructions when the extension is missing -- there
is deliberately NO eager-torch fallback on GPU tensors."""

from __future__ import annotations

_MOD = None


def load():
    global _MOD
    if _MOD is None:
        from . import _hip  # built by setup.py build_ext --inplace
        _MOD = _hip
    return _MOD
