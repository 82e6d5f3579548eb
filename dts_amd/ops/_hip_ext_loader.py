"""Loader for the in-tree HIP extension.

The extension is built IN-TREE (dts_amd/ops/_dts_hip.<abi>.so) by
`python -m dts_amd.ops.build` (or __graft_entry__.build()), so the .so
travels to the GPU box with the repo snapshot — a JIT cache under
~/.cache would not.
"""

from __future__ import annotations


def load():
    import torch  # noqa: F401 — the extension links against libtorch

    from dts_amd.ops import _dts_hip  # type: ignore[attr-defined]

    return _dts_hip
