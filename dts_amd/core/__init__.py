"""Native (C++) runtime core: block manager + scheduler (_dts_core)."""

from __future__ import annotations

_core = None
_err: Exception | None = None


def load_core():
    """Import the built _dts_core extension (None if not built)."""
    global _core, _err
    if _core is not None or _err is not None:
        return _core
    try:
        import torch  # noqa: F401

        from dts_amd.core import _dts_core  # type: ignore[attr-defined]

        _core = _dts_core
    except Exception as e:  # noqa: BLE001
        _err = e
    return _core
