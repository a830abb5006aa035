"""Locate lib_migbm.so (parity target: reference python-package/lightgbm/libpath.py)."""
import os
from pathlib import Path

__all__ = ["find_lib_path"]


def find_lib_path():
    """Return candidate paths of the migbm shared library."""
    here = Path(__file__).resolve().parent
    candidates = [
        here / "lib" / "lib_migbm.so",
        here.parent / "lightgbm_amd" / "lib" / "lib_migbm.so",
        here.parent / "build" / "lib_migbm.so",
    ]
    env = os.environ.get("MIGBM_LIBRARY_PATH")
    if env:
        candidates.insert(0, Path(env))
    found = [str(p) for p in candidates if p.is_file()]
    if not found:
        raise RuntimeError(
            f"Cannot find lib_migbm.so. Looked in: {[str(c) for c in candidates]}. "
            "Run `make` at the repo root (or __graft_entry__.build())."
        )
    return found
