"""Technique library: a directory of dill-serialized UDP classes.

Parity with reference ``saturn/library/library.py:19-73``: techniques are
stored as ``$SATURN_LIBRARY_PATH/<name>.udp`` dill blobs so that user-defined
parallelisms survive process boundaries; the format and the env-var contract
are kept.  Fixes the reference ``deregister`` list-path bug (library.py:45-47
drops the ``.udp`` suffix).
"""

from __future__ import annotations

import os
from typing import List, Optional, Union

import dill

from saturn_amd.core.technique import BaseTechnique


def _lib_dir() -> str:
    try:
        d = os.environ["SATURN_LIBRARY_PATH"]
    except KeyError as e:
        raise RuntimeError(
            "Set SATURN_LIBRARY_PATH to a writable directory before using "
            "the technique library."
        ) from e
    os.makedirs(d, exist_ok=True)
    return d


def register(name: str, udp: type) -> None:
    """Persist a BaseTechnique subclass as ``<name>.udp``."""
    if not (isinstance(udp, type) and issubclass(udp, BaseTechnique)):
        raise RuntimeError(
            f"Parallelism {name!r} must be a subclass of "
            "saturn_amd.BaseTechnique."
        )
    with open(os.path.join(_lib_dir(), f"{name}.udp"), "wb") as s:
        dill.dump(udp, s)


def deregister(name: Union[str, List[str]]) -> None:
    names = name if isinstance(name, list) else [name]
    for n in names:
        os.remove(os.path.join(_lib_dir(), f"{n}.udp"))


def retrieve(name: Optional[Union[str, List[str]]] = None):
    """Load one, several, or (name=None) all registered techniques."""
    if name is None:
        name = sorted(
            os.path.splitext(f)[0]
            for f in os.listdir(_lib_dir())
            if f.endswith(".udp")
        )
    if isinstance(name, list):
        return [retrieve(n) for n in name]
    with open(os.path.join(_lib_dir(), f"{name}.udp"), "rb") as s:
        return dill.load(s)
