"""Registry key paths (reference pkg/oim-common/path.go:15-38).

Registry keys are slash-separated paths like ``controller-id/address``.
Elements must be non-empty and must not be ``.`` or ``..`` so a DB backed
by a filesystem or etcd prefix scan cannot be escaped.
"""

from __future__ import annotations

from typing import List


class RegistryPathError(ValueError):
    pass


def split_registry_path(path: str) -> List[str]:
    # Lenient like the reference: empty elements (leading/trailing/
    # doubled slashes) are dropped, so "/a//b/" normalizes to a/b.
    # Callers get the CANONICAL path back from GetValues — paths are
    # identifiers up to normalization, not byte strings.
    elements = [e for e in path.split("/") if e != ""]
    for element in elements:
        if element in (".", ".."):
            raise RegistryPathError(f"invalid path element {element!r} in {path!r}")
    if not elements:
        raise RegistryPathError(f"empty registry path: {path!r}")
    return elements


def join_registry_path(elements: List[str]) -> str:
    for element in elements:
        if not element or "/" in element or element in (".", ".."):
            raise RegistryPathError(f"invalid path element: {element!r}")
    return "/".join(elements)
