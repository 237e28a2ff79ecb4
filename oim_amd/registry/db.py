"""Registry DB backends (reference RegistryDB interface registry.go:31-41,
memdb.go:15-52).

The interface is deliberately tiny — store / lookup / list by element
prefix — so backends can be swapped: in-memory (tests, single registry),
file-backed (durable single registry), etcd (HA, see etcddb.py).
"""

from __future__ import annotations

import json
import os
import threading
from typing import Dict, List, Optional, Sequence, Tuple

from ..common import join_registry_path, split_registry_path


class RegistryDB:
    """Abstract key/value store keyed by registry paths."""

    def store(self, elements: Sequence[str], value: str) -> None:
        """Set value at path; empty value deletes the entry."""
        raise NotImplementedError

    def lookup(self, elements: Sequence[str]) -> Optional[str]:
        raise NotImplementedError

    def list(self, prefix: Sequence[str]) -> List[Tuple[List[str], str]]:
        """All (path elements, value) pairs at or beneath prefix."""
        raise NotImplementedError


def _has_prefix(elements: Sequence[str], prefix: Sequence[str]) -> bool:
    return len(elements) >= len(prefix) and list(elements[: len(prefix)]) == list(prefix)


class MemRegistryDB(RegistryDB):
    """Mutex-guarded dict (reference memdb.go:15-52)."""

    def __init__(self):
        self._mutex = threading.Lock()
        self._data: Dict[str, str] = {}

    def store(self, elements: Sequence[str], value: str) -> None:
        key = join_registry_path(list(elements))
        with self._mutex:
            if value == "":
                self._data.pop(key, None)
            else:
                self._data[key] = value

    def lookup(self, elements: Sequence[str]) -> Optional[str]:
        key = join_registry_path(list(elements))
        with self._mutex:
            return self._data.get(key)

    def list(self, prefix: Sequence[str]) -> List[Tuple[List[str], str]]:
        with self._mutex:
            snapshot = dict(self._data)
        out = []
        for key, value in sorted(snapshot.items()):
            elements = split_registry_path(key)
            if _has_prefix(elements, prefix):
                out.append((elements, value))
        return out


class FileRegistryDB(RegistryDB):
    """Durable single-file backend (JSON, atomic rename on write).

    Gives a single-registry deployment persistence across restarts —
    the middle ground between memdb and the etcd backend.
    """

    def __init__(self, path: str):
        self._path = path
        self._mutex = threading.Lock()
        self._data: Dict[str, str] = {}
        if os.path.exists(path):
            # Refuse to start on a corrupt DB rather than silently
            # discarding registry state (writes are atomic-rename, so
            # corruption means something external broke the file —
            # surface it with the path, don't guess).
            try:
                with open(path) as f:
                    data = json.load(f)
            except (json.JSONDecodeError, UnicodeDecodeError) as exc:
                raise RuntimeError(
                    f"registry db {path} is corrupt: {exc}") from exc
            if not isinstance(data, dict) or not all(
                    isinstance(k, str) and isinstance(v, str)
                    for k, v in data.items()):
                raise RuntimeError(
                    f"registry db {path} is corrupt: not a "
                    "string-to-string map")
            self._data = data

    def _flush_locked(self) -> None:
        tmp = self._path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(self._data, f)
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, self._path)

    def store(self, elements: Sequence[str], value: str) -> None:
        key = join_registry_path(list(elements))
        with self._mutex:
            if value == "":
                self._data.pop(key, None)
            else:
                self._data[key] = value
            self._flush_locked()

    def lookup(self, elements: Sequence[str]) -> Optional[str]:
        key = join_registry_path(list(elements))
        with self._mutex:
            return self._data.get(key)

    def list(self, prefix: Sequence[str]) -> List[Tuple[List[str], str]]:
        with self._mutex:
            snapshot = dict(self._data)
        out = []
        for key, value in sorted(snapshot.items()):
            elements = split_registry_path(key)
            if _has_prefix(elements, prefix):
                out.append((elements, value))
        return out
