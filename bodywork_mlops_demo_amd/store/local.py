"""Local-filesystem artefact store.

A directory tree mirroring the reference's S3 bucket layout
(``bodywork-mlops-project`` with the four prefixes — SURVEY.md §1/L1), so
the whole pipeline runs hermetically.  Writes are atomic
(tmp-file + rename) so a stage killed mid-upload never leaves a torn
artefact — stronger than the reference, whose upload is not atomic either.
"""
from __future__ import annotations

import os
import tempfile

from bodywork_mlops_demo_amd.store.base import ArtefactStore


class LocalStore(ArtefactStore):
    def __init__(self, root: str):
        self.root = os.path.abspath(root)
        os.makedirs(self.root, exist_ok=True)

    def _path(self, key: str) -> str:
        path = os.path.normpath(os.path.join(self.root, key))
        # separator-aware root check: a plain prefix test lets a sibling
        # dir sharing the root as a string prefix through
        # (root='/x/store', key '../store2/f' -> '/x/store2/f')
        if path != self.root and not path.startswith(self.root + os.sep):
            raise ValueError(f"key escapes store root: {key!r}")
        return path

    def list_keys(self, prefix: str) -> list[str]:
        base = self._path(prefix.rstrip("/"))
        if not os.path.isdir(base):
            return []
        keys = []
        for dirpath, _dirs, files in os.walk(base):
            rel = os.path.relpath(dirpath, self.root)
            for f in files:
                if f.startswith("."):
                    continue
                keys.append(os.path.join(rel, f).replace(os.sep, "/"))
        return sorted(keys)

    def get_bytes(self, key: str) -> bytes:
        with open(self._path(key), "rb") as f:
            return f.read()

    def put_bytes(self, key: str, data: bytes) -> None:
        path = self._path(key)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(path), prefix=".tmp-")
        try:
            with os.fdopen(fd, "wb") as f:
                f.write(data)
            os.replace(tmp, path)
        except BaseException:
            if os.path.exists(tmp):
                os.unlink(tmp)
            raise

    from contextlib import contextmanager as _cm

    @_cm
    def put_stream(self, key: str):
        """Direct streaming write with the same atomicity (tmp+rename)."""
        path = self._path(key)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(path), prefix=".tmp-")
        try:
            with os.fdopen(fd, "wb", buffering=1 << 20) as f:
                yield f
            os.replace(tmp, path)
        except BaseException:
            if os.path.exists(tmp):
                os.unlink(tmp)
            raise

    def exists(self, key: str) -> bool:
        return os.path.isfile(self._path(key))

    def delete(self, key: str) -> None:
        os.unlink(self._path(key))

    @property
    def uri(self) -> str:
        """URI that re-opens this store (``open_store(uri)``) — e.g. for
        handing to a serving-replica subprocess."""
        return self.root

    def __repr__(self) -> str:
        return f"LocalStore({self.root!r})"
