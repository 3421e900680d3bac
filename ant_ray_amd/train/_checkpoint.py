"""Train Checkpoint: a directory of files + a `.metadata.json` sidecar.

Role parity: reference python/ray/train/_checkpoint.py:56 (class Checkpoint,
_METADATA_FILE_NAME ".metadata.json" :23, from_directory :179, to_directory
:190, as_directory :234). Same on-disk format: an opaque directory the user
populates (torch.save files etc.) plus optional metadata in `.metadata.json`,
so checkpoints written by the reference restore here and vice versa.

Local-filesystem URIs only (no network in this environment); the `path`
attribute is a plain directory path, `filesystem` kept for API parity.
"""
from __future__ import annotations

import contextlib
import json
import os
import shutil
import tempfile
from typing import Any, Dict, Iterator, Optional

_METADATA_FILE_NAME = ".metadata.json"


class Checkpoint:
    """A reference to a checkpoint directory persisted somewhere durable."""

    def __init__(self, path: str, filesystem: Optional[Any] = None):
        if "://" in str(path):
            path = str(path).split("://", 1)[1]
        self.path = os.fspath(path)
        self.filesystem = filesystem

    def __repr__(self):
        return f"Checkpoint(path={self.path})"

    def __eq__(self, other):
        return isinstance(other, Checkpoint) and self.path == other.path

    def __hash__(self):
        return hash(self.path)

    def __fspath__(self):
        return self.path

    # ------------------------------------------------------------ metadata

    def _metadata_path(self) -> str:
        return os.path.join(self.path, _METADATA_FILE_NAME)

    def get_metadata(self) -> Dict[str, Any]:
        p = self._metadata_path()
        if not os.path.exists(p):
            return {}
        with open(p) as f:
            return json.load(f)

    def set_metadata(self, metadata: Dict[str, Any]) -> None:
        with open(self._metadata_path(), "w") as f:
            json.dump(metadata, f)

    def update_metadata(self, metadata: Dict[str, Any]) -> None:
        md = self.get_metadata()
        md.update(metadata)
        self.set_metadata(md)

    # ------------------------------------------------------------ contents

    @classmethod
    def from_directory(cls, path) -> "Checkpoint":
        return cls(os.fspath(path))

    def to_directory(self, path: Optional[str] = None) -> str:
        """Copy checkpoint contents into `path` (or a fresh temp dir)."""
        target = os.fspath(path) if path else tempfile.mkdtemp(prefix="antray-ckpt-")
        os.makedirs(target, exist_ok=True)
        shutil.copytree(self.path, target, dirs_exist_ok=True)
        return target

    @contextlib.contextmanager
    def as_directory(self) -> Iterator[str]:
        """Local checkpoints are yielded in place (no copy, not deleted)."""
        yield self.path
