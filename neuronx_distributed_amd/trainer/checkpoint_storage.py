"""Checkpoint storage backends (reference trainer/checkpoint_storage.py:
46,138,287 — local FS + S3-with-retries).

The engine talks to a small interface so remote backends plug in without
touching the save/load logic.  Shipped:

* ``LocalStorage`` — plain filesystem (default).
* ``FsspecStorage`` — any fsspec URL (``s3://``, ``gs://``, ``memory://``,
  ...), with jittered-decrementing retry on transient errors like the
  reference's tenacity-wrapped S3 client (checkpoint_storage.py:236-285).
  ``s3://`` resolves through fsspec's s3fs when installed on the cluster;
  ``memory://`` backs the CI tests.
"""

import io
import os
import random
import shutil
import time
from typing import Any

import torch


class BaseCheckpointStorage:
    def __init__(self, root: str):
        self.root = root

    def save_object(self, obj: Any, rel_path: str) -> None:
        raise NotImplementedError

    def load_object(self, rel_path: str, map_location="cpu") -> Any:
        raise NotImplementedError

    def exists(self, rel_path: str) -> bool:
        raise NotImplementedError

    def write_text(self, rel_path: str, text: str) -> None:
        raise NotImplementedError

    def listdir(self, rel_path: str = ""):
        raise NotImplementedError

    def remove_tree(self, rel_path: str) -> None:
        raise NotImplementedError


class LocalStorage(BaseCheckpointStorage):
    def _full(self, rel_path: str) -> str:
        return os.path.join(self.root, rel_path) if rel_path else self.root

    def save_object(self, obj: Any, rel_path: str) -> None:
        full = self._full(rel_path)
        os.makedirs(os.path.dirname(full), exist_ok=True)
        torch.save(obj, full)

    def load_object(self, rel_path: str, map_location="cpu") -> Any:
        return torch.load(self._full(rel_path), map_location=map_location,
                          weights_only=False)

    def exists(self, rel_path: str) -> bool:
        return os.path.exists(self._full(rel_path))

    def write_text(self, rel_path: str, text: str) -> None:
        full = self._full(rel_path)
        os.makedirs(os.path.dirname(full), exist_ok=True)
        with open(full, "w") as f:
            f.write(text)

    def listdir(self, rel_path: str = ""):
        full = self._full(rel_path)
        return os.listdir(full) if os.path.isdir(full) else []

    def remove_tree(self, rel_path: str) -> None:
        shutil.rmtree(self._full(rel_path), ignore_errors=True)


def retry_transient(fn, attempts: int = 5, first_wait: float = 2.0):
    """Call ``fn`` retrying transient errors with DECREMENTING jittered
    waits (reference checkpoint_storage.py:236-285 retries S3 slow-downs
    with tenacity wait_random_exponential-then-decrement): the first wait
    is the longest so a throttled fleet spreads out, later waits shrink."""
    last = None
    for i in range(attempts):
        try:
            return fn()
        except (OSError, IOError, TimeoutError) as e:  # transient classes
            last = e
            if i == attempts - 1:
                break
            wait = first_wait * (attempts - 1 - i) / (attempts - 1)
            time.sleep(random.uniform(0, max(wait, 0.01)))
    raise last


class FsspecStorage(BaseCheckpointStorage):
    """Checkpoint storage over any fsspec filesystem (s3://, gs://,
    memory://, ...).  Objects are serialized with torch.save into an
    in-memory buffer and written in one put (object stores have no
    append); every call retries transient errors."""

    def __init__(self, root: str):
        super().__init__(root)
        import fsspec

        self.fs, self._root = fsspec.core.url_to_fs(root)

    def _full(self, rel_path: str) -> str:
        return f"{self._root}/{rel_path}" if rel_path else self._root

    def save_object(self, obj: Any, rel_path: str) -> None:
        buf = io.BytesIO()
        torch.save(obj, buf)
        data = buf.getvalue()

        def put():
            with self.fs.open(self._full(rel_path), "wb") as f:
                f.write(data)

        retry_transient(put)

    def load_object(self, rel_path: str, map_location="cpu") -> Any:
        def get():
            with self.fs.open(self._full(rel_path), "rb") as f:
                return f.read()

        data = retry_transient(get)
        return torch.load(io.BytesIO(data), map_location=map_location,
                          weights_only=False)

    def exists(self, rel_path: str) -> bool:
        return retry_transient(lambda: self.fs.exists(self._full(rel_path)))

    def write_text(self, rel_path: str, text: str) -> None:
        def put():
            with self.fs.open(self._full(rel_path), "w") as f:
                f.write(text)

        retry_transient(put)

    def listdir(self, rel_path: str = ""):
        full = self._full(rel_path)
        try:
            entries = retry_transient(lambda: self.fs.ls(full, detail=False))
        except FileNotFoundError:
            return []
        return [e.rstrip("/").rsplit("/", 1)[-1] for e in entries]

    def remove_tree(self, rel_path: str) -> None:
        try:
            retry_transient(
                lambda: self.fs.rm(self._full(rel_path), recursive=True))
        except FileNotFoundError:
            pass


def get_storage(path: str) -> BaseCheckpointStorage:
    """Local paths -> LocalStorage; URL-style paths (s3://, memory://, ...)
    -> FsspecStorage (reference create_checkpoint_storage,
    checkpoint_storage.py:287)."""
    if "://" in path:
        return FsspecStorage(path)
    return LocalStorage(path)


# reference API name (trainer/checkpoint_storage.py:287)
create_checkpoint_storage = get_storage
