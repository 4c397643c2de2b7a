"""Checkpoint storage backends (reference trainer/checkpoint_storage.py:
46,138,287 — local FS + S3-with-retries).

The engine talks to a small interface so remote backends plug in without
touching the save/load logic.  Shipped: LocalStorage.  ``s3://`` URIs
raise with a pointer to the extension point (no cluster object store in
this environment; the reference's S3 path needs boto3+CRT)."""

import os
import shutil
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


def get_storage(path: str) -> BaseCheckpointStorage:
    if path.startswith("s3://"):
        raise NotImplementedError(
            "S3 checkpoint storage: subclass BaseCheckpointStorage with a "
            "boto3/CRT client and pass it to the checkpoint engine "
            "(reference checkpoint_storage.py:138-287); this environment "
            "has no object store")
    return LocalStorage(path)
