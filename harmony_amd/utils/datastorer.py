"""Data storer SPI (reference: common/datastorer — DataStorer + the
LocalFSDataStorer default): persist named byte/tensor blobs."""

from __future__ import annotations

import os
from pathlib import Path

import torch


class DataStorer:
    def store(self, name: str, data) -> None:
        raise NotImplementedError

    def load(self, name: str):
        raise NotImplementedError

    def exists(self, name: str) -> bool:
        raise NotImplementedError


class LocalFSDataStorer(DataStorer):
    def __init__(self, root: str = "/tmp/harmony_data"):
        self.root = Path(root)
        self.root.mkdir(parents=True, exist_ok=True)

    def _p(self, name: str) -> Path:
        p = (self.root / name).resolve()
        assert str(p).startswith(str(self.root.resolve())), "path escape"
        return p

    def store(self, name: str, data) -> None:
        p = self._p(name)
        p.parent.mkdir(parents=True, exist_ok=True)
        if isinstance(data, bytes):
            p.write_bytes(data)
        else:
            torch.save(data, p)

    def load(self, name: str):
        p = self._p(name)
        try:
            return torch.load(p, weights_only=True)
        except Exception:  # noqa: BLE001
            return p.read_bytes()

    def exists(self, name: str) -> bool:
        return self._p(name).exists()
