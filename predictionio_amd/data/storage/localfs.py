"""LocalFS model-data backend: one blob file per engine instance.

Parity: storage/localfs/.../LocalFSModels.scala:33-62 — model blob stored as
`pio_model_<id>` under a base directory.
"""

from __future__ import annotations

import os
from typing import Optional

from predictionio_amd.data.storage import base
from predictionio_amd.data.storage.base import Model


class LocalFSClient:
    def __init__(self, path: str):
        self.path = path
        os.makedirs(path, exist_ok=True)


class LocalFSModels(base.Models):
    def __init__(self, client: LocalFSClient):
        self.base_dir = client.path

    def _path(self, mid: str) -> str:
        return os.path.join(self.base_dir, f"pio_model_{mid}")

    def insert(self, m: Model) -> None:
        with open(self._path(m.id), "wb") as f:
            f.write(m.models)

    def get(self, mid: str) -> Optional[Model]:
        p = self._path(mid)
        if not os.path.exists(p):
            return None
        with open(p, "rb") as f:
            return Model(mid, f.read())

    def delete(self, mid: str) -> bool:
        p = self._path(mid)
        if os.path.exists(p):
            os.remove(p)
            return True
        return False
