"""fsspec-backed model-blob store — the HDFS/S3/remote-FS equivalent.

Parity with the reference's remote model-data backends:
- storage/hdfs/.../HDFSModels.scala:36-63 (model blob as HDFS file)
- storage/s3/.../S3Models.scala:41-101 (model blob as S3 object)

One implementation covers both here: any fsspec URL (file://, s3://,
hdfs://, gs://, ...) works when the matching fsspec driver is installed;
plain local paths work everywhere. Configure with
  PIO_STORAGE_SOURCES_<NAME>_TYPE=fsspec
  PIO_STORAGE_SOURCES_<NAME>_PATH=<url-or-path>
"""

from __future__ import annotations

from typing import Optional

from predictionio_amd.data.storage import base
from predictionio_amd.data.storage.base import Model


class FsspecClient:
    def __init__(self, path: str):
        import fsspec
        self.fs, self.root = fsspec.core.url_to_fs(path)
        self.fs.makedirs(self.root, exist_ok=True)


class FsspecModels(base.Models):
    def __init__(self, client: FsspecClient):
        self.fs = client.fs
        self.root = client.root

    def _path(self, mid: str) -> str:
        return f"{self.root}/pio_model_{mid}"

    def insert(self, m: Model) -> None:
        with self.fs.open(self._path(m.id), "wb") as f:
            f.write(m.models)

    def get(self, mid: str) -> Optional[Model]:
        p = self._path(mid)
        if not self.fs.exists(p):
            return None
        with self.fs.open(p, "rb") as f:
            return Model(mid, f.read())

    def delete(self, mid: str) -> bool:
        p = self._path(mid)
        if not self.fs.exists(p):
            return False
        self.fs.rm(p)
        return True
