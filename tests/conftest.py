import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture()
def mem_storage(monkeypatch, tmp_path):
    """Fresh isolated sqlite-file storage for each test."""
    from predictionio_amd.data import storage
    storage.reset()
    monkeypatch.setenv("PIO_FS_BASEDIR", str(tmp_path))
    monkeypatch.setenv("PIO_STORAGE_SOURCES_TEST_TYPE", "sqlite")
    monkeypatch.setenv("PIO_STORAGE_SOURCES_TEST_PATH",
                       str(tmp_path / "pio.sqlite"))
    for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
        monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE", "TEST")
        monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME", "test")
    yield storage
    storage.reset()
