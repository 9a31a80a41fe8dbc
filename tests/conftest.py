import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires a real MI355X (gfx950) GPU")


@pytest.fixture
def tmp_db(tmp_path):
    return str(tmp_path / "meta.db")
