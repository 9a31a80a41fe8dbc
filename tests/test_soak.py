"""Short soak run in the CPU suite (tools/soak.py is the long-form driver)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(180)
def test_soak_short():
    out = subprocess.run(
        [sys.executable, "tools/soak.py", "--seconds", "10", "--workers", "3"],
        cwd=REPO, capture_output=True, text=True, timeout=150,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-1000:]
    assert "soak OK" in out.stdout
