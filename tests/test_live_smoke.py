"""CI-style live smoke (reference .github/workflows/ci.yml:146-209): real
gateway + real hello backend processes, curl transcript checks."""

import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.timeout(180)
@pytest.mark.parametrize("frontend", ["asyncio", "native"])
def test_live_smoke(frontend):
    p = subprocess.run(
        ["bash", str(REPO / "tools" / "live_smoke.sh"), "--frontend", frontend],
        capture_output=True, text=True, timeout=150, cwd=str(REPO),
    )
    assert p.returncode == 0, p.stdout + "\n" + p.stderr
    assert "PASS" in p.stdout
