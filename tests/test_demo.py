"""The examples/demo.py tour must keep working (doubles as an e2e check of
every scheduling mode through the native server)."""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_demo_runs():
    out = subprocess.run([sys.executable, str(REPO / "examples" / "demo.py")],
                         capture_output=True, text=True, timeout=300,
                         cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "xGMI-adjacent" in out.stdout
    assert "spread" in out.stdout
