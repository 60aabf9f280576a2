"""The walkthrough examples must keep running (they are the reference's
notebook-level surface; rot here is API breakage users would hit)."""
import subprocess
import sys

import pytest

ROOT = __file__.rsplit("/", 2)[0]


@pytest.mark.parametrize("script,needle", [
    ("examples/02_hyperopt.py", "distributed best C"),
    ("examples/05_serving.py", "classify top-3"),
])
def test_example_runs(script, needle):
    out = subprocess.run([sys.executable, f"{ROOT}/{script}"],
                         capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert needle in out.stdout, out.stdout[-2000:]
