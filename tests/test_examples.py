"""Example scripts run as tests (the reference executes its example
notebooks in CI — tests/test_examples.py there)."""
import os
import subprocess
import sys

import pytest

EXAMPLES = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples"
)


@pytest.mark.timeout(300)
@pytest.mark.parametrize(
    "script", ["minimal_build_and_serve.py", "fleet_and_client.py",
               "training_controls.py"]
)
def test_example_runs(script):
    proc = subprocess.run(
        [sys.executable, os.path.join(EXAMPLES, script)],
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
