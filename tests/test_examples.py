"""The examples/ scripts must stay runnable (they are the reference
notebooks' replacement and double as end-to-end smoke tests)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script", ["prompt_tuning_classification.py",
                                    "multi_turn_chat.py"])
def test_example_runs(script):
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", script)],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    assert out.stdout.strip(), "example should print progress"
