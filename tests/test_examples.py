"""The examples/ demos are advertised in README — keep them running.
(The apo_demo cache-overflow regression shipped unnoticed because nothing
executed the demos; these subprocess smokes close that hole.)"""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, timeout):
    return subprocess.run([sys.executable, os.path.join(ROOT, "examples", script)],
                          capture_output=True, text=True, timeout=timeout)


@pytest.mark.parametrize("script,needle,timeout", [
    ("apo_demo.py", "state persisted", 420),
    ("agent_demo.py", "demo ok", 300),
    ("serving_demo.py", "demo ok", 420),
])
def test_demo_runs(script, needle, timeout):
    r = _run(script, timeout)
    assert r.returncode == 0, r.stderr[-800:]
    assert needle in r.stdout, r.stdout[-800:]
