# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""bfrun launcher tests: env rendezvous wiring and failure propagation."""

import os
import subprocess
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_BFRUN = os.path.join(_ROOT, "bfrun")


def _run(np_, code, timeout=120):
    return subprocess.run(
        [sys.executable, _BFRUN, "-np", str(np_), sys.executable, "-c", code],
        capture_output=True, text=True, timeout=timeout, cwd=_ROOT,
    )


def test_bfrun_env_wiring():
    out = _run(3, "import os; print('R', os.environ['RANK'], os.environ['WORLD_SIZE'])")
    assert out.returncode == 0, out.stderr
    got = sorted(l for l in out.stdout.splitlines() if l.startswith("R "))
    assert got == ["R 0 3", "R 1 3", "R 2 3"]


def test_bfrun_failure_propagates():
    out = _run(2, "import os, sys; sys.exit(7 if os.environ['RANK'] == '1' else 0)")
    assert out.returncode == 7
    assert "rank 1 exited with code 7" in out.stderr


def test_bfrun_kills_stragglers_on_failure():
    code = (
        "import os, sys, time\n"
        "if os.environ['RANK'] == '0':\n"
        "    sys.exit(3)\n"
        "time.sleep(600)\n"
    )
    out = _run(2, code, timeout=90)
    assert out.returncode == 3
