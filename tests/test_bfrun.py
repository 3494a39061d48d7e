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


def test_parse_hosts_and_blocks():
    from bluefog_amd.run.bfrun import parse_hosts, rank_blocks

    hosts = parse_hosts("nodeA:8, nodeB:8", 16)
    assert hosts == [("nodeA", 8), ("nodeB", 8)]
    assert rank_blocks(hosts, 16) == [("nodeA", 0, 8), ("nodeB", 8, 8)]
    # -np smaller than total slots: trailing hosts get fewer/no ranks
    assert rank_blocks(hosts, 10) == [("nodeA", 0, 8), ("nodeB", 8, 2)]
    assert rank_blocks(hosts, 8) == [("nodeA", 0, 8)]
    import pytest

    with pytest.raises(ValueError):
        parse_hosts("nodeA:4", 8)  # not enough slots
    with pytest.raises(ValueError):
        parse_hosts("nodeA:x", 1)


def test_remote_command_composition():
    from bluefog_amd.run.bfrun import remote_command

    cmd = remote_command("nodeB", 8, 8, 16, "nodeA", 29501,
                         ["python", "train.py", "--lr", "0.1"], ["FOO=1"])
    assert cmd[0] == "ssh" and "nodeB" in cmd
    joined = cmd[-1]
    assert "--local-first-rank 8" in joined
    assert "--local-num 8" in joined
    assert "--master-addr nodeA" in joined
    assert "--extra-env FOO=1" in joined
    assert joined.endswith("-- python train.py --lr 0.1")


def test_multihost_rank_blocks_rendezvous():
    """Simulate the two halves of a 2-host launch as two local bfrun
    invocations with --local-first-rank/--local-num (exactly what the ssh
    side runs): all ranks must rendezvous into one world."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    code = (
        "import os, torch.distributed as dist\n"
        "dist.init_process_group('gloo')\n"
        "print('RZ', dist.get_rank(), dist.get_world_size(), flush=True)\n"
        "dist.barrier()\n"
        "dist.destroy_process_group()\n"
    )
    halves = []
    for first, cnt in [(0, 1), (1, 1)]:
        halves.append(subprocess.Popen(
            [sys.executable, "-m", "bluefog_amd.run.bfrun", "-np", "2",
             "--master-addr", "127.0.0.1", "--master-port", str(port),
             "--local-first-rank", str(first), "--local-num", str(cnt),
             "--", sys.executable, "-c", code],
            cwd=_ROOT, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True))
    outs = [p.communicate(timeout=120) for p in halves]
    assert all(p.returncode == 0 for p in halves), outs
    lines = sorted(l for o, _ in outs for l in o.splitlines() if l.startswith("RZ"))
    assert lines == ["RZ 0 2", "RZ 1 2"], outs
