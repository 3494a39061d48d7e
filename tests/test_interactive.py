# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""ibfrun interactive-cluster tests (reference analog: the ipyparallel
ibfrun of bluefog/run/interactive_run.py)."""

import os

import pytest

from bluefog_amd.run import interactive as ib


@pytest.fixture
def cluster(tmp_path, monkeypatch):
    monkeypatch.setattr(ib, "_STATE_DIR", str(tmp_path))
    state = ib.start_cluster(2, profile="test", extra_env={"BLUEFOG_LOG_LEVEL": "error"})
    yield state
    ib.stop_cluster(profile="test")


def _init_and_allreduce():
    import torch

    import bluefog_amd as bf

    bf.init()
    t = torch.ones(4) * (bf.rank() + 1)
    out = bf.allreduce(t, average=True, name="ib_t")
    return float(out[0])


def test_interactive_roundtrip(cluster):
    c = ib.InteractiveClient(profile="test")
    assert c.ping() == ["0", "1"]
    # persistent namespace
    c.run_code("x = int(__import__('os').environ['RANK']) * 10")
    assert c.pull("x") == [0, 10]
    # real collective over gloo inside the workers
    outs = c.run(_init_and_allreduce)
    assert outs == [1.5, 1.5]
    c.close()


def _boom():
    raise RuntimeError("worker exploded")


def test_interactive_error_propagates(cluster):
    c = ib.InteractiveClient(profile="test")

    with pytest.raises(ib.ClusterError, match="worker exploded"):
        c.run(_boom)
    # cluster still usable afterwards
    assert c.ping() == ["0", "1"]
    c.close()


def test_ibfrun_cli_start_stop(tmp_path, monkeypatch):
    """The ibfrun CLI itself: start 2 workers, connect, stop."""
    import subprocess

    env = dict(os.environ, BLUEFOG_IBFRUN_DIR=str(tmp_path))
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [os.path.join(root, "ibfrun"), "start", "-np", "2",
         "--ipython-profile", "cli"],
        capture_output=True, text=True, timeout=120, env=env, cwd=root,
    )
    assert r.returncode == 0, r.stderr
    assert "started 2 workers" in r.stdout
    try:
        monkeypatch.setattr(ib, "_STATE_DIR", str(tmp_path))
        c = ib.InteractiveClient(profile="cli")
        assert c.ping() == ["0", "1"]
        c.close()
    finally:
        r = subprocess.run(
            [os.path.join(root, "ibfrun"), "stop", "--ipython-profile", "cli"],
            capture_output=True, text=True, timeout=60, env=env, cwd=root,
        )
        assert r.returncode == 0, r.stderr
