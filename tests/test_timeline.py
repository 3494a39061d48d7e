# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Timeline tracing test (reference analog: test/timeline_test.py): run ops
with BLUEFOG_TIMELINE set, then parse the emitted Chrome-trace JSON."""

import json
import os
import tempfile

import torch

from tests.util import run_dist

_TL_DIR = tempfile.gettempdir()
_TL_BASE = os.path.join(_TL_DIR, "bf_test_timeline")


def w_timeline():
    import bluefog_amd as bf

    bf.init()
    rank = bf.rank()
    for i in range(3):
        bf.neighbor_allreduce(torch.ones(8) * rank, name=f"x{i}")
    with bf.timeline_context("user_tensor", "COMPUTE"):
        _ = torch.ones(4) * 2
    h = bf.allreduce_nonblocking(torch.ones(2), name="ar")
    bf.synchronize(h)
    from bluefog_amd.utils.timeline import timeline

    timeline().shutdown()
    fname = f"{_TL_BASE}_{rank}.json"
    assert os.path.exists(fname), fname
    with open(fname) as f:
        content = f.read()
    data = json.loads(content)
    names = {r.get("name") for r in data if isinstance(r, dict)}
    assert any(n and "COMMUNICATE" in str(n) for n in names), names
    assert any(n and "COMPUTE" in str(n) for n in names), names
    pids = {r["args"]["name"] for r in data if r.get("ph") == "M"}
    assert any("neighbor.allreduce.x0" in p for p in pids), pids
    assert "user_tensor" in pids


def test_timeline():
    run_dist(w_timeline, 2, env={"BLUEFOG_TIMELINE": _TL_BASE})
