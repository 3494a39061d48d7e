# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Prometheus metrics (optional, beyond reference parity)."""

import urllib.request

import pytest

pytest.importorskip("prometheus_client")


def w_metrics_endpoint():
    import os

    import torch

    import bluefog_amd as bf

    bf.init()
    port = 19300 + int(os.environ["BF_TEST_PORT_OFF"]) + bf.rank()
    bf.start_metrics_server(port)
    for i in range(3):
        t = torch.ones(256) * bf.rank()
        bf.allreduce(t, name=f"m{i}")
    bf.neighbor_allreduce(torch.ones(64))
    import urllib.request as u

    body = u.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
    assert 'bluefog_ops_total{op="allreduce"}' in body, body[:500]
    assert "bluefog_op_seconds" in body
    bf.barrier()


def test_metrics_endpoint():
    import random

    from tests.util import run_dist

    off = random.randint(0, 400)
    run_dist(w_metrics_endpoint, 2, env={"BF_TEST_PORT_OFF": str(off)})
