# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Guard the driver's bench.py contract: one JSON line from rank 0 with the
required keys and the BASELINE metric/config names."""

import json
import os
import subprocess
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, os.path.join(_ROOT, "bench.py"),
         "--model", "resnet18", "--batch-size", "2", "--steps", "2",
         "--warmup", "1", "--device", "cpu"],
        cwd=_ROOT, env=env, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == 1
    assert rec["steps"] == 2
    assert rec["data"] == "synthetic"
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0
    assert set(rec["config"]) >= {"model", "global_batch", "seq_len", "parallelism"}


def test_bench_default_metric_is_baseline_config():
    """Default invocation must measure the BASELINE.json headline config."""
    import argparse

    sys.path.insert(0, _ROOT)
    import bench

    args = bench.parse_args.__wrapped__() if hasattr(bench.parse_args, "__wrapped__") else None
    # parse with no CLI args
    old = sys.argv
    sys.argv = ["bench.py"]
    try:
        args = bench.parse_args()
    finally:
        sys.argv = old
    assert args.model == "resnet50"
    assert args.batch_size == 64
    assert args.dist_optimizer == "neighbor_allreduce"
    assert args.dtype == "fp32"


def test_bench_self_spawn_8rank():
    """The driver's SCALE invocation is `python bench.py --gpus N` with NO
    rendezvous env: bench.py must self-spawn N ranks and report n_gpus=N
    (a 1-rank number labeled n_gpus=8 would invalidate the record)."""
    env = dict(os.environ)
    for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    out = subprocess.run(
        [sys.executable, os.path.join(_ROOT, "bench.py"), "--gpus", "8",
         "--model", "resnet18", "--batch-size", "1", "--steps", "2",
         "--warmup", "1", "--device", "cpu"],
        cwd=_ROOT, env=env, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 8, rec
    assert rec["config"]["global_batch"] == 8, rec


def test_bench_world_size_mismatch_fails():
    """--gpus disagreeing with the launched world must fail loudly, not
    report a mislabeled measurement."""
    env = dict(os.environ)
    env.update(RANK="0", LOCAL_RANK="0", WORLD_SIZE="1",
               MASTER_ADDR="127.0.0.1", MASTER_PORT="29399")
    out = subprocess.run(
        [sys.executable, os.path.join(_ROOT, "bench.py"), "--gpus", "4",
         "--model", "resnet18", "--batch-size", "1", "--steps", "1",
         "--warmup", "0", "--device", "cpu"],
        cwd=_ROOT, env=env, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode != 0
    assert "world size" in (out.stderr + out.stdout)


def test_bench_config_presets():
    """--config 2..5 presets the BASELINE.json model/batch/optimizer."""
    sys.path.insert(0, _ROOT)
    import bench

    old = sys.argv
    try:
        sys.argv = ["bench.py", "--config", "3"]
        a = bench.parse_args()
        assert (a.model, a.batch_size, a.dist_optimizer) == (
            "resnet50", 32, "neighbor_allreduce")
        sys.argv = ["bench.py", "--config", "5"]
        a = bench.parse_args()
        assert a.model == "bert_base" and a.seq_len == 512
        assert a.dist_optimizer == "hierarchical_neighbor_allreduce"
        sys.argv = ["bench.py", "--config", "4", "--batch-size", "8"]
        a = bench.parse_args()  # explicit flags win over the preset
        assert a.batch_size == 8 and a.dist_optimizer == "win_put"
    finally:
        sys.argv = old


def test_bench_torchrun_default_optimizer():
    """The driver's SCALE tier launches bench.py under torch.distributed.run
    with the DEFAULT optimizer; rehearse that exact invocation at ws=2."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(_ROOT, "bench.py"), "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--model", "resnet18", "--batch-size", "2",
         "--device", "cpu"],
        cwd=_ROOT, env=dict(os.environ), capture_output=True, text=True,
        timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["config"]["global_batch"] == 4, rec
    assert "dynamic one-peer exp2" in rec["config"]["parallelism"], rec


def test_bench_hierarchical_faked_machines():
    """BASELINE config 5's code path end to end: 4 ranks, 2 faked machines
    (BLUEFOG_NODES_PER_MACHINE), hierarchical_neighbor_allreduce."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ, BLUEFOG_NODES_PER_MACHINE="2")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(_ROOT, "bench.py"), "--gpus", "4", "--steps", "2",
         "--warmup", "1", "--model", "resnet18", "--batch-size", "2",
         "--device", "cpu", "--dist-optimizer", "hierarchical_neighbor_allreduce"],
        cwd=_ROOT, env=env, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:] + out.stdout[-500:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert "hierarchical" in rec["config"]["parallelism"], rec["config"]
