# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Examples as end-to-end tests (reference analog: test/test_all_example.sh
run through scripts/wrap_examples.sh). Each example runs for a few
iterations on 2 CPU ranks via bfrun."""

import os
import subprocess
import sys

import pytest

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_BFRUN = os.path.join(_ROOT, "bfrun")


def _run_example(np_, script, *args, timeout=420):
    out = subprocess.run(
        [sys.executable, _BFRUN, "-np", str(np_), sys.executable,
         os.path.join(_ROOT, "examples", script), *args],
        capture_output=True, text=True, timeout=timeout, cwd=_ROOT,
    )
    assert out.returncode == 0, f"{script} failed:\n{out.stdout[-1500:]}\n{out.stderr[-1500:]}"
    return out.stdout


def test_example_average_consensus():
    out = _run_example(2, "pytorch_average_consensus.py")
    assert "consensus reached" in out or "relative error" in out


def test_example_optimization_exact_diffusion():
    out = _run_example(2, "pytorch_optimization.py", "--method",
                       "exact_diffusion", "--iters", "40")
    assert "consensus error" in out


def test_example_mnist_tiny():
    out = _run_example(
        2, "pytorch_mnist.py", "--epochs", "1", "--samples-per-rank", "64",
        "--batch-size", "16",
    )
    assert "epoch 0" in out


def test_example_benchmark_tiny():
    out = _run_example(
        2, "pytorch_benchmark.py", "--model", "resnet18", "--batch-size", "2",
        "--num-warmup-batches", "1", "--num-batches-per-iter", "1",
        "--num-iters", "1", "--no-cuda",
    )
    assert "Total img/sec" in out


def test_example_resnet_tiny():
    out = _run_example(
        2, "pytorch_resnet.py", "--epochs", "1", "--train-samples", "32",
        "--val-samples", "16", "--batch-size", "8", "--no-checkpoint",
    )
    assert "epoch 0" in out


def test_example_average_consensus_async():
    out = _run_example(2, "pytorch_average_consensus.py", "--asynchronous-mode",
                       "--data-size", "1000")
    assert "consensus reached" in out or "relative error" in out
