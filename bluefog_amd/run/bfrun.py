# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""``bfrun`` — process launcher.

The reference's bfrun composes an ``mpirun`` command line with NIC
discovery over vendored Horovod driver/task services (reference:
bluefog/run/run.py:121-203, horovod_driver.py). This framework has no MPI:
``bfrun -np N python train.py`` spawns N local processes with the
torchrun-style env rendezvous (RANK / LOCAL_RANK / WORLD_SIZE /
MASTER_ADDR / MASTER_PORT) that ``bf.init()`` consumes — one process per
GPU on one MI355X node. It propagates failures (first non-zero exit kills
the job) and forwards SIGINT/SIGTERM to children.

Usage:
    bfrun -np 8 python examples/pytorch_benchmark.py
    python -m bluefog_amd.run.bfrun -np 8 python train.py
"""

import argparse
import os
import signal
import socket
import subprocess
import sys
import time


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        prog="bfrun", description="bluefog_amd local process launcher"
    )
    p.add_argument("-np", "--num-proc", type=int, required=True,
                   help="number of processes (one per GPU)")
    p.add_argument("--master-addr", default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=0,
                   help="rendezvous port (0 = pick a free one)")
    p.add_argument("--extra-env", action="append", default=[],
                   help="KEY=VALUE to add to every rank's environment")
    p.add_argument("command", nargs=argparse.REMAINDER,
                   help="program and arguments to launch")
    args = p.parse_args(argv)
    if not args.command:
        p.error("no command given")
    if args.command[0] == "--":
        args.command = args.command[1:]
    return args


def main(argv=None) -> int:
    args = parse_args(argv)
    n = args.num_proc
    port = args.master_port or _free_port()
    procs = []
    try:
        for rank in range(n):
            env = dict(os.environ)
            env.update(
                RANK=str(rank),
                LOCAL_RANK=str(rank),
                WORLD_SIZE=str(n),
                MASTER_ADDR=args.master_addr,
                MASTER_PORT=str(port),
            )
            env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
            # make bluefog_amd importable from scripts in subdirectories
            repo_root = os.path.dirname(
                os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
            )
            env["PYTHONPATH"] = repo_root + os.pathsep + env.get("PYTHONPATH", "")
            for kv in args.extra_env:
                k, _, v = kv.partition("=")
                env[k] = v
            procs.append(subprocess.Popen(args.command, env=env))

        def forward(signum, frame):
            for p in procs:
                try:
                    p.send_signal(signum)
                except Exception:
                    pass

        signal.signal(signal.SIGINT, forward)
        signal.signal(signal.SIGTERM, forward)

        exit_code = 0
        live = {p.pid: (r, p) for r, p in enumerate(procs)}
        while live:
            for pid, (rank, p) in list(live.items()):
                rc = p.poll()
                if rc is None:
                    continue
                live.pop(pid, None)
                if rc != 0:
                    sys.stderr.write(
                        f"bfrun: rank {rank} exited with code {rc}; "
                        "terminating remaining ranks\n"
                    )
                    exit_code = rc
                    for _, q in live.values():
                        q.terminate()
                    for _, q in live.values():
                        try:
                            q.wait(timeout=10)
                        except subprocess.TimeoutExpired:
                            q.kill()
                    live.clear()
                    break
            time.sleep(0.1)
        return exit_code
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()


if __name__ == "__main__":
    sys.exit(main())
