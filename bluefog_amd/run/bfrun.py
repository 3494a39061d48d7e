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

Multi-host (reference run.py ``-H``): ``bfrun -np 16 -H hostA:8,hostB:8
python train.py`` launches the hostA ranks locally (bfrun must run on the
first listed host) and the remaining rank blocks over ``ssh`` — each
remote invokes this module with the same rendezvous env pointing at the
first host. No NIC discovery dance: the rendezvous address is the first
hostname (override with --master-addr), which is the interface every
host must route to anyway.

Usage:
    bfrun -np 8 python examples/pytorch_benchmark.py
    bfrun -np 16 -H node0:8,node1:8 python train.py
    python -m bluefog_amd.run.bfrun -np 8 python train.py
"""

import argparse
import os
import shlex
import signal
import socket
import subprocess
import sys
import time


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def parse_hosts(spec: str, np_: int):
    """``"hostA:8,hostB:8"`` -> [(hostA, 8), (hostB, 8)]; slot sum must
    cover -np (reference: run.py hosts argument)."""
    out = []
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        host, _, slots = part.partition(":")
        if not host or not slots.isdigit() or int(slots) < 1:
            raise ValueError(f"bad host spec {part!r}; expected host:slots")
        out.append((host, int(slots)))
    total = sum(s for _, s in out)
    if total < np_:
        raise ValueError(
            f"host list provides {total} slots but -np is {np_}"
        )
    return out


def rank_blocks(hosts, np_):
    """Assign global rank ranges to hosts in order: [(host, first_rank,
    n_ranks), ...] until np_ ranks are placed."""
    blocks = []
    next_rank = 0
    for host, slots in hosts:
        if next_rank >= np_:
            break
        n = min(slots, np_ - next_rank)
        blocks.append((host, next_rank, n))
        next_rank += n
    return blocks


def remote_command(host, first_rank, n, world_size, master_addr, port,
                   command, extra_env):
    """Compose the ssh command that runs a rank block on ``host`` by
    re-invoking this module there with --local-* placement flags."""
    inner = [
        "python3", "-m", "bluefog_amd.run.bfrun",
        "-np", str(world_size),
        "--master-addr", master_addr,
        "--master-port", str(port),
        "--local-first-rank", str(first_rank),
        "--local-num", str(n),
    ]
    for kv in extra_env:
        inner += ["--extra-env", kv]
    inner += ["--", *command]
    # run from the same cwd; PYTHONPATH must make bluefog_amd importable
    wrapped = f"cd {shlex.quote(os.getcwd())} && " + " ".join(
        shlex.quote(a) for a in inner
    )
    return ["ssh", "-o", "StrictHostKeyChecking=no", host, wrapped]


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        prog="bfrun", description="bluefog_amd process launcher"
    )
    p.add_argument("-np", "--num-proc", type=int, required=True,
                   help="total number of processes (one per GPU)")
    p.add_argument("-H", "--hosts", default=None,
                   help="comma-separated host:slots list for multi-host "
                        "launch (run bfrun on the first host)")
    p.add_argument("--master-addr", default=None)
    p.add_argument("--master-port", type=int, default=0,
                   help="rendezvous port (0 = pick a free one)")
    p.add_argument("--extra-env", action="append", default=[],
                   help="KEY=VALUE to add to every rank's environment")
    p.add_argument("--local-first-rank", type=int, default=0,
                   help=argparse.SUPPRESS)  # internal: remote block start
    p.add_argument("--local-num", type=int, default=None,
                   help=argparse.SUPPRESS)  # internal: remote block size
    p.add_argument("command", nargs=argparse.REMAINDER,
                   help="program and arguments to launch")
    args = p.parse_args(argv)
    if not args.command:
        p.error("no command given")
    if args.command[0] == "--":
        args.command = args.command[1:]
    return args


def _spawn_local_rank(args, rank, local_rank, world_size, master_addr, port):
    env = dict(os.environ)
    env.update(
        RANK=str(rank),
        LOCAL_RANK=str(local_rank),
        WORLD_SIZE=str(world_size),
        MASTER_ADDR=master_addr,
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
    return subprocess.Popen(args.command, env=env)


def main(argv=None) -> int:
    args = parse_args(argv)
    n = args.num_proc
    procs = []  # (label, Popen)
    try:
        if args.hosts and args.local_num is None:
            # multi-host: this process runs on the first listed host
            hosts = parse_hosts(args.hosts, n)
            blocks = rank_blocks(hosts, n)
            master_addr = args.master_addr or blocks[0][0]
            port = args.master_port or _free_port()
            first_host, first_rank0, first_n = blocks[0]
            for lr in range(first_n):
                procs.append(
                    (f"rank {first_rank0 + lr}",
                     _spawn_local_rank(args, first_rank0 + lr, lr, n,
                                       master_addr, port)))
            for host, fr, cnt in blocks[1:]:
                cmd = remote_command(host, fr, cnt, n, master_addr, port,
                                     args.command, args.extra_env)
                procs.append((f"host {host} (ranks {fr}..{fr + cnt - 1})",
                              subprocess.Popen(cmd)))
        else:
            # single host, or the remote half of a multi-host launch
            # (--local-first-rank/--local-num set by remote_command)
            master_addr = args.master_addr or "127.0.0.1"
            port = args.master_port or _free_port()
            first = args.local_first_rank
            count = args.local_num if args.local_num is not None else n
            for lr in range(count):
                procs.append(
                    (f"rank {first + lr}",
                     _spawn_local_rank(args, first + lr, lr, n, master_addr,
                                       port)))

        def forward(signum, frame):
            for _, p in procs:
                try:
                    p.send_signal(signum)
                except Exception:
                    pass

        signal.signal(signal.SIGINT, forward)
        signal.signal(signal.SIGTERM, forward)

        exit_code = 0
        live = {p.pid: (label, p) for label, p in procs}
        while live:
            for pid, (label, p) in list(live.items()):
                rc = p.poll()
                if rc is None:
                    continue
                live.pop(pid, None)
                if rc != 0:
                    sys.stderr.write(
                        f"bfrun: {label} exited with code {rc}; "
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
        for _, p in procs:
            if p.poll() is None:
                p.kill()


if __name__ == "__main__":
    sys.exit(main())
