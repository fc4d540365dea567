#!/usr/bin/env python3
"""Local N-process launcher (the reference's `mpirun -np N` / `horovodrun`
slot — SURVEY.md §3.5).  One process per GPU, env rendezvous:

    python tools/launch.py --nproc 8 [--master-port P] script.py [args...]

Sets RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT and streams
children's output; first non-zero exit kills the job (fail-fast whole-job
semantics, as the reference's MPI launch — SURVEY.md §5.3)."""
import argparse
import os
import signal
import socket
import subprocess
import sys


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--nproc", type=int, required=True)
    p.add_argument("--master-addr", default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=0)
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    args = p.parse_args()
    if not args.cmd:
        p.error("missing command")
    cmd = args.cmd
    if cmd[0].endswith(".py"):
        cmd = [sys.executable] + cmd
    port = args.master_port or free_port()

    procs = []
    for rank in range(args.nproc):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(args.nproc),
            "LOCAL_WORLD_SIZE": str(args.nproc),
            "MASTER_ADDR": args.master_addr,
            "MASTER_PORT": str(port),
        })
        procs.append(subprocess.Popen(cmd, env=env))

    def kill_all(*_):
        for pr in procs:
            if pr.poll() is None:
                pr.terminate()

    signal.signal(signal.SIGINT, kill_all)
    signal.signal(signal.SIGTERM, kill_all)

    rc = 0
    try:
        while any(pr.poll() is None for pr in procs):
            for pr in procs:
                code = pr.poll()
                if code is not None and code != 0:
                    sys.stderr.write(f"rank exited rc={code}; killing job\n")
                    kill_all()
                    rc = code
            try:
                procs[0].wait(timeout=1)
            except subprocess.TimeoutExpired:
                pass
    finally:
        kill_all()
        for pr in procs:
            pr.wait()
            rc = rc or pr.returncode
    sys.exit(rc)


if __name__ == "__main__":
    main()
