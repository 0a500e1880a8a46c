"""CLI (reference python/pathway/cli.py): spawn / spawn-from-env / replay.

`python -m pathway_amd spawn --processes N program.py` launches one worker
process per GPU (RANK/WORLD_SIZE + PATHWAY_* env), supervises them, and
implements the elastic up/downscale restart protocol
(reference cli.py:211-374 + workload_tracker.rs exit codes).
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys

EXIT_CODE_UPSCALE = 77
EXIT_CODE_DOWNSCALE = 78
UPSCALING_FACTOR = 2


def _launch(program: list[str], processes: int, threads: int, first_port: int, extra_env: dict) -> int:
    procs = []
    for pid in range(processes):
        env = dict(os.environ)
        env.update(extra_env)
        env.update(
            {
                "PATHWAY_THREADS": str(threads),
                "PATHWAY_PROCESSES": str(processes),
                "PATHWAY_PROCESS_ID": str(pid),
                "PATHWAY_FIRST_PORT": str(first_port),
                "RANK": str(pid),
                "LOCAL_RANK": str(pid),
                "WORLD_SIZE": str(processes),
                "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(first_port),
            }
        )
        procs.append(subprocess.Popen([sys.executable, *program], env=env))
    exit_code = 0
    try:
        for p in procs:
            rc = p.wait()
            if rc != 0:
                exit_code = rc
                for q in procs:
                    if q.poll() is None and rc not in (EXIT_CODE_UPSCALE, EXIT_CODE_DOWNSCALE):
                        q.terminate()
    except KeyboardInterrupt:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGINT)
        for p in procs:
            p.wait()
        raise
    return exit_code


def spawn(args) -> None:
    processes = args.processes
    import os as _os

    from pathway_amd.internals.license import check_worker_limit

    check_worker_limit(
        processes * max(args.threads, 1),
        _os.environ.get("PATHWAY_LICENSE_KEY"),
    )
    while True:
        rc = _launch(
            args.program, processes, args.threads, args.first_port, {}
        )
        if rc == EXIT_CODE_UPSCALE:
            processes = min(processes * UPSCALING_FACTOR, 8)
            print(f"[pathway_amd] upscaling to {processes} workers", file=sys.stderr)
            continue
        if rc == EXIT_CODE_DOWNSCALE:
            processes = max(processes // UPSCALING_FACTOR, 1)
            print(f"[pathway_amd] downscaling to {processes} workers", file=sys.stderr)
            continue
        sys.exit(rc)


def spawn_from_env(args) -> None:
    program = os.environ.get("PATHWAY_SPAWN_PROGRAM", "")
    if not program:
        print("PATHWAY_SPAWN_PROGRAM not set", file=sys.stderr)
        sys.exit(2)
    ns = argparse.Namespace(
        program=program.split(),
        processes=int(os.environ.get("PATHWAY_SPAWN_PROCESSES", "1")),
        threads=int(os.environ.get("PATHWAY_SPAWN_THREADS", "1")),
        first_port=int(os.environ.get("PATHWAY_FIRST_PORT", "29500")),
    )
    spawn(ns)


def replay(args) -> None:
    env = {
        "PATHWAY_REPLAY_STORAGE": args.record_path,
        "PATHWAY_SNAPSHOT_ACCESS": args.mode,
    }
    rc = _launch(args.program, 1, 1, args.first_port, env)
    sys.exit(rc)


def main(argv: list[str] | None = None) -> None:
    p = argparse.ArgumentParser(prog="pathway_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    ps = sub.add_parser("spawn", help="launch a multi-worker program")
    ps.add_argument("--processes", "-n", type=int, default=1)
    ps.add_argument("--threads", "-t", type=int, default=1)
    ps.add_argument("--first-port", type=int, default=29500)
    ps.add_argument("program", nargs=argparse.REMAINDER)
    ps.set_defaults(fn=spawn)

    pe = sub.add_parser("spawn-from-env")
    pe.set_defaults(fn=spawn_from_env)

    pr = sub.add_parser("replay", help="record/replay a program run")
    pr.add_argument("--record-path", default="./record")
    pr.add_argument("--mode", choices=["record", "replay"], default="replay")
    pr.add_argument("--first-port", type=int, default=29500)
    pr.add_argument("program", nargs=argparse.REMAINDER)
    pr.set_defaults(fn=replay)

    args = p.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
