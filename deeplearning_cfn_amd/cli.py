"""``dlx`` — the job-submit CLI (L3 public surface).

Maps the reference's operator workflow onto one MI355X node
(SURVEY.md §2, §4):

* ``dlx up``      — preflight: enumerate GPUs, write the hostfile, print
                    the env contract (== stack CREATE_COMPLETE).
* ``dlx run``     — launch a distributed job: ``dlx run -n 8 -- python -m
                    deeplearning_cfn_amd.train --network resnet ...``.
                    ``-n/--num-ranks`` and ``-H/--hostfile`` mirror dmlc
                    ``launch.py``; ``--elastic`` adds watchdog relaunch.
* ``dlx kill``    — the README's "kill stuck runs" loop, local-process
                    edition (kills the recorded pgid of the last run).
* ``dlx profile`` — wrap a command in rocprofv3 kernel-trace capture.

Usage: ``python -m deeplearning_cfn_amd.cli <subcommand> ...``
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
from pathlib import Path

from .contract import make_contract
from .launcher import LaunchSpec, launch_ranks
from .topo import preflight, probe

_STATE = Path(os.environ.get("DLX_STATE_DIR", "/tmp/dlx_state"))


def cmd_up(args) -> int:
    topo = probe(require_gpu=False)
    n = args.num_ranks or max(topo.n_gpus, 1)
    if topo.n_gpus:
        preflight(n)
    contract = make_contract(n, topo.n_gpus or 8, args.efs_mount,
                             args.hostfile)
    print(f"# node: {topo.n_gpus} GPU(s) "
          f"{topo.gpu_names[0] if topo.gpu_names else '(none visible)'}")
    print("# export these (or let `dlx run` export them per rank):")
    for k, v in [("DEEPLEARNING_WORKERS_PATH", contract.workers_path),
                 ("DEEPLEARNING_WORKERS_COUNT", contract.workers_count),
                 ("DEEPLEARNING_WORKER_GPU_COUNT", contract.worker_gpu_count),
                 ("EFS_MOUNT", contract.efs_mount)]:
        print(f"export {k}={v}")
    return 0


def cmd_run(args, cmd: list[str]) -> int:
    if not cmd:
        print("dlx run: no command given (use: dlx run -n 8 -- python ...)",
              file=sys.stderr)
        return 2
    spec = LaunchSpec(
        cmd=cmd, num_ranks=args.num_ranks, efs_mount=args.efs_mount,
        hostfile=args.hostfile, elastic=args.elastic,
        max_restarts=args.max_restarts,
        gpus_per_host=args.gpus_per_host,
        cuda_visible=not args.no_pin_gpus,
        extra_env={"DLX_IMPL": args.impl} if args.impl else {},
    )
    _STATE.mkdir(parents=True, exist_ok=True)
    (_STATE / "last_run.json").write_text(json.dumps(
        {"pid": os.getpid(), "cmd": cmd, "num_ranks": args.num_ranks}))
    results = launch_ranks(spec)
    bad = [r for r in results if r.returncode != 0]
    for r in results:
        print(f"rank {r.rank}: rc={r.returncode} log={r.log_path}")
    return 1 if bad else 0


def cmd_kill(args) -> int:
    f = _STATE / "last_run.json"
    if not f.exists():
        print("dlx kill: no recorded run")
        return 0
    info = json.loads(f.read_text())
    pid = info["pid"]
    try:
        os.killpg(os.getpgid(pid), signal.SIGTERM)
        print(f"sent SIGTERM to launcher pgid of pid {pid}")
    except ProcessLookupError:
        print(f"launcher pid {pid} already gone")
    return 0


def cmd_profile(args, cmd: list[str]) -> int:
    out = Path(args.output)
    out.mkdir(parents=True, exist_ok=True)
    wrapped = ["rocprofv3", "--kernel-trace", "--stats", "-d", str(out),
               "--"] + cmd
    print("+ " + " ".join(wrapped))
    return subprocess.call(wrapped)


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    # split off the trailing `-- cmd...`
    tail: list[str] = []
    if "--" in argv:
        i = argv.index("--")
        argv, tail = argv[:i], argv[i + 1:]

    p = argparse.ArgumentParser(prog="dlx")
    sub = p.add_subparsers(dest="sub", required=True)

    up = sub.add_parser("up", help="preflight + env contract")
    up.add_argument("-n", "--num-ranks", type=int, default=None)
    up.add_argument("-H", "--hostfile", default=None)
    up.add_argument("--efs-mount", default=os.environ.get("EFS_MOUNT",
                                                          "/tmp/dlx_efs"))

    run = sub.add_parser("run", help="launch a distributed job")
    run.add_argument("-n", "--num-ranks", type=int, default=1)
    run.add_argument("-H", "--hostfile", default=None)
    run.add_argument("--efs-mount", default=os.environ.get("EFS_MOUNT",
                                                           "/tmp/dlx_efs"))
    run.add_argument("--elastic", action="store_true")
    run.add_argument("--max-restarts", type=int, default=2)
    run.add_argument("--impl", default=None, help="auto|native|torch")
    run.add_argument("--gpus-per-host", type=int, default=8,
                     help="devices per host; rank r is pinned to GPU "
                          "r %% gpus-per-host via HIP_VISIBLE_DEVICES "
                          "(1 => all ranks share device 0)")
    run.add_argument("--no-pin-gpus", action="store_true",
                     help="do not set per-rank HIP_VISIBLE_DEVICES")

    sub.add_parser("kill", help="terminate the last launched run")

    prof = sub.add_parser("profile", help="rocprofv3 capture of a command")
    prof.add_argument("-o", "--output", default="gpurun_out/prof")

    args = p.parse_args(argv)
    if args.sub == "up":
        return cmd_up(args)
    if args.sub == "run":
        return cmd_run(args, tail)
    if args.sub == "kill":
        return cmd_kill(args)
    if args.sub == "profile":
        return cmd_profile(args, tail)
    return 2


if __name__ == "__main__":
    sys.exit(main())
