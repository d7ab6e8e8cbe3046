"""L3 — the job launcher.

The reference launches jobs with external fan-out tools: dmlc
``tools/launch.py -n $COUNT -H $PATH`` (SSH per-host, sets ``DMLC_*``) for
the MXNet example, and ``mpirun -np N -hostfile $PATH`` (orted, sets
``OMPI_*``) for the Horovod example (SURVEY.md §2 L3, §4.2–4.3).

On one MI355X node both collapse to fork/exec of N local rank processes:

* one process per GPU (``LOCAL_RANK`` selects the device; RCCL — the
  "nccl" backend of torch.distributed on ROCm — rides xGMI between them);
* the env contract (contract.py) plus rendezvous vars exported per rank;
* stdout/stderr multiplexed with a ``[rank k]`` prefix and mirrored to
  per-rank log files under ``$EFS_MOUNT/logs``;
* a watchdog: the first rank to die non-zero kills the world (the
  reference's "a dead worker hangs the job" contract, modernised —
  SURVEY.md §6.3), with optional auto-relaunch from the newest checkpoint
  (``--elastic``);
* fault injection for tests: ``DLX_FAULT_RANK`` / ``DLX_FAULT_STEP`` make
  a chosen rank exit mid-run to exercise the watchdog path.
"""

from __future__ import annotations

import os
import signal
import socket
import subprocess
import sys
import threading
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Callable, List, Optional, Sequence

from ..contract import EnvContract, make_contract


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@dataclass
class LaunchSpec:
    """What to run and how wide."""

    cmd: Sequence[str]                       # argv of the training script
    num_ranks: int = 1
    gpus_per_host: int = 8
    efs_mount: str = "/tmp/dlx_efs"
    hostfile: Optional[str] = None           # -H; default $EFS_MOUNT/workers
    master_addr: str = "127.0.0.1"
    master_port: Optional[int] = None
    cuda_visible: bool = True                # pin HIP_VISIBLE_DEVICES per rank
    elastic: bool = False                    # auto-relaunch on rank failure
    max_restarts: int = 2
    log_dir: Optional[str] = None            # default $EFS_MOUNT/logs
    extra_env: dict = field(default_factory=dict)
    tee: bool = True                         # mirror rank output to console


@dataclass
class RankResult:
    rank: int
    returncode: int
    log_path: str


class _RankProc:
    def __init__(self, rank: int, proc: subprocess.Popen, log_path: Path,
                 tee: bool):
        self.rank = rank
        self.proc = proc
        self.log_path = log_path
        self._log = open(log_path, "ab", buffering=0)
        self._thread = threading.Thread(
            target=self._pump, args=(tee,), daemon=True)
        self._thread.start()

    def _pump(self, tee: bool) -> None:
        prefix = f"[rank {self.rank}] ".encode()
        assert self.proc.stdout is not None
        for line in self.proc.stdout:
            self._log.write(line)
            if tee:
                try:
                    sys.stdout.buffer.write(prefix + line)
                    sys.stdout.buffer.flush()
                except ValueError:
                    break  # interpreter shutting down

    def close(self) -> None:
        self._thread.join(timeout=5)
        self._log.close()


def _rank_env(spec: LaunchSpec, contract: EnvContract, rank: int,
              port: int) -> dict:
    env = contract.export()
    env.update(spec.extra_env)
    env.update({
        "RANK": str(rank),
        "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(spec.num_ranks),
        "MASTER_ADDR": spec.master_addr,
        "MASTER_PORT": str(port),
        # dmlc launch.py compatibility shadows (SURVEY.md §3.2 PS-Lite row):
        # consumers of the old contract can still read role/counts.
        "DMLC_ROLE": "worker",
        "DMLC_NUM_WORKER": str(spec.num_ranks),
        "DMLC_NUM_SERVER": "0",
    })
    # dmabuf IPC is the only mode the host driver supports (env note).
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if spec.cuda_visible:
        # Pin one GPU per rank BEFORE the child imports torch (setting it
        # inside the process after context init is ineffective).  The
        # child then sees exactly one device; engine/dist.py's
        # ``set_device(local_rank % count)`` resolves to device 0.
        env.setdefault("HIP_VISIBLE_DEVICES",
                       str(rank % max(spec.gpus_per_host, 1)))
    return env


def launch_ranks(
    spec: LaunchSpec,
    on_event: Optional[Callable[[str], None]] = None,
) -> List[RankResult]:
    """Spawn the world, babysit it, return per-rank results.

    Failure semantics: first non-zero exit → SIGTERM the
    rest (SIGKILL after a grace period).  With ``spec.elastic`` the whole
    world is relaunched up to ``max_restarts`` times; the training script
    is expected to resume from its newest checkpoint (engine/checkpoint).
    """
    say = on_event or (lambda m: print(f"[dlx] {m}", flush=True))
    attempt = 0
    while True:
        results = _launch_once(spec, say)
        bad = [r for r in results if r.returncode != 0]
        if not bad:
            return results
        if spec.elastic and attempt < spec.max_restarts:
            attempt += 1
            say(f"rank(s) {[r.rank for r in bad]} failed "
                f"(rc={[r.returncode for r in bad]}); elastic relaunch "
                f"{attempt}/{spec.max_restarts}")
            continue
        return results


def _launch_once(spec: LaunchSpec, say: Callable[[str], None]) -> List[RankResult]:
    contract = make_contract(spec.num_ranks, spec.gpus_per_host,
                             spec.efs_mount, spec.hostfile)
    port = spec.master_port or find_free_port()
    log_dir = Path(spec.log_dir or (Path(spec.efs_mount) / "logs"))
    log_dir.mkdir(parents=True, exist_ok=True)

    procs: List[_RankProc] = []
    say(f"launching {spec.num_ranks} rank(s): {' '.join(spec.cmd)} "
        f"(rendezvous {spec.master_addr}:{port})")
    for rank in range(spec.num_ranks):
        env = _rank_env(spec, contract, rank, port)
        p = subprocess.Popen(
            list(spec.cmd), env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, start_new_session=True)
        procs.append(_RankProc(rank, p, log_dir / f"rank{rank}.log", spec.tee))

    # Watchdog: poll for exits; first failure kills the world.
    results: dict[int, int] = {}
    try:
        while len(results) < len(procs):
            for rp in procs:
                if rp.rank in results:
                    continue
                rc = rp.proc.poll()
                if rc is None:
                    continue
                results[rp.rank] = rc
                if rc != 0:
                    say(f"rank {rp.rank} exited rc={rc}; terminating world")
                    _kill_world(procs, results)
            time.sleep(0.05)
    except KeyboardInterrupt:
        say("interrupted; terminating world")
        _kill_world(procs, results, sig=signal.SIGKILL)
        raise
    finally:
        for rp in procs:
            rp.close()
    return [RankResult(rp.rank, results.get(rp.rank, -1), str(rp.log_path))
            for rp in procs]


def _kill_world(procs: List["_RankProc"], results: dict,
                sig: int = signal.SIGTERM, grace: float = 5.0) -> None:
    """Terminate every still-running rank by exact PID/process group."""
    for rp in procs:
        if rp.rank in results:
            continue
        try:
            os.killpg(os.getpgid(rp.proc.pid), sig)
        except (ProcessLookupError, PermissionError):
            pass
    deadline = time.time() + grace
    for rp in procs:
        if rp.rank in results:
            continue
        timeout = max(0.0, deadline - time.time())
        try:
            rp.proc.wait(timeout=timeout)
        except subprocess.TimeoutExpired:
            try:
                os.killpg(os.getpgid(rp.proc.pid), signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                pass
