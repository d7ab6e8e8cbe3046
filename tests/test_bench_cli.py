"""bench.py contract tests — the driver invokes it exactly like this.

Single-process default run and a 2-rank gloo run (the shape of the
driver's `torch.distributed.run --nproc-per-node N bench.py` launch),
both on CPU.  Asserts the ONE-JSON-line output contract.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _last_json(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


def test_bench_single_process(tmp_path):
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch-size", "4"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json(r.stdout)
    assert j["n_gpus"] == 1
    assert j["unit"] == "images/sec"
    assert j["value"] > 0
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["config"]["model"] == "resnet50"


def test_bench_two_ranks_gloo(tmp_path):
    """Two ranks rendezvous on 127.0.0.1 and only rank 0 prints JSON."""
    port = 29731
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), LOCAL_RANK=str(rank),
                   WORLD_SIZE="2", MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
             "--batch-size", "4"],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=300) for p in procs]
    for p, (out, err) in zip(procs, outs):
        assert p.returncode == 0, err[-2000:]
    j = _last_json(outs[0][0])
    assert j["n_gpus"] == 2
    assert j["config"]["global_batch"] == 8
    # rank 1 must not print the JSON line
    assert "{" not in outs[1][0]
