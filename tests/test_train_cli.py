"""End-to-end trainer CLI tests on CPU.

Exercises the real surface: ``dlx run -n 2 -- python -m
deeplearning_cfn_amd.train --network resnet --kv-store dist_device_sync``
with the env contract, checkpoint layout, resume, and fault injection.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

PY = sys.executable
REPO = Path(__file__).resolve().parent.parent


def _run(args, env_extra=None, timeout=600):
    env = dict(os.environ)
    env["DLX_IMPL"] = "torch"
    if env_extra:
        env.update(env_extra)
    return subprocess.run(args, cwd=str(REPO), env=env, timeout=timeout,
                          capture_output=True, text=True)


def test_single_process_train_and_resume(tmp_path):
    efs = tmp_path / "efs"
    prefix = str(tmp_path / "ckpt" / "r20")
    base = [PY, "-m", "deeplearning_cfn_amd.train", "--network", "resnet",
            "--num-layers", "20", "--dataset", "cifar10",
            "--batch-size", "8", "--steps-per-epoch", "4",
            "--disp-batches", "2", "--num-epochs", "1",
            "--model-prefix", prefix, "--dtype", "fp32"]
    r = _run(base, env_extra={"EFS_MOUNT": str(efs)})
    assert r.returncode == 0, r.stdout + r.stderr
    assert Path(prefix + "-0001.params").exists()
    assert Path(prefix + "-symbol.json").exists()
    sym = json.loads(Path(prefix + "-symbol.json").read_text())
    assert sym["network"] == "resnet"
    metrics = (efs / "metrics.jsonl").read_text().splitlines()
    kinds = [json.loads(m)["kind"] for m in metrics]
    assert "train" in kinds and "checkpoint" in kinds

    # resume from epoch 1, run to epoch 2
    r = _run(base + ["--load-epoch", "1", "--num-epochs", "2"],
             env_extra={"EFS_MOUNT": str(efs)})
    assert r.returncode == 0, r.stdout + r.stderr
    assert Path(prefix + "-0002.params").exists()


def test_dlx_run_two_ranks_cifar(tmp_path):
    efs = tmp_path / "efs"
    r = _run([PY, "-m", "deeplearning_cfn_amd.cli", "run", "-n", "2",
              "--efs-mount", str(efs), "--impl", "torch", "--",
              PY, "-m", "deeplearning_cfn_amd.train", "--network", "resnet",
              "--num-layers", "20", "--dataset", "cifar10",
              "--batch-size", "4", "--steps-per-epoch", "3",
              "--disp-batches", "3", "--num-epochs", "1",
              "--checkpoint-every", "0", "--dtype", "fp32"])
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-2000:]
    assert "rank 0: rc=0" in r.stdout and "rank 1: rc=0" in r.stdout


def test_fault_injection_kills_world(tmp_path):
    efs = tmp_path / "efs"
    r = _run([PY, "-m", "deeplearning_cfn_amd.cli", "run", "-n", "2",
              "--efs-mount", str(efs), "--",
              PY, "-m", "deeplearning_cfn_amd.train", "--network", "resnet",
              "--num-layers", "20", "--dataset", "cifar10",
              "--batch-size", "4", "--steps-per-epoch", "50",
              "--num-epochs", "1", "--checkpoint-every", "0",
              "--dtype", "fp32", "--fault-step", "2"],
             env_extra={"DLX_FAULT_RANK": "1"})
    assert r.returncode == 1  # world failed, reported
    assert "rank 1: rc=13" in r.stdout


def test_bert_tiny_step():
    r = _run([PY, "-m", "deeplearning_cfn_amd.train", "--network", "bert",
              "--batch-size", "2", "--seq-len", "16",
              "--steps-per-epoch", "1", "--disp-batches", "1",
              "--num-epochs", "1", "--checkpoint-every", "0",
              "--dtype", "fp32", "--lr", "1e-4"], timeout=900)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-2000:]
