"""Elastic auto-resume + optimizer-state-in-checkpoint.

Reference semantics: an elastic relaunch resumes from the last checkpoint
on the shared mount; MXNet's ``--model-prefix`` /
``--load-epoch`` workflow assumes momentum survives resume (SURVEY.md
§6.4).  These tests prove: (a) a resumed run reproduces an uninterrupted
run's loss curve exactly (momentum + running stats intact), (b) ``dlx run
--elastic`` with mid-epoch fault injection completes by resuming at the
right epoch, (c) the per-epoch Validation-accuracy line appears.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

import torch

PY = sys.executable
REPO = Path(__file__).resolve().parent.parent

# pool == steps-per-epoch so the data stream position coincides with the
# epoch boundary: a resumed process replays exactly the batches the
# uninterrupted run saw in its second epoch.
BASE = ["--network", "resnet", "--num-layers", "20", "--dataset", "cifar10",
        "--batch-size", "8", "--steps-per-epoch", "4", "--data-pool", "4",
        "--disp-batches", "2", "--dtype", "fp32", "--eval-steps", "2"]


def _run(args, env_extra=None, timeout=600):
    env = dict(os.environ)
    env["DLX_IMPL"] = "torch"
    if env_extra:
        env.update(env_extra)
    return subprocess.run(args, cwd=str(REPO), env=env, timeout=timeout,
                          capture_output=True, text=True)


def _train_losses(efs):
    out = []
    for line in (Path(efs) / "metrics.jsonl").read_text().splitlines():
        rec = json.loads(line)
        if rec["kind"] == "train":
            out.append((rec["epoch"], rec["step"], rec["loss"]))
    return out


def test_auto_resume_matches_uninterrupted(tmp_path):
    # run A: 2 epochs straight through
    efs_a = tmp_path / "efs_a"
    pre_a = str(tmp_path / "a" / "r20")
    r = _run([PY, "-m", "deeplearning_cfn_amd.train", *BASE,
              "--num-epochs", "2", "--model-prefix", pre_a],
             env_extra={"EFS_MOUNT": str(efs_a)})
    assert r.returncode == 0, r.stdout + r.stderr

    # run B: 1 epoch, then a fresh process with NO --load-epoch must
    # auto-resume from epoch 1 (latest checkpoint under the prefix)
    efs_b = tmp_path / "efs_b"
    pre_b = str(tmp_path / "b" / "r20")
    r = _run([PY, "-m", "deeplearning_cfn_amd.train", *BASE,
              "--num-epochs", "1", "--model-prefix", pre_b],
             env_extra={"EFS_MOUNT": str(efs_b)})
    assert r.returncode == 0, r.stdout + r.stderr
    r = _run([PY, "-m", "deeplearning_cfn_amd.train", *BASE,
              "--num-epochs", "2", "--model-prefix", pre_b],
             env_extra={"EFS_MOUNT": str(efs_b)})
    assert r.returncode == 0, r.stdout + r.stderr
    assert "resume" in r.stdout, r.stdout

    a = [x for x in _train_losses(efs_a) if x[0] == 1]
    b = [x for x in _train_losses(efs_b) if x[0] == 1]
    assert a and len(a) == len(b)
    for (ea, sa, la), (eb, sb, lb) in zip(a, b):
        assert (ea, sa) == (eb, sb)
        # optimizer momentum + fp32 exactness: curves must coincide
        assert abs(la - lb) < 1e-6, (a, b)

    # checkpointed weights identical too
    wa = torch.load(pre_a + "-0002.params", map_location="cpu",
                    weights_only=False)["model"]
    wb = torch.load(pre_b + "-0002.params", map_location="cpu",
                    weights_only=False)["model"]
    for k in wa:
        assert torch.equal(wa[k], wb[k]), k

    # optimizer state must be in the checkpoint
    st = torch.load(pre_a + "-0002.params", map_location="cpu",
                    weights_only=False)
    assert st["optimizer"] is not None
    assert "momentum_bufs" in st["optimizer"]


def test_elastic_relaunch_resumes_at_checkpoint(tmp_path):
    efs = tmp_path / "efs"
    prefix = str(tmp_path / "ck" / "r20")
    # fault at global step 6 = epoch 1 step 2 (after the epoch-0
    # checkpoint); the relaunched process resumes at epoch 1 where
    # global_step restarts from 0 and never reaches 6 again.
    r = _run([PY, "-m", "deeplearning_cfn_amd.cli", "run", "-n", "2",
              "--efs-mount", str(efs), "--elastic", "--max-restarts", "1",
              "--impl", "torch", "--",
              PY, "-m", "deeplearning_cfn_amd.train", *BASE,
              "--num-epochs", "2", "--model-prefix", prefix,
              "--fault-step", "6"],
             env_extra={"DLX_FAULT_RANK": "1"})
    assert r.returncode == 0, r.stdout[-4000:] + r.stderr[-2000:]
    assert "elastic relaunch" in r.stdout
    assert "resume" in r.stdout          # relaunched ranks picked up epoch 1
    assert Path(prefix + "-0002.params").exists()


def test_validation_accuracy_line(tmp_path):
    efs = tmp_path / "efs"
    r = _run([PY, "-m", "deeplearning_cfn_amd.train", *BASE,
              "--num-epochs", "1", "--checkpoint-every", "0"],
             env_extra={"EFS_MOUNT": str(efs)})
    assert r.returncode == 0, r.stdout + r.stderr
    # the MXNet example's per-epoch line shape
    assert "Epoch[0] Validation-accuracy=" in r.stdout, r.stdout
    evals = [json.loads(l) for l in
             (efs / "metrics.jsonl").read_text().splitlines()
             if json.loads(l)["kind"] == "eval"]
    assert evals and 0.0 <= evals[0]["validation_accuracy"] <= 1.0
