"""L2 env-contract unit tests (SURVEY.md §5 plan item 1)."""

import os

from deeplearning_cfn_amd import contract


def test_roundtrip(tmp_path):
    c = contract.make_contract(8, 8, tmp_path / "efs")
    env = c.export({})
    assert env["DEEPLEARNING_WORKERS_COUNT"] == "8"
    assert env["DEEPLEARNING_WORKER_GPU_COUNT"] == "8"
    assert os.path.isdir(env["EFS_MOUNT"])
    back = contract.EnvContract.from_env(env)
    assert back == c


def test_hostfile(tmp_path):
    hf = tmp_path / "workers"
    contract.write_hostfile(hf, ["localhost"] * 3)
    assert contract.read_hostfile(hf) == ["localhost"] * 3
    hf.write_text("# comment\nhostA slots=8\n\nhostB\n")
    assert contract.read_hostfile(hf) == ["hostA", "hostB"]


def test_missing_env_raises():
    try:
        contract.EnvContract.from_env({})
    except KeyError as e:
        assert "DEEPLEARNING" in str(e)
    else:
        raise AssertionError("expected KeyError")


def test_topo_probe_no_gpu_ok():
    from deeplearning_cfn_amd import topo
    t = topo.probe(require_gpu=False)
    assert t.n_gpus >= 0
    assert t.xgmi_links_per_gpu == 7
