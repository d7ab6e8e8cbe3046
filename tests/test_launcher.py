"""L3 launcher tests: env fan-out, failure propagation, fault injection,
elastic relaunch."""

import sys

from deeplearning_cfn_amd.launcher import LaunchSpec, launch_ranks

PY = sys.executable

CHECK_ENV = (
    "import os,sys;"
    "assert os.environ['DEEPLEARNING_WORKERS_COUNT']=='2';"
    "assert os.environ['WORLD_SIZE']=='2';"
    "assert os.environ['RANK']==os.environ['LOCAL_RANK'];"
    "assert os.path.exists(os.environ['DEEPLEARNING_WORKERS_PATH']);"
    "print('rank-ok', os.environ['RANK'])"
)


def test_launch_two_ranks_env(tmp_path):
    spec = LaunchSpec(cmd=[PY, "-c", CHECK_ENV], num_ranks=2,
                      efs_mount=str(tmp_path / "efs"), tee=False)
    results = launch_ranks(spec, on_event=lambda m: None)
    assert [r.returncode for r in results] == [0, 0]
    for r in results:
        assert "rank-ok" in open(r.log_path).read()


def test_failure_kills_world(tmp_path):
    # rank 1 exits 7 quickly; rank 0 would sleep for 60 s — the watchdog
    # must kill it long before that.
    code = ("import os,time,sys;"
            "sys.exit(7) if os.environ['RANK']=='1' else time.sleep(60)")
    spec = LaunchSpec(cmd=[PY, "-c", code], num_ranks=2,
                      efs_mount=str(tmp_path / "efs"), tee=False)
    import time
    t0 = time.time()
    results = launch_ranks(spec, on_event=lambda m: None)
    assert time.time() - t0 < 30
    rcs = {r.rank: r.returncode for r in results}
    assert rcs[1] == 7
    assert rcs[0] != 0  # killed


def test_elastic_relaunch(tmp_path):
    # First attempt fails (marker file absent -> rank 0 exits 3 and creates
    # it); relaunch succeeds.
    marker = tmp_path / "marker"
    code = (f"import os,sys,pathlib;p=pathlib.Path({str(marker)!r});\n"
            "if os.environ['RANK']=='0' and not p.exists():\n"
            "    p.touch(); sys.exit(3)\n"
            "print('ok')")
    spec = LaunchSpec(cmd=[PY, "-c", code], num_ranks=2, elastic=True,
                      max_restarts=2, efs_mount=str(tmp_path / "efs"),
                      tee=False)
    events = []
    results = launch_ranks(spec, on_event=events.append)
    assert [r.returncode for r in results] == [0, 0]
    assert any("elastic relaunch" in e for e in events)


def test_cli_up_and_run(tmp_path, capsys):
    from deeplearning_cfn_amd import cli
    rc = cli.main(["up", "-n", "2", "--efs-mount", str(tmp_path / "efs")])
    assert rc == 0
    out = capsys.readouterr().out
    assert "DEEPLEARNING_WORKERS_PATH" in out
    rc = cli.main(["run", "-n", "2", "--efs-mount", str(tmp_path / "efs"),
                   "--", PY, "-c", "print('hi')"])
    assert rc == 0
