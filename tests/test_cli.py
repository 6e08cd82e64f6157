"""CLI drivers: 13-positional-arg parsing + the stdout contract
(reference run template, README.md; log lines SparkASGDThread.scala:29-65,
196, 349, 355-362, 400-410)."""

import io
import os
import re
from contextlib import redirect_stdout

import pytest

from asyncframework_amd.cli import drivers


ARGS13 = ["synthetic", "synthetic", "16", "200", "2", "30", "0.5", "1000000",
          "0.3", "0.5", "10", "0", "42"]
ARGS8 = ["synthetic", "synthetic", "16", "200", "2", "30", "0.5", "0.3"]


def _capture(fn, args):
    buf = io.StringIO()
    with redirect_stdout(buf):
        fn(args)
    return buf.getvalue()


def _check_contract(out, app):
    lines = out.splitlines()
    assert lines[0] == f"Spark {app} application started"
    assert lines[1] == "Input arguments:"
    assert any(l.startswith("Input format:") for l in lines)
    assert any(re.match(r"Elapsed time\(ms\): \d+", l) for l in lines)
    assert "finished" == lines[-1]
    # final time,objective CSV lines
    csv = [l for l in lines if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    assert len(csv) >= 1
    return lines


def test_asgd_thread_contract():
    out = _capture(drivers.asgd_thread, ARGS13)
    lines = _check_contract(out, "ASGD")
    assert any(re.match(r"Iteration \d+ is finished", l) for l in lines)
    assert "Individual waiting times:" in lines
    assert any(l.startswith("Average waiting time(ms) per worker and "
                            "iteration:") for l in lines)


def test_asgd_sync_contract():
    out = _capture(drivers.asgd_sync, ARGS13)
    _check_contract(out, "ASGDSync")


def test_asaga_thread_contract():
    args = list(ARGS13)
    args[6] = "0.05"
    out = _capture(drivers.asaga_thread, args)
    _check_contract(out, "ASAGA")


def test_asaga_sync_contract():
    args = list(ARGS13)
    args[6] = "0.05"
    out = _capture(drivers.asaga_sync, args)
    _check_contract(out, "ASAGASync")


def test_sgd_mllib_contract():
    out = _capture(drivers.sgd_mllib, ARGS8)
    _check_contract(out, "MLlib SGD")


def test_arg_echo():
    out = _capture(drivers.asgd_thread, ARGS13)
    assert "num columns: 16" in out
    assert "num rows: 200" in out
    assert "num partitions: 2" in out
    assert "step size: 0.5" in out
    assert "taw: 1000000" in out
    assert "batch rate: 0.3" in out
    assert "bucket ratio: 0.5" in out
    assert "printer freq: 10" in out
    assert "coeff: 0" in out
    assert "seed: 42" in out


def test_objective_decreases_in_output():
    out = _capture(drivers.asgd_thread, ARGS13)
    lines = out.splitlines()
    # objective CSV comes after the LAST ********* separator
    # (waiting-time lines share the "<int>,<num>" shape — reference format)
    sep = max(i for i, l in enumerate(lines) if l.startswith("*********"))
    csv = [l for l in lines[sep + 1:] if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    objs = [float(l.split(",")[1]) for l in csv]
    assert len(objs) >= 2
    assert objs[-1] < objs[0]


def test_cli_checkpoint_flag(tmp_path):
    ck = str(tmp_path / "run.ckpt")
    args = ARGS13 + ["--checkpoint-path", ck, "--checkpoint-every", "10"]
    _capture(drivers.asgd_thread, args)
    import os
    assert os.path.exists(ck)
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    state = load_checkpoint(ck)
    assert state["k"] >= 10


def test_cli_resume_from_checkpoint(tmp_path):
    ck = str(tmp_path / "resume.ckpt")
    _capture(drivers.asgd_thread,
             ARGS13 + ["--checkpoint-path", ck, "--checkpoint-every", "10"])
    # resume with a larger iteration budget: the run continues from the
    # checkpointed k rather than restarting
    args = list(ARGS13)
    args[5] = "60"  # numIter
    out = _capture(drivers.asgd_thread, args + ["--resume-from", ck])
    assert "finished" in out


@pytest.mark.timeout(300)
@pytest.mark.parametrize("engine", ["threads", "native"])
def test_cli_torchrun_dist_contract(engine, tmp_path):
    """13-arg drivers under torchrun (2 ranks, gloo): numPart stays the
    logical worker count; rank 0 prints the unchanged stdout contract."""
    import subprocess
    import sys
    args = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node=2", "--master-addr=127.0.0.1",
            "--master-port=29733" if engine == "threads"
            else "--master-port=29734",
            "-m", "asyncframework_amd.cli", "asgd-thread",
            "synthetic", "synthetic", "16", "200", "4", "40", "0.5",
            "1000000", "0.3", "0.5", "10", "0", "42", "--engine", engine]
    out = subprocess.run(args, capture_output=True, text=True, timeout=240,
                         cwd=os.path.dirname(os.path.dirname(
                             os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    lines = out.stdout.splitlines()
    assert lines[0] == "Spark ASGD application started"
    assert any(re.match(r"Iteration \d+ is finished", l) for l in lines)
    assert any(re.match(r"Elapsed time\(ms\): \d+", l) for l in lines)
    assert lines[-1] == "finished"
    sep = max(i for i, l in enumerate(lines) if l.startswith("*********"))
    csv = [l for l in lines[sep + 1:] if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    objs = [float(l.split(",")[1]) for l in csv]
    assert len(objs) >= 2 and objs[-1] < objs[0]


@pytest.mark.timeout(300)
def test_cli_torchrun_dist_checkpoint_resume(tmp_path):
    """Checkpoint + resume through the torchrun CLI path (asaga, threads
    dist engine): the checkpoint carries BOTH ranks' history tables and a
    resumed run restores them."""
    import subprocess
    import sys
    ck = str(tmp_path / "dist.ckpt")

    def cmd(port, extra):
        return [sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", "--nproc-per-node=2",
                "--master-addr=127.0.0.1", f"--master-port={port}",
                "-m", "asyncframework_amd.cli", "asaga-thread",
                "synthetic", "synthetic", "16", "200", "4", "80", "0.05",
                "1000000", "0.3", "0.5", "20", "0", "42"] + extra

    cwd = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(cmd(29741, ["--checkpoint-path", ck,
                                     "--checkpoint-every", "25"]),
                         capture_output=True, text=True, timeout=240,
                         cwd=cwd)
    assert out.returncode == 0, out.stderr[-2000:]
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    state = load_checkpoint(ck)
    assert set(state["alpha"].keys()) == {0, 1, 2, 3}
    assert state["k"] >= 25

    out2 = subprocess.run(cmd(29742, ["--resume-from", ck]),
                          capture_output=True, text=True, timeout=240,
                          cwd=cwd)
    assert out2.returncode == 0, out2.stderr[-2000:]
    assert out2.stdout.splitlines()[-1] == "finished"


def test_cli_main_usage_error():
    import subprocess
    import sys
    cwd = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run([sys.executable, "-m", "asyncframework_amd.cli",
                          "bogus-driver"], capture_output=True, text=True,
                         timeout=120, cwd=cwd)
    assert out.returncode == 2
    assert "usage:" in out.stderr
    out2 = subprocess.run([sys.executable, "-m", "asyncframework_amd.cli"],
                          capture_output=True, text=True, timeout=120,
                          cwd=cwd)
    assert out2.returncode == 2
