"""bench.py argument resolution: preset values, CLI overrides, and the
BASELINE config-1 objective (logistic) — the driver contract's knobs."""

import importlib.util
import os
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
spec = importlib.util.spec_from_file_location(
    "bench_mod", os.path.join(ROOT, "bench.py"))
bench = importlib.util.module_from_spec(spec)
spec.loader.exec_module(bench)


def _parse(argv):
    old = sys.argv
    sys.argv = ["bench.py"] + argv
    try:
        return bench.parse_args()
    finally:
        sys.argv = old


def test_default_model_is_flagship():
    a = _parse([])
    assert a.model == "asgd-mnist8m"
    assert a.rows == 8_100_000 and a.cols == 784
    assert a.dtype == "bf16" and a.algo == "asgd"
    assert a.engine == "native" and a.preset_workers == 32
    assert a.objective == "lsq"


def test_config1_is_logistic_sync_cpu():
    a = _parse(["--model", "sync-tiny-cpu"])
    assert a.objective == "logistic"  # BASELINE config 1 naming
    assert a.sync and a.device == "cpu" and a.preset_workers == 2


def test_cli_overrides_beat_presets():
    a = _parse(["--model", "asgd-mnist8m", "--rows", "1000",
                "--cols", "32", "--dtype", "fp32", "--workers", "4",
                "--objective", "logistic", "--engine", "threads"])
    assert (a.rows, a.cols, a.dtype, a.preset_workers) == (1000, 32,
                                                           "fp32", 4)
    assert a.objective == "logistic" and a.engine == "threads"


def test_all_five_baseline_models_present():
    assert set(bench.MODELS) == {"sync-tiny-cpu", "asgd-mnist8m",
                                 "asaga-rcv1", "asgd-epsilon-delay",
                                 "asaga-mnist8m-hostspill"}
    assert bench.MODELS["asaga-rcv1"]["sparse"]
    assert bench.MODELS["asgd-epsilon-delay"]["delay_coeff"] == 1.0
    assert bench.MODELS["asaga-mnist8m-hostspill"]["history"] == "host"


def test_make_cfg_iteration_budget():
    a = _parse(["--steps", "100", "--warmup", "10", "--model",
                "sync-tiny-cpu"])
    cfg = bench.make_cfg(a, 2, "cpu")
    assert cfg.num_iterations == 111
    assert cfg.snapshot_weights is False


def test_warmup_and_steps_clamped():
    """marks record at exact post-increment update counts, so k=0 is
    unreachable: bench must clamp --warmup 0 (and degenerate --steps)
    instead of failing with 'marks missing'."""
    a = _parse(["--warmup", "0", "--steps", "1"])
    assert a.warmup == 1
    assert a.steps == 2
    a = _parse(["--warmup", "3000", "--steps", "30000"])
    assert a.warmup == 3000 and a.steps == 30000
