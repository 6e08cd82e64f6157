"""Straggler-injection subsystem tests (reference SparkASGDThread.scala:
124-141 selection, 177-186/247-252 calibration, 287-312 injection)."""

from asyncframework_amd.engine.delay import DelayInjector


def test_straggler_selection_matches_reference_rule():
    # P=32: length=8, lengthNormal=round(0.8*8)=6, lengthLongTail=2
    inj = DelayInjector(32, coeff=-1.0)
    assert inj.straggler_longtail == {0, 4}
    assert inj.straggler_normal == {8, 12, 16, 20, 24, 28}


def test_no_delay_before_calibration():
    inj = DelayInjector(4, coeff=-1.0)
    assert inj.delay_ms(0, 5) == 0.0


def test_calibration_and_cloud_delay():
    inj = DelayInjector(4, coeff=-1.0, calib_window=10)
    for k in range(10):
        inj.record_task(k, 100.0)
    inj.maybe_activate(11)
    assert inj.flag
    assert inj.avg_delay_ms == 100.0
    # P=4: length=1, lengthNormal=round(0.8)=1, longtail=0 -> worker 0 normal
    assert inj.straggler_normal == {0}
    d = inj.delay_ms(0, 3)
    assert 150.0 <= d <= 250.0  # (u+1.5)*avg, u in [0,1)
    assert inj.delay_ms(1, 3) == 0.0


def test_deterministic_delay_same_round():
    inj = DelayInjector(4, coeff=-1.0, calib_window=0)
    inj._cul_time, inj._cul_count = 100.0, 1
    inj.maybe_activate(1)
    assert inj.delay_ms(0, 7) == inj.delay_ms(0, 7)
    assert inj.delay_ms(0, 7) != inj.delay_ms(0, 8)  # varies per round


def test_fixed_coeff_slows_worker0_only():
    inj = DelayInjector(4, coeff=2.0, calib_window=0)
    inj._cul_time, inj._cul_count = 50.0, 1
    inj.maybe_activate(1)
    assert inj.delay_ms(0, 1) == 100.0
    for wid in (1, 2, 3):
        assert inj.delay_ms(wid, 1) == 0.0


def test_coeff_zero_disables():
    inj = DelayInjector(4, coeff=0.0, calib_window=0)
    inj.maybe_activate(1)
    assert inj.delay_ms(0, 1) == 0.0


def test_activation_without_samples_keeps_zero_delay():
    inj = DelayInjector(4, coeff=-1.0, calib_window=0)
    inj.maybe_activate(1)  # no recorded tasks
    assert inj.flag
    assert inj.avg_delay_ms == 0.0
    assert inj.delay_ms(0, 1) == 0.0
