"""Byte-exact stdout contract helpers (reference log lines)."""

import io
from contextlib import redirect_stdout

from asyncframework_amd.utils import logfmt


def _cap(fn, *a):
    buf = io.StringIO()
    with redirect_stdout(buf):
        fn(*a)
    return buf.getvalue()


def test_iteration_line():
    assert _cap(logfmt.iteration_finished, 200) == "Iteration 200 is finished\n"


def test_elapsed_line():
    assert _cap(logfmt.elapsed, 12345) == "Elapsed time(ms): 12345\n"


def test_waiting_block_integer_division():
    out = _cap(logfmt.waiting_times, {0: 10, 1: 25}, 3)
    lines = out.splitlines()
    assert lines[0] == "*" * 33
    assert lines[1] == "Individual waiting times:"
    assert lines[2] == "0,10" and lines[3] == "1,25"
    # (10+25) // (2*3) = 5 — Scala Long integer division semantics
    assert lines[4] == "Average waiting time(ms) per worker and iteration:5"


def test_objective_lines_and_finished():
    out = _cap(logfmt.objective_lines, [(0, 1.5), (120, 0.25)])
    lines = out.splitlines()
    assert lines[0] == "*" * 33
    assert lines[1] == "0,1.5" and lines[2] == "120,0.25"
    assert lines[-1] == "finished"
