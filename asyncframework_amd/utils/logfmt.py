"""The stdout contract — byte-compatible with the reference drivers.

The reference's scientific output channel is stdout:
``Iteration <k> is finished`` (SparkASGDThread.scala:196), ``Elapsed
time(ms): <t>`` (:349), the waiting-time block (:355-362), and the final
``<time_ms>,<objective>`` CSV lines (:400-404) ending in ``finished`` (:410).
These helpers centralize the format so every driver emits identical text.
"""

from __future__ import annotations

from typing import Dict, Iterable, Sequence, Tuple


def print_header(app_name: str, arg_names: Sequence[str],
                 arg_values: Sequence[object]) -> None:
    """Mirrors the argument echo of the reference drivers
    (SparkASGDThread.scala:29-65)."""
    print(f"Spark {app_name} application started")
    print("Input arguments:")
    print("Input format: [" + "] [".join(arg_names) + "]")
    for name, val in zip(arg_names, arg_values):
        print(f"{name}: {val}")


def iteration_finished(k: int) -> None:
    print(f"Iteration {k} is finished")


def elapsed(ms: int) -> None:
    print(f"Elapsed time(ms): {ms}")


def waiting_times(table: Dict[int, int], k: int) -> None:
    """Reference SparkASGDThread.scala:352-362 — note the integer division."""
    print("*********************************")
    print("Individual waiting times:")
    total = 0
    count = 0
    for wid, ms in table.items():
        print(f"{wid},{ms}")
        total += int(ms)
        count += 1
    denom = count * max(k, 1)
    avg = total // denom if denom > 0 else 0
    print(f"Average waiting time(ms) per worker and iteration:{avg}")


def objective_lines(pairs: Iterable[Tuple[int, float]]) -> None:
    """Final ``time_ms,objective`` CSV (reference :400-404) + ``finished``."""
    print("*********************************")
    for t_ms, obj in pairs:
        print(f"{t_ms},{obj}")
    print("finished")
