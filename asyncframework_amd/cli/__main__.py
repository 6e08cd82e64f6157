"""``python -m asyncframework_amd.cli <driver> <13 args...>`` — the
spark-submit-equivalent entry point (reference bin/spark-submit --class
SparkASGDThread ...)."""

import sys

from . import drivers

DRIVERS = {
    "asgd-thread": drivers.asgd_thread,
    "asgd-sync": drivers.asgd_sync,
    "asaga-thread": drivers.asaga_thread,
    "asaga-sync": drivers.asaga_sync,
    "sgd-mllib": drivers.sgd_mllib,
}


def main() -> None:
    if len(sys.argv) < 2 or sys.argv[1] not in DRIVERS:
        print("usage: python -m asyncframework_amd.cli "
              f"{{{','.join(DRIVERS)}}} <args...>", file=sys.stderr)
        sys.exit(2)
    DRIVERS[sys.argv[1]](sys.argv[2:])


if __name__ == "__main__":
    main()
