"""ResNet images/sec benchmark, runnable inside the worker image as
``python -m mpi_operator_amd.benchmarks.resnet`` (the tf_cnn_benchmarks
role, reference README.md:127-130). Delegates to the repo-root bench.py
main when run from a source checkout; inlined here for wheel installs."""
from __future__ import annotations

import os
import sys


def main():
    root = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    sys.path.insert(0, root)
    bench = os.path.join(root, "bench.py")
    if os.path.exists(bench):
        import runpy

        sys.argv[0] = bench
        runpy.run_path(bench, run_name="__main__")
    else:
        raise SystemExit("bench.py not found next to the package — "
                         "run from a source checkout or the amdrun image")


if __name__ == "__main__":
    main()
