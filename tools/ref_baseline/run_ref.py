"""Run the UNMODIFIED reference (/root/reference) through the dependency
shim, for baseline measurement (BASELINE.md protocol).

Usage:
    python tools/ref_baseline/run_ref.py train --env DubinsCar -n 16 \
        --steps 2000 --batch-size 512 [--cpu] [--log-path results/refruns]
    python tools/ref_baseline/run_ref.py test --path <log_dir> --epi 10 ...

The reference scripts run exactly as published (runpy, __main__); only the
unavailable third-party imports are shimmed (see pygshim.py).
"""
import os
import runpy
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
REF = "/root/reference"


def main():
    sys.path.insert(0, HERE)
    import pygshim
    pygshim.install()
    sys.path.insert(0, REF)
    script = sys.argv[1]
    assert script in ("train", "test", "plot_cbf"), script
    sys.argv = [script + ".py"] + sys.argv[2:]
    runpy.run_path(os.path.join(REF, script + ".py"), run_name="__main__")


if __name__ == "__main__":
    main()
