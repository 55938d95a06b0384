"""No-code-change runner (reference __main__.py runpy wrapper):

  python -m spark_rapids_ml_amd app.py [args...]
  python -m torch.distributed.run --nproc-per-node 8 --master-addr 127.0.0.1 \\
      -m spark_rapids_ml_amd app.py [args...]

Installs the `spark_rapids_ml` compatibility aliases, initializes the SPMD
communicator, then runs the target script as __main__.
"""

from __future__ import annotations

import runpy
import sys

from .install import install_aliases
from .parallel.context import init_comm, shutdown_comm


def main() -> None:
    if len(sys.argv) < 2:
        print("usage: python -m spark_rapids_ml_amd <script.py> [args...]", file=sys.stderr)
        sys.exit(2)
    install_aliases()
    init_comm()
    script = sys.argv[1]
    sys.argv = sys.argv[1:]
    try:
        runpy.run_path(script, run_name="__main__")
    finally:
        shutdown_comm()


if __name__ == "__main__":
    main()
