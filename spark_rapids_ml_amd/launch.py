"""Multi-GPU launch wrapper (reference `spark_rapids_submit.py` /
`pyspark_rapids.py` analog).

The reference wraps `spark-submit`/`pyspark` so a user script runs on the
cluster with the plugin configured. Here the cluster is N SPMD ranks over
RCCL, so the wrapper execs `torch.distributed.run` with one rank per
visible GPU and routes the script through the no-code-change runner
(`python -m spark_rapids_ml_amd`, which installs the `spark_rapids_ml`
aliases and initializes the communicator):

  srml-amd-launch app.py [args...]            # all visible GPUs
  srml-amd-launch --gpus 4 app.py [args...]   # explicit rank count
  srml-amd-launch --gpus 0 app.py             # single process, CPU (gloo)
"""

from __future__ import annotations

import argparse
import os
import runpy
import sys


def _n_gpus() -> int:
    try:
        import torch

        return torch.cuda.device_count() if torch.cuda.is_available() else 0
    except Exception:
        return 0


def main() -> None:
    ap = argparse.ArgumentParser(
        prog="srml-amd-launch", description=__doc__, add_help=True
    )
    ap.add_argument("--gpus", type=int, default=None, help="ranks to launch (default: all visible GPUs)")
    ap.add_argument("--master-port", type=int, default=29531)
    ap.add_argument("script", help="user script to run")
    ap.add_argument("script_args", nargs=argparse.REMAINDER)
    args = ap.parse_args()

    n = args.gpus if args.gpus is not None else _n_gpus()
    if n <= 1:
        # single rank: run in-process, no rendezvous needed
        sys.argv = [args.script] + args.script_args
        from .install import install_aliases
        from .parallel.context import init_comm, shutdown_comm

        install_aliases()
        init_comm()
        try:
            runpy.run_path(args.script, run_name="__main__")
        finally:
            shutdown_comm()
        return

    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    sys.argv = [
        "torchrun",
        "--nnodes=1",
        f"--nproc-per-node={n}",
        "--master-addr=127.0.0.1",
        f"--master-port={args.master_port}",
        "-m",
        "spark_rapids_ml_amd",
        args.script,
    ] + args.script_args
    from torch.distributed.run import main as torchrun_main

    torchrun_main(sys.argv[1:])


if __name__ == "__main__":
    main()
