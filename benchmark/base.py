"""Benchmark harness base (reference benchmark/benchmark/base.py:232-283
`run_once` timing pattern): each bench builds/loads a dataset, times fit
(barriers + device sync on both sides), optionally times transform and a
quality score, and prints one JSON report line from rank 0.
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from typing import Any, Callable, Dict, Optional

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.parallel.context import get_comm, init_comm


class BenchmarkBase:
    name = "base"

    def add_arguments(self, ap: argparse.ArgumentParser) -> None:
        pass

    def make_data(self, args: argparse.Namespace) -> DataFrame:
        raise NotImplementedError

    def make_estimator(self, args: argparse.Namespace):
        raise NotImplementedError

    def score(self, model, df: DataFrame, args: argparse.Namespace) -> Optional[Dict[str, float]]:
        return None

    def run_fit(self, est, df):
        return est.fit(df)

    def run(self, argv=None) -> Dict[str, Any]:
        ap = argparse.ArgumentParser(prog=f"bench_{self.name}")
        ap.add_argument("--num_rows", type=int, default=100000)
        ap.add_argument("--num_cols", type=int, default=300)
        ap.add_argument("--train_path", default=None, help="parquet input (else synthetic)")
        ap.add_argument("--num_runs", type=int, default=1)
        ap.add_argument("--seed", type=int, default=0)
        ap.add_argument("--no_transform", action="store_true")
        self.add_arguments(ap)
        args = ap.parse_args(argv)

        comm = init_comm()
        if args.train_path:
            df = DataFrame.read_parquet(args.train_path, vector_cols=["features"])
        else:
            df = self.make_data(args)

        def _sync():
            comm.barrier()
            if comm.device.type == "cuda":
                torch.cuda.synchronize(comm.device)

        fit_times = []
        model = None
        for _ in range(args.num_runs):
            est = self.make_estimator(args)
            _sync()
            t0 = time.perf_counter()
            model = self.run_fit(est, df)
            _sync()
            fit_times.append(comm.allreduce_scalar(time.perf_counter() - t0, "max"))

        transform_time = None
        out_df = df
        if not args.no_transform and hasattr(model, "transform"):
            try:
                _sync()
                t0 = time.perf_counter()
                out_df = model.transform(df)
                _sync()
                transform_time = comm.allreduce_scalar(time.perf_counter() - t0, "max")
            except NotImplementedError:
                pass

        report: Dict[str, Any] = {
            "bench": self.name,
            "n_gpus": comm.world_size,
            "num_rows_total": df.count(),
            "num_cols": args.num_cols,
            "fit_sec": float(np.mean(fit_times)),
            "fit_sec_all": fit_times,
            "transform_sec": transform_time,
            "device": comm.device.type,
        }
        scores = self.score(model, out_df, args)
        if scores:
            report.update(scores)
        if comm.rank == 0:
            print(json.dumps(report))
        return report
