"""Benchmark dispatcher (reference benchmark_runner.py):

  python benchmark_runner.py kmeans --num_rows 1000000 --num_cols 3000 --k 1000
  python -m torch.distributed.run --nproc-per-node 8 --master-addr 127.0.0.1 \\
      benchmark_runner.py logistic_regression --num_rows 1000000 ...
"""

import sys
import os

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from benchmark.benches import BENCHMARKS


def main() -> None:
    if len(sys.argv) < 2 or sys.argv[1] not in BENCHMARKS:
        print(f"usage: benchmark_runner.py <{'|'.join(sorted(BENCHMARKS))}> [args]")
        sys.exit(2)
    bench = BENCHMARKS[sys.argv[1]]
    bench.run(sys.argv[2:])


if __name__ == "__main__":
    main()
