#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""Test driver (the reference drives legate.tester from a root test.py;
here the same role is a thin pytest wrapper with a multi-process axis).

    python test.py                 # CPU suite
    python test.py --gpu           # GPU-marked suite (run on an MI355X)
    python test.py --ranks 2       # distributed suite at world size 2
"""
import argparse
import os
import subprocess
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpu", action="store_true",
                    help="run the GPU-marked tests")
    ap.add_argument("--ranks", type=int, default=0,
                    help="also run the distributed worker at this world "
                         "size (gloo on CPU, RCCL on GPUs)")
    ap.add_argument("pytest_args", nargs="*", default=[])
    args = ap.parse_args()
    repo = os.path.dirname(os.path.abspath(__file__))
    rc = 0
    marker = "gpu" if args.gpu else "not gpu"
    rc |= subprocess.call([sys.executable, "-m", "pytest", "tests", "-q",
                           "-m", marker, *args.pytest_args], cwd=repo)
    if args.ranks > 1:
        rc |= subprocess.call(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             f"--nproc-per-node={args.ranks}", "--master-addr", "127.0.0.1",
             "--master-port", "29779",
             os.path.join(repo, "tests", "dist_worker.py")], cwd=repo)
    sys.exit(rc)


if __name__ == "__main__":
    main()
