#!/usr/bin/env python3
# A/B eager vs hipGraph CG.
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import legate_sparse as lsp
from legate_sparse.gallery import poisson_2d


def bench(nx, iters):
    A = poisson_2d(nx, nx)
    b = torch.ones(A.shape[0], dtype=torch.float64, device="cuda")
    res = {}
    for mode in ("0", "1"):
        os.environ["LS_CG_GRAPH"] = mode
        lsp.linalg.cg(A, b, rtol=0.0, maxiter=30, conv_test_iters=0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        lsp.linalg.cg(A, b, rtol=0.0, maxiter=iters, conv_test_iters=0)
        torch.cuda.synchronize()
        res[mode] = (time.perf_counter() - t0) / iters * 1e3
    print("nx=%d: eager %.4f ms/iter, graph %.4f ms/iter" %
          (nx, res["0"], res["1"]), flush=True)


bench(512, 2000)
bench(2048, 500)
bench(4096, 300)
