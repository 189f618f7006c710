#!/usr/bin/env python3
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples"))
import torch
from common import d2_mat_dirichlet_2d
from gmg import GMG

k = int(sys.argv[1]) if len(sys.argv) > 1 else 11
nx = 2 ** k - 1
A = d2_mat_dirichlet_2d(nx, nx)
for mode in ("0", "1"):
    os.environ["LS_GMG_GRAPH"] = mode
    M = GMG(A, k)
    r = torch.rand(M.levels[0][1].numel(), dtype=torch.float64,
                   device="cuda")
    z = M.matvec(r)  # includes capture in graph mode
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30):
        z = M.matvec(r, out=z)
    torch.cuda.synchronize()
    per = (time.perf_counter() - t0) / 30 * 1e3
    print(f"k={k} graph={mode}: {per:.3f} ms per V-cycle", flush=True)
