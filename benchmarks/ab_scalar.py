import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
import legate_sparse as lsp
from legate_sparse.gallery import poisson_2d
sys.path.insert(0, "/root/repo/examples")
from common import banded_matrix

flag = os.environ.get("LS_SPGEMM_SCALAR", "0")
A = poisson_2d(2048, 2048)
for _ in range(3): C = A @ A
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(20): C = A @ A
torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / 20
print(f"scalar={flag} poisson2048: {dt*1e3:.3f} ms  {C.nnz/dt/1e9:.2f} Gnnz/s")
B = banded_matrix(2**22, 5, seed=1)
for _ in range(3): C2 = B @ B
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(20): C2 = B @ B
torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / 20
print(f"scalar={flag} banded4M: {dt*1e3:.3f} ms  {C2.nnz/dt/1e9:.2f} Gnnz/s")
