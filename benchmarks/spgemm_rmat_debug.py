#!/usr/bin/env python3
# Phase-timed R-MAT SpGEMM probe.
import argparse, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from legate_sparse.gallery import rmat
from legate_sparse import ops, _cext
from legate_sparse.settings import settings

ap = argparse.ArgumentParser()
ap.add_argument("--scale", type=int, default=16)
args = ap.parse_args()

A = rmat(args.scale, 16, seed=1)
torch.cuda.synchronize()
n = A.shape[0]
print(f"scale={args.scale} n={n} nnz={A.nnz}", flush=True)

ext = _cext.require_hip()
st = torch.cuda.current_stream().cuda_stream
dev = A._data.device
ub = torch.empty(n, dtype=torch.int64, device=dev)
t0=time.perf_counter()
ext.spgemm_row_ub(A._indptr.data_ptr(), A._indices.data_ptr(),
                  A._indptr.data_ptr(), ub.data_ptr(), n, 0, st)
torch.cuda.synchronize()
print(f"ub kernel {time.perf_counter()-t0:.3f}s; flops={int(ub.sum())/1e9:.2f}G "
      f"max_ub={int(ub.max())}", flush=True)
cuts = ext.spgemm_lds_bins
prev=-1
for c in list(cuts)+[1<<62]:
    m = ((ub > prev) & (ub <= c))
    print(f"  bin <= {c}: rows={int(m.sum())} work={int(ub[m].sum())/1e9:.3f}G", flush=True)
    prev=c

t0=time.perf_counter()
C_indptr, C_idx, C_val = ops.spgemm_local(A._indptr, A._indices, A._data,
                                          A._indptr, A._indices, A._data, n)
torch.cuda.synchronize()
print(f"spgemm total {time.perf_counter()-t0:.2f}s nnzC={C_idx.numel()}", flush=True)
