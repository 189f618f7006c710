#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""The README demo (BASELINE config 1): 5x5 tridiagonal SpMV + SpGEMM,
checked for EXACT agreement with scipy on the CPU path."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import scipy.sparse as sp

import legate_sparse as lsp

n = 5
diagonals = [np.full(n, 2.0), -np.ones(n - 1), -np.ones(n - 1)]
A = lsp.diags(diagonals, [0, -1, 1], shape=(n, n), format="csr")
S = sp.diags(diagonals, [0, -1, 1], shape=(n, n), format="csr")

x = np.arange(1.0, n + 1)
y = A @ x
y_ref = S @ x
assert np.array_equal(np.asarray(y.cpu() if hasattr(y, "cpu") else y),
                      y_ref), (y, y_ref)
print("SpMV  :", np.asarray(y.cpu() if hasattr(y, "cpu") else y))

C = A @ A
C_ref = (S @ S).toarray()
Cd = C.todense()
Cd = Cd.cpu().numpy() if hasattr(Cd, "cpu") else np.asarray(Cd)
assert np.array_equal(Cd, C_ref), (Cd, C_ref)
print("SpGEMM:")
print(Cd)
print("PASS: exact match vs scipy")
