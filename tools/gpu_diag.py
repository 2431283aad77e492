"""
Kernel-level diagnostic for the MFMA layouts (run on the GPU box).

Single sgd_step with squared loss, lr=1, no mask, W0=0:
  GT should equal  (X @ 0 - t)^T = -t broadcast  -> checks K1 epilogue
  (then with W warm) GT = (X@W - t)^T            -> checks K1 GEMM layout
  W after = -1 * X^T @ G / m                     -> checks K2/K3

Prints max-abs diffs and small corners so a transposed operand is
immediately visible.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from skdist_amd.ops import require_hip


def main():
    ext = require_hip()
    dev = torch.device("cuda")
    torch.manual_seed(0)
    n, fa, ncols = 256, 64, 128  # 2 row tiles, 1 col tile, fa%32==0
    bs = 256
    gts = 256

    X = torch.randn(n, fa, device=dev).to(torch.bfloat16)
    # asymmetric, nonzero W so the forward GEMM layout is exercised
    W = torch.randn(fa, ncols, device=dev) * 0.1
    WbfT = W.t().contiguous().to(torch.bfloat16)
    y = torch.randn(n, device=dev)
    fold = torch.full((n,), 3, dtype=torch.int32, device=dev)
    col_class = torch.full((ncols,), -1, dtype=torch.int32, device=dev)
    col_fold = torch.full((ncols,), -2, dtype=torch.int32, device=dev)
    col_lr = torch.ones(ncols, device=dev)
    col_l2 = torch.zeros(ncols, device=dev)
    GT = torch.empty(ncols, gts, dtype=torch.bfloat16, device=dev)
    V = torch.empty(0, device=dev)
    splitk = 4
    partial = torch.empty(splitk, fa, ncols, device=dev)

    fa_store = (fa + 127) // 128 * 128
    XsT = torch.zeros(fa_store, n, dtype=torch.bfloat16, device=dev)
    XsT[:fa] = X.t()
    W0 = W.clone()
    col_class2 = torch.full((ncols,), -1, dtype=torch.int32, device=dev)
    ext.sgd_step(X, XsT, GT, W, V, WbfT, partial, y, fold,
                 col_class, col_fold, col_class2, col_lr, col_l2,
                 0, n, 2, 1.0, 0.0, fa - 1)
    torch.cuda.synchronize()

    Xf = X.float()
    Zref = Xf @ WbfT.float().t()  # note: WbfT was refreshed by K3 -> wrong!
    # use the ORIGINAL bf16 weights for the forward reference:
    Zref = Xf @ W0.t().contiguous().to(torch.bfloat16).float().t()
    Gref = (Zref - y.unsqueeze(1)).to(torch.bfloat16).float()
    gt = GT.float()[:, :n]
    dG = (gt - Gref.t()).abs().max().item()
    print(f"GT  max|diff| = {dG:.6f}")
    print("GT[0:3,0:3]    =", gt[:3, :3].cpu().numpy().round(3).tolist())
    print("Gref.T[0:3,0:3]=",
          Gref.t()[:3, :3].cpu().numpy().round(3).tolist())

    grad_ref = Xf.t() @ Gref / n
    W_expected = W0 - grad_ref
    dW = (W - W_expected).abs().max().item()
    print(f"W   max|diff| = {dW:.6f}")
    print("W[0:3,0:3]     =", W[:3, :3].cpu().numpy().round(4).tolist())
    print("Wexp[0:3,0:3]  =",
          W_expected[:3, :3].cpu().numpy().round(4).tolist())
    print("partial sum check:",
          (partial.sum(0) / n - grad_ref).abs().max().item())
    ok = dG < 0.02 and dW < 0.02
    print("DIAG", "PASS" if ok else "FAIL")


if __name__ == "__main__":
    main()
