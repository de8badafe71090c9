"""Tiny kernel microbench for PMC counter collection (rocprofv3 --pmc
serializes dispatches — run ~10 launches, not a whole bench)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from volcano_amd.ops import hip

N, R, W = 10000, 8, 1
dev = "cuda"
alloc = torch.full((R, N), 32000.0, device=dev)
used = torch.zeros((R, N), device=dev)
extra = torch.zeros((R, N), device=dev)
ready = torch.ones(N, dtype=torch.uint8, device=dev)
taints = torch.zeros(N, dtype=torch.int64, device=dev)
planes = torch.zeros((W, N), dtype=torch.int64, device=dev)
req = torch.tensor([1000.0] * R, device=dev)
require = torch.zeros(W, dtype=torch.int64, device=dev)
forbid = torch.zeros(W, dtype=torch.int64, device=dev)
dim_w = torch.ones(R, device=dev)
score = torch.empty(N, device=dev)
cap = torch.empty(N, dtype=torch.int32, device=dev)
qa = torch.zeros(R, device=dev)
ql = torch.full((R,), 1e18, device=dev)
K = N
ln = torch.zeros(K, dtype=torch.int32, device=dev)
lc = torch.zeros(K, dtype=torch.int32, device=dev)
ll = torch.zeros(1, dtype=torch.int32, device=dev)
placed = torch.zeros(1, dtype=torch.int32, device=dev)
jp = torch.zeros(1, dtype=torch.int32, device=dev)
scratch = torch.empty(4 * N, dtype=torch.int32, device=dev)

for i in range(5):
    hip.score_cap(alloc, used, extra, ready, taints, planes, req, -1,
                  require, forbid, 1.0, 1.0, 1.0, dim_w, None, score, cap)
for i in range(3):
    used.zero_()
    qa.zero_()
    hip.score_cap(alloc, used, extra, ready, taints, planes, req, -1,
                  require, forbid, 1.0, 1.0, 1.0, dim_w, None, score, cap)
    hip.select_commit(score, cap, req, 100000, used, qa, ql, ln, lc, ll,
                      placed, jp, -1, sort_scratch=scratch)
torch.cuda.synchronize()
print("microbench done; placed:", int(placed.item()))
