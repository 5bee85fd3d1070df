"""Exercise every extension host wrapper once on small GPU shapes (run
under ASan via tools/sanitize_check.sh)."""
import importlib.util
import os
import sys

import torch

spec = importlib.util.spec_from_file_location(
    "_C_asan", os.path.join("gpurun_out", "asan_build", "_C_asan.so"))
ext = importlib.util.module_from_spec(spec)
spec.loader.exec_module(ext)

dev = "cuda"
t = torch.randn(64, 8, device=dev)
rows = torch.randint(0, 64, (200,), device=dev)
out = ext.gather(t, rows)
ext.apply_add(t, rows, torch.randn(200, 8, device=dev), False)
ext.apply_add(t, torch.arange(64, device=dev), torch.randn(64, 8, device=dev), True)
acc = torch.zeros(64, 8, device=dev)
sr, _ = torch.sort(rows)
ext.apply_adagrad_sorted(t, acc, sr.contiguous(), torch.randn(200, 8, device=dev), 0.1, 1e-8)
ext.hash_init(torch.arange(10, device=dev), 4, -1.0, 1.0, 3)
ext.hash_init_rows(torch.empty(10, 4, device=dev), 0, 1, -1.0, 1.0, 3)
users = torch.randint(0, 50, (500,), dtype=torch.int32, device=dev)
slots = torch.randint(0, 30, (500,), dtype=torch.int32, device=dev)
ut = torch.randn(50, 8, device=dev)
iv = torch.randn(30, 8, device=dev)
r = torch.randn(500, device=dev)
for tag in (0, 77, 77):  # no-reuse, first visit, revisit
    ext.mf_sgd_step(ut, users, iv, slots, r, 0.05, 0.01, tag, False)
ext.mf_sgd_step(ut, users, iv, slots, r, 0.05, 0.01, 0, True)
ext.mf_sgd_step(torch.randn(50, 6, device=dev), users,
                torch.randn(30, 6, device=dev), slots, r, 0.05, 0.0, 0,
                False)  # envelope (dim % 4 != 0)
vecs = torch.randn(40, 8, device=dev)
c = torch.randint(0, 20, (300,), dtype=torch.int64, device=dev)
x = torch.randint(20, 40, (300,), dtype=torch.int64, device=dev)
y = (torch.rand(300, device=dev) < 0.3).float()
ext.sgns_step(vecs, c, x, y, 0.05)
ext.sgns_expand(c, c, 3, 20, 123)
ids = torch.randint(0, 10**8, (5000,), device=dev)
uids, sl, keys, smap = ext.hash_coalesce(ids)
ext.hash_lookup(keys, smap, uids[:100])
B, nnz = 40, 8
indptr = torch.arange(0, B * nnz + 1, nnz, dtype=torch.int64, device=dev)
fs = torch.randint(0, 60, (B * nnz,), dtype=torch.int64, device=dev)
xv = torch.randn(B * nnz, device=dev)
lab = (torch.rand(B, device=dev) < 0.5).float()
w = torch.randn(60, device=dev)
ext.pa_step(indptr, fs, xv, lab * 2 - 1, w, "PA-I", 1.0)
ext.lr_step(indptr, fs, xv, lab, w)
wm = torch.randn(60, 7, device=dev)
labm = torch.randint(-1, 7, (B,), dtype=torch.int64, device=dev)
ext.pa_multiclass_step(indptr, fs, xv, labm, wm, "PA-II", 1.0)
torch.cuda.synchronize()
print("all wrappers exercised")
