import os, sys, torch
sys.path.insert(0, "/root/repo")
from flink_parameter_server_amd import ops
from flink_parameter_server_amd.ops import reference as R
DEV = "cuda"
torch.manual_seed(11)
n, U, M, dim = 5000, 300, 200, 32
users = torch.randint(0, U, (n,), dtype=torch.int64)
slots = torch.randint(0, M, (n,), dtype=torch.int64)
r = torch.randn(n); iv = torch.randn(M, dim)
ut0 = torch.randn(U, dim)
ut_cpu = ut0.clone()
d_cpu = R.mf_sgd_step(ut_cpu, users, iv, slots, r, 0.05, 0.01)
ut_gpu = ut0.to(DEV)
d_gpu = ops.mf_sgd_step(ut_gpu, users.to(DEV), iv.to(DEV),
                        slots.to(DEV), r.to(DEV), 0.05, 0.01)
torch.cuda.synchronize()
du = (ut_cpu - ut_gpu.cpu()).abs().max(dim=1).values
bad = (du > 1e-3).nonzero().reshape(-1)
cnt = torch.bincount(users, minlength=U)
print("seg_chunk", os.environ.get("FPS_SEG_CHUNK", "64"),
      "bad users", bad.numel(), "of", U, "maxdiff", float(du.max()))
if bad.numel():
    print("bad user dup-counts:", cnt[bad][:20].tolist())
    print("good user dup-counts:", cnt[du <= 1e-3][:20].tolist())
    b = int(bad[0])
    print("user", b, "cnt", int(cnt[b]))
    print("cpu ", ut_cpu[b][:6].tolist())
    print("gpu ", ut_gpu.cpu()[b][:6].tolist())
    print("init", ut0[b][:6].tolist())
