import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from flink_parameter_server_amd import ops
from flink_parameter_server_amd.ops import reference as R

torch.manual_seed(11)
n, U, M, dim = 5000, 300, 200, 32
users = torch.randint(0, U, (n,), dtype=torch.int32)
slots = torch.randint(0, M, (n,), dtype=torch.int32)
users_g, slots_g = users.cuda(), slots.cuda()
ut_tag = torch.randn(U, dim).cuda()
ut_notag = ut_tag.clone()
for visit in range(3):
    r = torch.randn(n); iv = torch.randn(M, dim)
    ut_ref = ut_tag.cpu().clone()
    d_ref = R.mf_sgd_step(ut_ref, users.long(), iv, slots.long(), r, 0.05, 0.01)
    d_tag = ops.mf_sgd_step(ut_tag, users_g, iv.cuda(), slots_g, r.cuda(),
                            0.05, 0.01, reuse_tag=987654321)
    d_nt = ops.mf_sgd_step(ut_notag, users_g, iv.cuda(), slots_g, r.cuda(),
                           0.05, 0.01, reuse_tag=0)
    torch.cuda.synchronize()
    print(f"visit {visit}: "
          f"d_tag-ref {(d_tag.cpu()-d_ref).abs().max():.2e}  "
          f"ut_tag-ref {(ut_tag.cpu()-ut_ref).abs().max():.2e}  "
          f"d_nt-ref {(d_nt.cpu()-d_ref).abs().max():.2e}  "
          f"ut_nt-ref {(ut_notag.cpu()-ut_ref).abs().max():.2e}")
