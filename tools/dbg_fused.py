import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from gymfx_amd.models.lstm import ActorCriticLSTM
from gymfx_amd.ops import api, native
H, M, D = 64, 528, 32
m = ActorCriticLSTM(D, 3, H, device=torch.device("cuda"), seed=9)
g = torch.Generator(device="cuda").manual_seed(0)
h = torch.randn(M, H, generator=g, device="cuda").to(torch.bfloat16)
c = torch.randn(M, H, generator=g, device="cuda")
gates = torch.randn(M, 4*H, generator=g, device="cuda").to(torch.bfloat16)
done = torch.rand(M, generator=g, device="cuda") < 0.3
hm_f = torch.empty(M, H, dtype=torch.bfloat16, device="cuda"); cm_f = torch.empty(M, H, device="cuda")
hm_u = torch.empty_like(hm_f); cm_u = torch.empty_like(cm_f)
gh_u = torch.empty(M, 4*H, dtype=torch.bfloat16, device="cuda")
c_u = torch.empty(M, H, device="cuda"); h_u = torch.empty(M, H, dtype=torch.bfloat16, device="cuda")
api.gemm(h, m.wt("Wh"), None, gh_u, act=1, trans_b=True)
api.lstm_cell_fwd(gates, gh_u, c, c_u, h_u, done, hm_u, cm_u)
gh_f = torch.empty_like(gh_u); c_f = torch.empty_like(c_u); h_f = torch.empty_like(h_u)
ok = native.require().lstm_gemm_cell_fwd(h, m.wt("Wh"), gates, gh_f, c, c_f, h_f, done, hm_f, cm_f)
torch.cuda.synchronize()
print("ok", ok, "gh eq", torch.equal(gh_f, gh_u))
d = (c_f != c_u)
print("c mismatches", int(d.sum()), "of", c_u.numel())
idx = d.nonzero()[:10]
print("first idx", idx.tolist())
if len(idx):
    r, u = idx[0].tolist()
    print("row", r, "unit", u, "done", bool(done[r]), "c_f", float(c_f[r,u]), "c_u", float(c_u[r,u]))
rows = d.any(dim=1).nonzero().flatten()
print("rows affected", rows[:20].tolist(), "count", len(rows))
print("done rows among affected:", done[rows].float().mean().item() if len(rows) else None)
