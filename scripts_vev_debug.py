import torch, sys
sys.path.insert(0, "/root/repo")
from dmosopt_amd import _hipops
from dmosopt_amd import ops as dops

dev = torch.device("cuda")
torch.manual_seed(5)
K, d, C, M = 100, 30, 45, 22
pool = torch.rand(K, d, device=dev)
i1 = torch.randint(0, K, (C,), device=dev)
i2 = (i1 + 1 + torch.randint(0, K - 1, (C,), device=dev)) % K
im = torch.randint(0, K, (M,), device=dev)
di_c = torch.full((d,), 1.0, device=dev)
di_m = torch.full((d,), 20.0, device=dev)
lo = torch.zeros(d, device=dev)
hi = torch.ones(d, device=dev)
total = 2 * C + M
src_rows = torch.randperm(total, device=dev)
s1, s2 = 1234, 9876
c1, c2 = dops.sbx_from_pool(pool, i1, i2, di_c, lo, hi, seed=s1)
mm = dops.mutation_from_pool(pool, im, di_m, lo, hi, 0.1, seed=s2)
virt = torch.cat([c1, c2, mm], dim=0)
want = virt[src_rows]
inv = torch.argsort(src_rows)
c_idx = torch.empty(2 * C, dtype=torch.long, device=dev)
c_idx[0::2] = inv[:C]
c_idx[1::2] = inv[C:2 * C]
m_idx = inv[2 * C:].contiguous()
got2 = _hipops.variation_events(
    pool.contiguous(), c_idx.contiguous(), m_idx, i1.long().contiguous(),
    i2.long().contiguous(), im.long().contiguous(), di_c, di_m, lo, hi, 0.1,
    s1, s2)
bad = (got2 != want).any(dim=1)
print("mismatch rows:", int(bad.sum()), "of", total)
# are the mismatched rows correct rows in wrong places?
bs = bad.nonzero().flatten()[:10].tolist()
for s in bs[:5]:
    r = int(src_rows[s])
    kind = "c1" if r < C else ("c2" if r < 2 * C else "mut")
    # which virtual row does got2[s] equal, if any?
    eq = (virt == got2[s][None, :]).all(dim=1).nonzero().flatten().tolist()
    print(f"slot {s}: virtual row {r} ({kind}); got2 row equals virtual {eq}")
