import torch, sys
sys.path.insert(0, "/root/repo")
from dmosopt_amd import _hipops

dev = torch.device("cuda")
torch.manual_seed(1)
K, d, C, M = 6, 4, 2, 1
pool = torch.rand(K, d, device=dev)
i1 = torch.tensor([0, 2], dtype=torch.long, device=dev)
i2 = torch.tensor([1, 3], dtype=torch.long, device=dev)
im = torch.tensor([4], dtype=torch.long, device=dev)
di_c = torch.full((d,), 1.0, device=dev)
di_m = torch.full((d,), 20.0, device=dev)
lo = torch.zeros(d, device=dev)
hi = torch.ones(d, device=dev)
s1, s2 = 1234, 9876
c1, c2 = _hipops.sbx_batch(pool, i1.int().contiguous(), i2.int().contiguous(), di_c, lo, hi, s1)
mm = _hipops.mutation_batch(pool, im.int().contiguous(), di_c * 0 + 20.0, lo, hi, 0.1, s2)
# identity placement: slots [c1_0, c1_1? no: ci[2k]=slot of c1_k]
ci = torch.tensor([0, 1, 2, 3], dtype=torch.long, device=dev)  # pair0 -> slots 0,1; pair1 -> 2,3
mi = torch.tensor([4], dtype=torch.long, device=dev)
got = _hipops.variation_events(pool.contiguous(), ci, mi, i1.contiguous(), i2.contiguous(), im.contiguous(), di_c, di_m, lo, hi, 0.1, s1, s2)
print("c1 oracle:", c1.cpu().numpy())
print("c2 oracle:", c2.cpu().numpy())
print("mm oracle:", mm.cpu().numpy())
print("events   :", got.cpu().numpy())
print("slot0==c1_0:", torch.equal(got[0], c1[0]), " slot1==c2_0:", torch.equal(got[1], c2[0]))
print("slot2==c1_1:", torch.equal(got[2], c1[1]), " slot3==c2_1:", torch.equal(got[3], c2[1]))
print("slot4==mm_0:", torch.equal(got[4], mm[0]))
