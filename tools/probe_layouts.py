"""Decode the 32x32x16 MFMA layouts from raw dumps (tr16/permlane32 were
confirmed by the first probe run; see profiles/layout_probes.txt)."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

# Probe 1: C layout + A row map. A[m][k] = m + k/32 (bf16-exact),
# B one-hot row 0. If A rows map l&31 and B rows map l&31:
# D[m][0] = 16*m + 3.75, D[m][n>0] = 0.
a = torch.zeros(64, 8)
for l in range(64):
    m = l & 31
    for j in range(8):
        k = (l >> 5) * 8 + j
        a[l, j] = m + k / 32.0
b = torch.zeros(64, 8)
for l in range(64):
    if (l & 31) == 0:
        for j in range(8):
            b[l, j] = 1.0
d = hip._debug_mfma32_raw(a.flatten(), b.flatten()).cpu()
expect = {round(16 * m + 3.75, 2): m for m in range(32)}
got = {}
for l in range(64):
    for i in range(16):
        v = round(d[l, i].item(), 2)
        if v in expect:
            got[(l, i)] = expect[v]
hyp = all(got.get((l, i)) == ((l >> 5) * 4 + (i & 3) + 8 * (i >> 2))
          for l in range(64) for i in range(16) if (l & 31) == 0)
nonzero_off = [(l, i, d[l, i].item()) for l in range(64) for i in range(16)
               if abs(d[l, i].item()) > 0.01 and (l & 31) != 0]
print("probe1 decoded:", len(got), "of 512 expected 16*32... lanes n==0 only")
print("  C hypothesis m=(l>>5)*4+(i&3)+8*(i>>2), n=l&31:", hyp)
print("  nonzero at n!=0:", len(nonzero_off))
if not hyp:
    for l in (0, 32):
        print("  lane", l, [got.get((l, i)) for i in range(16)])

# Probe 2: k map. A one-hot at k=5 under the assumed map; B[n][k] = n + k/32.
# D[m][n] = B[n][k'] = n + k'/32 where k' is where the one-hot REALLY landed.
a2 = torch.zeros(64, 8)
for l in range(64):
    for j in range(8):
        if (l >> 5) * 8 + j == 5:
            a2[l, j] = 1.0
b2 = torch.zeros(64, 8)
for l in range(64):
    n = l & 31
    for j in range(8):
        k = (l >> 5) * 8 + j
        b2[l, j] = n + k / 32.0
d2 = hip._debug_mfma32_raw(a2.flatten(), b2.flatten()).cpu()
# read lane 0 i=0 (m=0, n=0 under hypothesis): value = 0 + k'/32
v = d2[0, 0].item()
print(f"probe2: one-hot placed at k=5 lands at k' = {v * 32:.1f} (want 5.0)")
v2 = d2[5, 0].item()  # lane 5 -> n=5: value = 5 + 5/32 = 5.15625
print(f"  lane5 i0 = {v2:.4f} (want 5.15625)")
