import sys
sys.path.insert(0, "/root/repo")
import numpy as np
import oracle
from tnc_amd import hiplib

rng = np.random.default_rng(0)
def r(shape):
    return (rng.standard_normal(shape) + 1j*rng.standard_normal(shape)).astype(np.complex128)

a = r((32, 8, 64)); b = r((64, 8, 32))
al, bl = [0,1,2], [2,3,4]
for out in ([0,1,3,4], [0,3,1,4]):
    ref = oracle.contract_ndarrays(out, al, a, bl, b)
    got = hiplib.einsum_c128(out, al, a, bl, b)
    d = np.abs(got-ref)
    print(out, "maxdiff", d.max(), "bad", np.count_nonzero(d>1e-8), "/", d.size)
    if d.max() > 1e-8:
        bad = np.argwhere(d > 1e-8)
        ok = np.argwhere(d <= 1e-8)
        print("  first bad:", bad[:4].tolist())
        print("  first ok:", ok[:4].tolist())
# K=64 with M=N=128 and symdiff order
a2 = r((128, 64)); b2 = r((64, 128))
ref = oracle.contract_ndarrays([0,2],[0,1],a2,[1,2],b2)
got = hiplib.einsum_c128([0,2],[0,1],a2,[1,2],b2)
print("M128K64N128 direct:", np.abs(got-ref).max())
# M=256 N=256 K=64 direct
a3 = r((256, 64)); b3 = r((64, 256))
ref = oracle.contract_ndarrays([0,2],[0,1],a3,[1,2],b3)
got = hiplib.einsum_c128([0,2],[0,1],a3,[1,2],b3)
d = np.abs(got-ref)
print("M256K64N256 direct:", d.max(), np.count_nonzero(d>1e-8), "/", d.size)
if d.max() > 1e-8:
    bad = np.argwhere(d>1e-8); print("  bad rows:", sorted(set(bad[:,0].tolist()))[:10], "...")
    print("  bad cols:", sorted(set(bad[:,1].tolist()))[:10])
