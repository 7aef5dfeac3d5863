import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
import __graft_entry__; __graft_entry__.build()
import tez_amd
from tez_amd._engine import lib, _ck

n = 500_000
conf = tez_amd.make_conf(256, key_type=tez_amd.KEY_TEXT, comparator=tez_amd.CMP_TEXT)
d, off, kl, part = tez_amd.generate(seed=0xB15, n=n, kind=1, klen=0, vlen=64, conf=conf)
offs = np.zeros(n+1, dtype=np.uint64)
_ck(lib().tzs_memcpy_d2h(offs.ctypes.data, off, 8*(n+1)), "d2h")
klen = np.zeros(n, dtype=np.uint32)
_ck(lib().tzs_memcpy_d2h(klen.ctypes.data, kl, 4*n), "d2h")
lens = np.diff(offs.astype(np.int64))
print("reclen min/max/mean:", lens.min(), lens.max(), lens.mean())
print("klen min/max:", klen.min(), klen.max())
print("monotonic offsets:", bool(np.all(lens > 0)))
parts = np.zeros(n, dtype=np.int32)
_ck(lib().tzs_memcpy_d2h(parts.ctypes.data, part, 4*n), "d2h")
print("parts min/max:", parts.min(), parts.max())
# distribution of keys: how many share the same first 5 content chars?
data = np.zeros(int(offs[-1]), dtype=np.uint8)
_ck(lib().tzs_memcpy_d2h(data.ctypes.data, d, int(offs[-1])), "d2h")
import collections
c5 = collections.Counter()
cfull = collections.Counter()
for i in range(0, n, 7):
    o0 = int(offs[i]); klc = int(klen[i])
    key = bytes(data[o0:o0+klc])
    c5[key[1:6]] += 1
    cfull[key] += 1
print("top first-5-content:", c5.most_common(3))
print("top full key dup count:", cfull.most_common(2))
tez_amd.free_device(d, off, kl, part)
