import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, glob, tempfile
from greptimedb_amd import _native, _hip_ops
from greptimedb_amd.engine import pagedec

rng = np.random.RandomState(2)
# build synthetic hybrid RLE streams directly and compare kernel vs cpu
def encode_hybrid(values, bw):
    out = bytearray([bw])
    i = 0
    n = len(values)
    while i < n:
        mode = rng.randint(2)
        if mode == 0:  # rle run
            run = min(n - i, rng.randint(1, 300))
            v = int(values[i]); values[i:i+run] = v
            # varint header
            h = run << 1
            while True:
                b = h & 0x7F; h >>= 7
                out.append(b | (0x80 if h else 0)); 
                if not h: break
            nb = (bw + 7)//8
            out += int(v).to_bytes(nb, "little")
            i += run
        else:  # bit packed groups
            groups = min((n - i)//8, rng.randint(1, 40))
            if groups == 0: continue
            cnt = groups*8
            h = (groups << 1) | 1
            while True:
                b = h & 0x7F; h >>= 7
                out.append(b | (0x80 if h else 0))
                if not h: break
            vals = values[i:i+cnt]
            bits = ((vals[:,None] >> np.arange(bw)) & 1).astype(np.uint8)
            out += np.packbits(bits.reshape(-1), bitorder="little").tobytes()
            i += cnt
    return bytes(out)

fails = 0
for trial in range(30):
    bw = rng.randint(1, 25)
    n = rng.randint(50, 20000)
    n -= n % 8  # keep bit-packed viable
    if n == 0: n = 8
    values = rng.randint(0, 1 << bw, n).astype(np.int64)
    payload = encode_hybrid(values.copy(), bw)
    blob = payload + b"\x00"*16
    runs, bw2 = _native.rle_run_table(blob, 0, len(payload), n)
    exp = pagedec._expand_cpu(np.asarray(runs), blob, bw2, n)
    runs_t = torch.as_tensor(np.ascontiguousarray(runs)).cuda()
    blob_t = torch.as_tensor(np.frombuffer(blob, np.uint8).copy()).cuda()
    got = _hip_ops.rle_expand_indices(runs_t, blob_t, bw2, n).cpu().numpy()
    if not np.array_equal(exp, got):
        bad = np.flatnonzero(exp != got)
        print("TRIAL", trial, "bw", bw2, "n", n, "bad", len(bad), "first", bad[:5],
              "exp", exp[bad[:5]], "got", got[bad[:5]])
        # dump runs around first bad
        i = bad[0]
        r = np.asarray(runs)
        k = np.searchsorted(r[:,3], i, side="right")-1
        print("run", r[k], "rel", i - r[k][3])
        fails += 1
        if fails > 3: break
print("done fails=", fails)
