import json
import random
import sys
import time

import numpy as np

from mcp_context_forge_amd.ops import hip

rng = random.Random(7)
rows = []
for i in range(1656):
    if i % 2 == 0:
        args = {"msg": f"payload {rng.randrange(1 << 30)} lorem ipsum dolor sit amet",
                "n": rng.randrange(100),
                "note": f"contact me at user{rng.randrange(1000)}@example.com"}
    else:
        args = {"msg": "this is   spaced\ttext", "n": i}
    rows.append(json.dumps(args, separators=(",", ":")).encode())
blob = b"".join(rows)
beg, end, off = [], [], 0
for r in rows:
    beg.append(off); off += len(r); end.append(off)
n = len(rows)
db = np.frombuffer(b"forbiddenblocked_word", np.uint8).copy()
do = np.asarray([0, 9, 21], np.int32)
harm = ["kill them all", "how to make a bomb", "ways to hurt myself", "buy stolen credit cards"]
hb = np.frombuffer(b"".join(p.encode() for p in harm), np.uint8).copy()
ho = np.zeros(len(harm) + 1, np.int32)
for i, p in enumerate(harm):
    ho[i + 1] = ho[i] + len(p)
A = (np.frombuffer(blob, np.uint8).copy(), np.asarray(beg, np.int32), np.asarray(end, np.int32))
fl = np.full(n, 15, np.uint8)
want = np.full(n, 0x3F, np.uint32)
for trial in (165, 1656, 1656, 1656):
    t0 = time.perf_counter()
    hip.rewrite_rows(A[0], A[1][:trial], A[2][:trial], fl[:trial], want[:trial],
                     0x3F, 0, True, True, deny_blob=db, deny_off=do, deny_ci=True,
                     harm_blob=hb, harm_off=ho)
    print(trial, "rows:", round((time.perf_counter() - t0) * 1000, 2), "ms", flush=True)
