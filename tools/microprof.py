import time, numpy as np, torch
torch.cuda.init()
from mcp_context_forge_amd.ops import hip
from mcp_context_forge_amd.gpu.batch import pack_texts, pad_rows
from mcp_context_forge_amd.models.classifier import HashedTextClassifier
from mcp_context_forge_amd.gpu.classifier import GpuClassifier
from mcp_context_forge_amd.ops.dfa import compile_literals

dev = "cuda"
m = 8192
texts = [b'{"msg":"payload 123 lorem ipsum dolor sit amet","n":42}'] * m
blob = b"".join(texts)
offs = np.zeros(m + 1, dtype=np.int32)
np.cumsum(np.fromiter(map(len, texts), dtype=np.int32, count=m), out=offs[1:])
data_np = np.frombuffer(blob, dtype=np.uint8)
pin = torch.empty(64 << 20, dtype=torch.uint8, pin_memory=True)

banks = [(n, hip.DeviceScanTables(compile_literals([f"needle{n}{i}" for i in range(8)], case_insensitive=True), dev)) for n in "abcde"]
bs = hip.ScanBankSet(banks, dev)
_m = HashedTextClassifier(dim=4096, hidden=1024, classes=8)
clf = GpuClassifier(_m, dev)

def t(label, fn, n=30):
    torch.cuda.synchronize()
    # measure CPU-side (launch) time per call
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    dt_launch = (time.perf_counter() - t0) / n
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    dt_full = (time.perf_counter() - t0) / n
    print(f"{label:28s} launch {dt_launch*1e3:7.3f} ms   e2e {dt_full*1e3:7.3f} ms")

def upload():
    nbytes = data_np.nbytes
    pin[:nbytes] = torch.from_numpy(data_np.view(np.uint8))
    d = torch.empty(nbytes, dtype=torch.uint8, device=dev)
    d.copy_(pin[:nbytes], non_blocking=True)
    ot = torch.from_numpy(offs).to(dev, non_blocking=True)
    return d, ot[:-1], ot[1:]

d, b, e = upload()
t("upload(1MB pin+h2d)", lambda: upload())
t("scan_multi(5 banks)", lambda: hip.scan_multi(d, b, e, bs))
fb = hip.featurize(d, b, e, 4096)[0]
t("featurize(8192x4096)", lambda: hip.featurize(d, b, e, 4096))
t("pad_rows", lambda: pad_rows(fb, 128))
t("clf.forward(8192)", lambda: clf.forward(fb))
t("gemm_bt only", lambda: hip.gemm_bt(fb, clf.w1t, clf.b1, act=hip.ACT_GELU, out_bf16=True))

# rescan-size versions
m2 = 192
texts2 = texts[:m2]
d2, b2, e2 = pack_texts(texts2, dev)
t("pack_texts(192)", lambda: pack_texts(texts2, dev))
f2 = hip.featurize(d2, b2, e2, 4096)[0]
t("featurize(192)", lambda: hip.featurize(d2, b2, e2, 4096))
t("pad+forward(192)", lambda: clf.forward(pad_rows(f2, 128)))
