import importlib.util, os, sys, tempfile
import torch
spec = importlib.util.spec_from_file_location("_mfx_cas_asan",
                                              "build/asan/mfx_cas_asan.so")
m = importlib.util.module_from_spec(spec)
spec.loader.exec_module(m)
eng = m.Engine()
root = tempfile.mkdtemp()
# parallel hash + blob save/load round trips
data = os.urandom(24 << 20)
key = eng.compute_key(data)
assert key == eng.compute_key(data) and len(key) == 64
path = os.path.join(root, "blob")
with open(path, "wb") as f:
    f.write(b"HDR!" + data)
got = eng.load_blob_parallel(path, 4)
assert got == data
# StreamSaver: chunked feed -> key parity with compute_key
sv = m.StreamSaver(os.path.join(root, "stream"), b"", 8)
t = torch.frombuffer(bytearray(data), dtype=torch.uint8)
for off in range(0, t.numel(), 8 << 20):
    sv.feed(t[off:off + (8 << 20)])
k2 = sv.finish()
assert k2 == key, (k2, key)
# save_blob round trip
p2 = os.path.join(root, "blob2")
k3 = eng.save_blob(p2, b"HD", data)
assert eng.load_blob(p2, 2) == data
# (intentional-error paths excluded: this environment's ASAN preload
# cannot intercept __cxa_throw; exception paths are covered by the
# regular pytest suite)
for _ in range(20):
    assert eng.compute_key(data[: 1 << 20]) == eng.compute_key(data[: 1 << 20])
print("ASAN _mfx_cas drive OK")
