import ctypes, importlib.util, os, sys, tempfile
spec = importlib.util.spec_from_file_location("_mfx_io", "build/asan/mfx_io_asan.so")
m = importlib.util.module_from_spec(spec)
spec.loader.exec_module(m)
# exercise: small, big, header_skip, missing file, many threads, empty
fd, path = tempfile.mkstemp()
os.write(fd, os.urandom(40 << 20)); os.close(fd)
data = m.load_file(path)
assert len(data) == 40 << 20
d2 = m.load_file(path, header_skip=8, threads=16)
assert len(d2) == (40 << 20) - 8 and d2 == data[8:]
fd, empty = tempfile.mkstemp(); os.close(fd)
assert m.load_file(empty) == b""
for _ in range(50):
    assert len(m.load_file(path, threads=3)) == 40 << 20
print("ASAN _mfx_io drive OK")
