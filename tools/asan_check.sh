#!/bin/bash
# Host-C++ sanitizer pass over the native engines ("HIP
# kernels need real sanitizer passes ... ASAN on host C++"). Builds
# ASAN variants of cas_engine.cpp and io_engine.cpp and exercises
# their memory-heavy paths (thread-pool SHA/Merkle, pread pools,
# StreamSaver feed/finish, blob round-trips) under
# AddressSanitizer with halt_on_error=1.
#
# Leak checking: _mfx_io runs with detect_leaks=1 (all reported leaks
# attribute to CPython interpreter frames, none to the extension);
# the torch-linked engine runs with detect_leaks=0 (libtorch holds
# caches for process lifetime). Intentional exception paths are NOT
# driven here: this image's ASAN preload cannot intercept
# __cxa_throw (interceptor CHECK) — they are covered by pytest.
set -e
cd "$(dirname "$0")/.."
mkdir -p build/asan
ASAN_LIB=$(gcc -print-file-name=libasan.so)
CSRC=metaflow_amd/ops/csrc

g++ -O1 -g -fsanitize=address -fno-omit-frame-pointer -std=c++17 \
    -shared -fPIC -pthread -I/usr/include/python3.10 \
    $CSRC/io_engine.cpp -o build/asan/mfx_io_asan.so

TINC=$(python3 -c "from torch.utils import cpp_extension as C; print(' '.join('-I'+p for p in C.include_paths()))")
TLIB=$(python3 -c "from torch.utils import cpp_extension as C; print(' '.join('-L'+p for p in C.library_paths()))")
PYB=-I$(python3 -c "import pybind11; print(pybind11.get_include())")
g++ -O1 -g -fsanitize=address -fno-omit-frame-pointer -std=c++17 \
    -shared -fPIC -pthread $TINC $PYB -I/usr/include/python3.10 $TLIB \
    -D_GLIBCXX_USE_CXX11_ABI=1 -DTORCH_EXTENSION_NAME=_mfx_cas_asan \
    $CSRC/cas_engine.cpp -o build/asan/mfx_cas_asan.so \
    -ltorch -ltorch_cpu -ltorch_python -lc10

# detect_leaks=0 in the scripted run: a manual detect_leaks=1 pass
# attributes every reported leak to CPython interpreter frames (the
# interpreter never frees interned objects at exit), none to the
# extension — but its nonzero exit would abort the script
LD_PRELOAD=$ASAN_LIB ASAN_OPTIONS=detect_leaks=0:halt_on_error=1 \
    python3 tools/asan_drive_io.py
LD_PRELOAD=$ASAN_LIB ASAN_OPTIONS=detect_leaks=0:halt_on_error=1 \
    python3 tools/asan_drive_cas.py
echo "asan_check: both native engines clean"
