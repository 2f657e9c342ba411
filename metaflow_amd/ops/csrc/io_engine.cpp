// _mfx_io: standalone parallel-pread loader — deliberately NOT linked
// against torch (unlike _mfx_cas): importing it costs ~10 ms, so every
// small torch-less task subprocess can use the native CAS read path.
// (The torch-linked engine's first import pulls the full ~1.5 s torch
// runtime — measured as a 10x task-startup regression, ops/cas_native.py.)
// One entry point:
//   load_file(path, header_skip=0, threads=8) -> bytes
// allocates the result once and fills it with 16 MiB-chunk preads from
// a thread pool (GIL released). Reference analog: the s3op worker pool
// (reference plugins/datatools/s3/s3op.py:171) — here the "store" is
// the local CAS file and parallelism is intra-file.
#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

static PyObject* load_file(PyObject*, PyObject* args, PyObject* kwargs) {
  const char* path = nullptr;
  Py_ssize_t header_skip = 0;
  int threads = 8;
  static const char* kwlist[] = {"path", "header_skip", "threads",
                                 nullptr};
  if (!PyArg_ParseTupleAndKeywords(args, kwargs, "s|ni",
                                   const_cast<char**>(kwlist), &path,
                                   &header_skip, &threads))
    return nullptr;
  struct stat st;
  if (::stat(path, &st) != 0 || (Py_ssize_t)st.st_size < header_skip) {
    PyErr_Format(PyExc_OSError, "_mfx_io: missing or short file %s",
                 path);
    return nullptr;
  }
  const size_t n = (size_t)(st.st_size - header_skip);
  PyObject* obj = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)n);
  if (!obj) return nullptr;
  char* dst = PyBytes_AS_STRING(obj);
  bool ok = true;
  Py_BEGIN_ALLOW_THREADS;
  int fd = ::open(path, O_RDONLY);
  if (fd < 0) {
    ok = false;
  } else {
    constexpr size_t kChunk = 16u << 20;
    const size_t nchunks = n ? (n + kChunk - 1) / kChunk : 0;
    const int nt = (int)std::min<size_t>(
        std::max(1, threads), std::max<size_t>(nchunks, 1));
    std::atomic<size_t> next{0};
    std::atomic<bool> failed{false};
    auto worker = [&]() {
      size_t i;
      while (!failed.load(std::memory_order_relaxed) &&
             (i = next.fetch_add(1)) < nchunks) {
        const size_t off = i * kChunk;
        const size_t len = std::min(kChunk, n - off);
        size_t done = 0;
        while (done < len) {
          ssize_t r = ::pread(fd, dst + off + done, len - done,
                              (off_t)(header_skip + off + done));
          if (r <= 0) {
            failed.store(true);
            return;
          }
          done += (size_t)r;
        }
      }
    };
    if (nt <= 1) {
      worker();
    } else {
      std::vector<std::thread> pool;
      for (int t = 0; t < nt; ++t) pool.emplace_back(worker);
      for (auto& t : pool) t.join();
    }
    ::close(fd);
    ok = !failed.load();
  }
  Py_END_ALLOW_THREADS;
  if (!ok) {
    Py_DECREF(obj);
    PyErr_Format(PyExc_OSError, "_mfx_io: read failed for %s", path);
    return nullptr;
  }
  return obj;
}

static PyMethodDef Methods[] = {
    {"load_file", (PyCFunction)load_file, METH_VARARGS | METH_KEYWORDS,
     "load_file(path, header_skip=0, threads=8) -> bytes (parallel "
     "chunked pread, GIL released)"},
    {nullptr, nullptr, 0, nullptr}};

static struct PyModuleDef Module = {PyModuleDef_HEAD_INIT, "_mfx_io",
                                    "standalone parallel file loader",
                                    -1, Methods};

PyMODINIT_FUNC PyInit__mfx_io(void) { return PyModule_Create(&Module); }
