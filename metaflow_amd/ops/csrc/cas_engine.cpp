// Native CAS engine: threaded SHA-256 hashing + direct blob file writes.
//
// Replaces the hot loop of the Python ContentAddressedStore for large blobs
// (the artifact-GB/s metric). SHA-256 is implemented here
// (public-domain style compression function, written fresh) and parallelized
// by chunking: large buffers are hashed as a Merkle-style root over 8 MiB
// leaf hashes so all cores contribute; small buffers hash single-threaded
// and identically to hashlib (callers must use the same convention on read,
// which holds because keys are opaque).

#include <torch/extension.h>

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>
#include <cstring>
#include <fstream>
#include <string>
#include <thread>
#include <vector>

namespace {

// ----------------------------------------------------------- sha-256 core
struct Sha256 {
  uint32_t h[8];
  uint64_t len = 0;
  uint8_t buf[64];
  size_t buf_len = 0;

  Sha256() {
    static const uint32_t init[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372,
                                     0xa54ff53a, 0x510e527f, 0x9b05688c,
                                     0x1f83d9ab, 0x5be0cd19};
    memcpy(h, init, sizeof(h));
  }

  static uint32_t rotr(uint32_t x, int n) {
    return (x >> n) | (x << (32 - n));
  }

  void block(const uint8_t* p) {
    static const uint32_t K[64] = {
        0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b,
        0x59f111f1, 0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01,
        0x243185be, 0x550c7dc3, 0x72be5d74, 0x80deb1fe, 0x9bdc06a7,
        0xc19bf174, 0xe49b69c1, 0xefbe4786, 0x0fc19dc6, 0x240ca1cc,
        0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da, 0x983e5152,
        0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
        0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc,
        0x53380d13, 0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85,
        0xa2bfe8a1, 0xa81a664b, 0xc24b8b70, 0xc76c51a3, 0xd192e819,
        0xd6990624, 0xf40e3585, 0x106aa070, 0x19a4c116, 0x1e376c08,
        0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a, 0x5b9cca4f,
        0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
        0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};
    uint32_t w[64];
    for (int i = 0; i < 16; ++i)
      w[i] = (uint32_t(p[i * 4]) << 24) | (uint32_t(p[i * 4 + 1]) << 16) |
             (uint32_t(p[i * 4 + 2]) << 8) | uint32_t(p[i * 4 + 3]);
    for (int i = 16; i < 64; ++i) {
      uint32_t s0 = rotr(w[i - 15], 7) ^ rotr(w[i - 15], 18) ^
                    (w[i - 15] >> 3);
      uint32_t s1 = rotr(w[i - 2], 17) ^ rotr(w[i - 2], 19) ^
                    (w[i - 2] >> 10);
      w[i] = w[i - 16] + s0 + w[i - 7] + s1;
    }
    uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5],
             g = h[6], hh = h[7];
    for (int i = 0; i < 64; ++i) {
      uint32_t S1 = rotr(e, 6) ^ rotr(e, 11) ^ rotr(e, 25);
      uint32_t ch = (e & f) ^ (~e & g);
      uint32_t t1 = hh + S1 + ch + K[i] + w[i];
      uint32_t S0 = rotr(a, 2) ^ rotr(a, 13) ^ rotr(a, 22);
      uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
      uint32_t t2 = S0 + maj;
      hh = g; g = f; f = e; e = d + t1;
      d = c; c = b; b = a; a = t1 + t2;
    }
    h[0] += a; h[1] += b; h[2] += c; h[3] += d;
    h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
  }

  void update(const uint8_t* p, size_t n) {
    len += n;
    if (buf_len) {
      while (n && buf_len < 64) {
        buf[buf_len++] = *p++;
        --n;
      }
      if (buf_len == 64) {
        block(buf);
        buf_len = 0;
      }
    }
    while (n >= 64) {
      block(p);
      p += 64;
      n -= 64;
    }
    while (n--) buf[buf_len++] = *p++;
  }

  void final(uint8_t out[32]) {
    uint64_t bits = len * 8;
    uint8_t pad = 0x80;
    update(&pad, 1);
    uint8_t z = 0;
    while (buf_len != 56) update(&z, 1);
    uint8_t lb[8];
    for (int i = 0; i < 8; ++i) lb[i] = (bits >> (56 - 8 * i)) & 0xff;
    update(lb, 8);
    for (int i = 0; i < 8; ++i) {
      out[i * 4] = h[i] >> 24;
      out[i * 4 + 1] = h[i] >> 16;
      out[i * 4 + 2] = h[i] >> 8;
      out[i * 4 + 3] = h[i];
    }
  }
};

std::string hex(const uint8_t* d, size_t n) {
  static const char* k = "0123456789abcdef";
  std::string s(n * 2, '0');
  for (size_t i = 0; i < n; ++i) {
    s[2 * i] = k[d[i] >> 4];
    s[2 * i + 1] = k[d[i] & 15];
  }
  return s;
}

std::string sha256_hex(const uint8_t* p, size_t n) {
  Sha256 s;
  s.update(p, n);
  uint8_t out[32];
  s.final(out);
  return hex(out, 32);
}

constexpr size_t kLeaf = 8 << 20;  // 8 MiB leaves for parallel hashing

// Parallel content key: sha256 over concatenated leaf sha256 digests,
// prefixed "MFXP1". Single-leaf inputs use plain sha256 (matches hashlib).
std::string content_key(const uint8_t* p, size_t n, int threads) {
  if (n <= kLeaf || threads <= 1) return sha256_hex(p, n);
  size_t nleaves = (n + kLeaf - 1) / kLeaf;
  std::vector<uint8_t> digests(nleaves * 32);
  std::atomic<size_t> next{0};
  auto worker = [&]() {
    size_t i;
    while ((i = next.fetch_add(1)) < nleaves) {
      size_t off = i * kLeaf;
      size_t len = std::min(kLeaf, n - off);
      Sha256 s;
      s.update(p + off, len);
      s.final(&digests[i * 32]);
    }
  };
  int nt = std::min<int>(threads, (int)nleaves);
  std::vector<std::thread> pool;
  for (int t = 0; t < nt; ++t) pool.emplace_back(worker);
  for (auto& t : pool) t.join();
  Sha256 root;
  root.update((const uint8_t*)"MFXP1", 5);
  root.update(digests.data(), digests.size());
  uint8_t out[32];
  root.final(out);
  return hex(out, 32);
}

class Engine {
 public:
  Engine() : threads_((int)std::thread::hardware_concurrency()) {}

  std::string compute_key(py::bytes data) {
    char* ptr = nullptr;
    Py_ssize_t n = 0;
    if (PyBytes_AsStringAndSize(data.ptr(), &ptr, &n) != 0)
      throw std::runtime_error("cas_engine: expected bytes");
    py::gil_scoped_release rel;
    return content_key((const uint8_t*)ptr, (size_t)n, threads_);
  }

  // atomic write: header + payload -> tmp -> rename
  void save_blob(const std::string& path, py::bytes header,
                 py::bytes payload) {
    std::string h = header;
    char* pptr = nullptr;
    Py_ssize_t pn = 0;
    if (PyBytes_AsStringAndSize(payload.ptr(), &pptr, &pn) != 0)
      throw std::runtime_error("cas_engine: expected bytes payload");
    std::string_view pv(pptr, (size_t)pn);
    py::gil_scoped_release rel;
    std::string dir = path.substr(0, path.find_last_of('/'));
    // mkdir -p without shelling out
    for (size_t i = 1; i < dir.size(); ++i) {
      if (dir[i] == '/') {
        ::mkdir(dir.substr(0, i).c_str(), 0755);
      }
    }
    ::mkdir(dir.c_str(), 0755);
    std::string tmp = path + ".tmp" + std::to_string(::getpid());
    {
      std::ofstream f(tmp, std::ios::binary);
      f.write(h.data(), h.size());
      f.write(pv.data(), pv.size());
      if (!f.good()) {
        ::unlink(tmp.c_str());
        throw std::runtime_error("cas_engine: write failed: " + tmp);
      }
    }
    if (::rename(tmp.c_str(), path.c_str()) != 0) {
      ::unlink(tmp.c_str());
      throw std::runtime_error("cas_engine: rename failed: " + path);
    }
  }

  py::bytes load_blob(const std::string& path, size_t header_skip) {
    std::ifstream f(path, std::ios::binary | std::ios::ate);
    if (!f.good()) throw std::runtime_error("cas_engine: missing " + path);
    size_t n = (size_t)f.tellg();
    if (n < header_skip) throw std::runtime_error("cas_engine: short blob");
    std::string out(n - header_skip, '\0');
    f.seekg(header_skip);
    f.read(&out[0], out.size());
    return py::bytes(out);
  }

  // Parallel pread straight into the result bytes object: one
  // allocation, zero intermediate copies (the Python path pays
  // f.read() + a header-slice copy, both single-threaded — the round-1
  // profile had load at 2.45 GB/s vs save at 8). Chunked pread from the
  // page cache scales with cores.
  py::bytes load_blob_parallel(const std::string& path,
                               size_t header_skip) {
    struct stat st;
    if (::stat(path.c_str(), &st) != 0)
      throw std::runtime_error("cas_engine: missing " + path);
    size_t total = (size_t)st.st_size;
    if (total < header_skip)
      throw std::runtime_error("cas_engine: short blob " + path);
    size_t n = total - header_skip;
    PyObject* obj = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)n);
    if (!obj) throw std::runtime_error("cas_engine: alloc failed");
    char* dst = PyBytes_AS_STRING(obj);
    bool ok = true;
    {
      py::gil_scoped_release rel;
      int fd = ::open(path.c_str(), O_RDONLY);
      if (fd < 0) {
        ok = false;
      } else {
        constexpr size_t kChunk = 16u << 20;
        size_t nchunks = (n + kChunk - 1) / kChunk;
        int nt = std::min<int>(threads_, (int)std::max<size_t>(nchunks,
                                                               1));
        std::atomic<size_t> next{0};
        std::atomic<bool> failed{false};
        auto worker = [&]() {
          size_t i;
          while (!failed.load(std::memory_order_relaxed) &&
                 (i = next.fetch_add(1)) < nchunks) {
            size_t off = i * kChunk;
            size_t len = std::min(kChunk, n - off);
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
        std::vector<std::thread> pool;
        for (int t = 0; t < nt; ++t) pool.emplace_back(worker);
        for (auto& t : pool) t.join();
        ::close(fd);
        ok = !failed.load();
      }
    }
    if (!ok) {
      Py_DECREF(obj);
      throw std::runtime_error("cas_engine: parallel read failed: " +
                               path);
    }
    return py::reinterpret_steal<py::bytes>(obj);
  }

  int threads() const { return threads_; }

 private:
  int threads_;
};

// ------------------------------------------------------------ StreamSaver
// Feed-based blob save with ZERO Python-side data work: each feed()
// releases the GIL, hashes the chunk's 8 MiB Merkle leaves across
// threads (reading straight from the caller's pinned buffer — measured
// 27 GB/s on MI355X hosts) while a writer thread appends the same bytes
// to the tmp file. finish() returns the content key (same convention as
// content_key/parallel_key, so dedup spans all save paths); the caller
// renames tmp into the CAS. Chunks must be 8 MiB multiples except the
// last (the checkpoint path's pinned halves are 256 MiB).
class StreamSaver {
 public:
  StreamSaver(std::string tmp_path, py::bytes header, int threads)
      : tmp_(std::move(tmp_path)),
        threads_(threads > 0 ? threads
                             : (int)std::thread::hardware_concurrency()) {
    std::string h = header;
    fd_ = ::open(tmp_.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
    if (fd_ < 0)
      throw std::runtime_error("StreamSaver: cannot open " + tmp_);
    if (::write(fd_, h.data(), h.size()) != (ssize_t)h.size())
      throw std::runtime_error("StreamSaver: header write failed");
  }

  ~StreamSaver() {
    if (fd_ >= 0) ::close(fd_);
  }

  void feed(torch::Tensor chunk) {
    TORCH_CHECK(chunk.device().is_cpu() &&
                chunk.scalar_type() == torch::kUInt8 &&
                chunk.is_contiguous(),
                "feed() wants a contiguous CPU uint8 tensor");
    TORCH_CHECK(fd_ >= 0, "StreamSaver already finished");
    TORCH_CHECK(total_ % kLeaf == 0,
                "only the final chunk may be a non-multiple of 8 MiB");
    const uint8_t* p = chunk.data_ptr<uint8_t>();
    const size_t n = (size_t)chunk.numel();
    py::gil_scoped_release rel;
    // running plain sha only matters while the blob could still end up
    // "small" (< 16 MiB total: the plain-sha key regime)
    if (plain_valid_) {
      if (total_ + n < (size_t)PARALLEL_MIN) {
        plain_.update(p, n);
      } else {
        plain_valid_ = false;
      }
    }
    const size_t nleaves = (n + kLeaf - 1) / kLeaf;
    const size_t base = digests_.size();
    digests_.resize(base + nleaves * 32);
    std::atomic<size_t> next{0};
    std::atomic<bool> wfail{false};
    std::thread writer([&] {
      size_t done = 0;
      while (done < n) {
        ssize_t w = ::write(fd_, p + done, n - done);
        if (w <= 0) {
          wfail.store(true);
          return;
        }
        done += (size_t)w;
      }
    });
    auto worker = [&] {
      size_t i;
      while ((i = next.fetch_add(1)) < nleaves) {
        const size_t off = i * kLeaf;
        Sha256 s;
        s.update(p + off, std::min(kLeaf, n - off));
        s.final(&digests_[base + i * 32]);
      }
    };
    int nt = std::min<int>(threads_, (int)std::max<size_t>(nleaves, 1));
    std::vector<std::thread> pool;
    for (int t = 0; t < nt; ++t) pool.emplace_back(worker);
    for (auto& t : pool) t.join();
    writer.join();
    if (wfail.load())
      throw std::runtime_error("StreamSaver: write failed: " + tmp_);
    total_ += n;
  }

  std::string finish() {
    TORCH_CHECK(fd_ >= 0, "StreamSaver already finished");
    ::close(fd_);
    fd_ = -1;
    py::gil_scoped_release rel;
    if (total_ < (size_t)PARALLEL_MIN && plain_valid_) {
      uint8_t out[32];
      plain_.final(out);
      return hex(out, 32);
    }
    Sha256 root;
    root.update((const uint8_t*)"MFXP1", 5);
    root.update(digests_.data(), digests_.size());
    uint8_t out[32];
    root.final(out);
    return hex(out, 32);
  }

  static constexpr long long PARALLEL_MIN = 16ll << 20;

 private:
  std::string tmp_;
  int threads_;
  int fd_ = -1;
  size_t total_ = 0;
  Sha256 plain_;
  bool plain_valid_ = true;
  std::vector<uint8_t> digests_;
};

}  // namespace

#include <sys/stat.h>
#include <unistd.h>

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  py::class_<Engine>(m, "Engine")
      .def(py::init<>())
      .def("compute_key", &Engine::compute_key)
      .def("save_blob", &Engine::save_blob)
      .def("load_blob", &Engine::load_blob)
      .def("load_blob_parallel", &Engine::load_blob_parallel)
      .def("threads", &Engine::threads);
  py::class_<StreamSaver>(m, "StreamSaver")
      .def(py::init<std::string, py::bytes, int>())
      .def("feed", &StreamSaver::feed)
      .def("finish", &StreamSaver::finish);
}
