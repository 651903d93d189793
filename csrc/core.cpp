// modal_amd._core — native runtime core (C++20, pybind11).
//
// The reference ships native client runtimes alongside Python (Go ~19k LoC,
// TypeScript ~11k). This module is the C++ core of the
// MI355X-native runtime's data plane:
//
//  * ShmRing — single-producer/single-consumer byte ring in a mmap'd file:
//    the per-worker payload channel ("per-GPU input/output ring
//    buffers in pinned host memory"). Large input/output frames ride the
//    ring; the Unix socket stays the control/doorbell path. Lock-free
//    (acquire/release atomics), frames length-prefixed, wrap-around via a
//    skip sentinel. memcpy runs with the GIL released.
//
//  * pack_payloads / unpack_payloads — batch pickle-payload framing used by
//    the map fan-out: N buffers <-> one contiguous [u32 len]* blob without
//    per-item Python bytes handling.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <atomic>
#include <functional>
#include <thread>
#include <cerrno>
#include <cstdint>
#include <cstring>
#include <fcntl.h>
#include <stdexcept>
#include <string>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <vector>

namespace py = pybind11;

namespace {

struct RingHeader {
  alignas(64) std::atomic<uint64_t> head;  // bytes produced
  alignas(64) std::atomic<uint64_t> tail;  // bytes consumed
  alignas(64) uint64_t capacity;
  uint64_t magic;
};

constexpr uint64_t kMagic = 0x6d6f64616c616d64ULL;  // "modalamd"
constexpr uint32_t kSkip = 0xFFFFFFFFu;

class ShmRing {
 public:
  ShmRing(const std::string& path, uint64_t capacity, bool create) {
    int flags = create ? (O_RDWR | O_CREAT) : O_RDWR;
    fd_ = ::open(path.c_str(), flags, 0600);
    if (fd_ < 0) throw std::runtime_error("ShmRing: open failed: " + path);
    uint64_t total = sizeof(RingHeader) + capacity;
    if (create) {
      if (::ftruncate(fd_, (off_t)total) != 0) {
        ::close(fd_);
        throw std::runtime_error("ShmRing: ftruncate failed");
      }
    } else {
      struct stat st;
      if (::fstat(fd_, &st) != 0 || (uint64_t)st.st_size < sizeof(RingHeader)) {
        ::close(fd_);
        throw std::runtime_error("ShmRing: bad ring file");
      }
      total = (uint64_t)st.st_size;
      capacity = total - sizeof(RingHeader);
    }
    void* mem = ::mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd_, 0);
    if (mem == MAP_FAILED) {
      ::close(fd_);
      throw std::runtime_error("ShmRing: mmap failed");
    }
    base_ = static_cast<uint8_t*>(mem);
    header_ = reinterpret_cast<RingHeader*>(base_);
    data_ = base_ + sizeof(RingHeader);
    total_ = total;
    if (create) {
      header_->head.store(0, std::memory_order_relaxed);
      header_->tail.store(0, std::memory_order_relaxed);
      header_->capacity = capacity;
      header_->magic = kMagic;
    } else if (header_->magic != kMagic) {
      throw std::runtime_error("ShmRing: magic mismatch");
    }
    capacity_ = header_->capacity;
  }

  ~ShmRing() {
    if (base_) ::munmap(base_, total_);
    if (fd_ >= 0) ::close(fd_);
  }

  ShmRing(const ShmRing&) = delete;
  ShmRing& operator=(const ShmRing&) = delete;

  // producer: returns false when the frame does not fit right now
  bool push(py::buffer buf) {
    py::buffer_info info = buf.request();
    const uint8_t* src = static_cast<const uint8_t*>(info.ptr);
    const uint64_t n = (uint64_t)info.size * (uint64_t)info.itemsize;
    if (n + 8 > capacity_) throw std::runtime_error("frame larger than ring");
    uint64_t head = header_->head.load(std::memory_order_relaxed);
    uint64_t tail = header_->tail.load(std::memory_order_acquire);
    uint64_t pos = head % capacity_;
    uint64_t to_end = capacity_ - pos;
    uint64_t need = 4 + n;
    uint64_t skip = 0;
    if (to_end < 4) {
      skip = to_end;  // too small even for a length: pad to start
    } else if (to_end < need) {
      // write a skip sentinel and start the frame at offset 0
      skip = to_end;
    }
    if (head + skip + need - tail > capacity_) return false;  // full
    {
      py::gil_scoped_release release;
      if (skip) {
        if (to_end >= 4) {
          uint32_t sentinel = kSkip;
          std::memcpy(data_ + pos, &sentinel, 4);
        }
        pos = 0;
      }
      uint32_t len32 = (uint32_t)n;
      std::memcpy(data_ + pos, &len32, 4);
      std::memcpy(data_ + pos + 4, src, n);
    }
    header_->head.store(head + skip + need, std::memory_order_release);
    return true;
  }

  // consumer: drain up to max_frames frames (0 = all available)
  py::list pop_all(size_t max_frames = 0) {
    py::list out;
    uint64_t tail = header_->tail.load(std::memory_order_relaxed);
    uint64_t head = header_->head.load(std::memory_order_acquire);
    size_t count = 0;
    while (tail < head) {
      uint64_t pos = tail % capacity_;
      uint64_t to_end = capacity_ - pos;
      if (to_end < 4) {
        tail += to_end;  // padding at end (no room for a length)
        continue;
      }
      uint32_t len32;
      std::memcpy(&len32, data_ + pos, 4);
      if (len32 == kSkip) {
        tail += to_end;
        continue;
      }
      py::bytes frame(reinterpret_cast<const char*>(data_ + pos + 4), len32);
      out.append(std::move(frame));
      tail += 4 + len32;
      if (max_frames && ++count >= max_frames) break;
    }
    header_->tail.store(tail, std::memory_order_release);
    return out;
  }

  uint64_t pending_bytes() const {
    return header_->head.load(std::memory_order_acquire) -
           header_->tail.load(std::memory_order_acquire);
  }

  uint64_t capacity() const { return capacity_; }

 private:
  int fd_ = -1;
  uint8_t* base_ = nullptr;
  uint8_t* data_ = nullptr;
  RingHeader* header_ = nullptr;
  uint64_t capacity_ = 0;
  uint64_t total_ = 0;
};

py::bytes pack_payloads(const std::vector<py::bytes>& items) {
  uint64_t total = 0;
  std::vector<std::pair<const char*, size_t>> views;
  views.reserve(items.size());
  for (const auto& item : items) {
    char* ptr;
    Py_ssize_t n;
    if (PyBytes_AsStringAndSize(item.ptr(), &ptr, &n) != 0)
      throw std::runtime_error("pack_payloads: not bytes");
    views.emplace_back(ptr, (size_t)n);
    total += 4 + (uint64_t)n;
  }
  std::string out;
  out.resize(total);
  {
    py::gil_scoped_release release;
    char* dst = out.data();
    for (const auto& [ptr, n] : views) {
      uint32_t len32 = (uint32_t)n;
      std::memcpy(dst, &len32, 4);
      std::memcpy(dst + 4, ptr, n);
      dst += 4 + n;
    }
  }
  return py::bytes(out);
}

py::list unpack_payloads(py::buffer buf) {
  py::buffer_info info = buf.request();
  const char* src = static_cast<const char*>(info.ptr);
  size_t n = (size_t)info.size * (size_t)info.itemsize;
  py::list out;
  size_t pos = 0;
  while (pos + 4 <= n) {
    uint32_t len32;
    std::memcpy(&len32, src + pos, 4);
    if (pos + 4 + len32 > n) throw std::runtime_error("unpack_payloads: truncated");
    out.append(py::bytes(src + pos + 4, len32));
    pos += 4 + len32;
  }
  if (pos != n) throw std::runtime_error("unpack_payloads: trailing bytes");
  return out;
}

}  // namespace


// ---------------------------------------------------------------------------
// LZ4 block codec + MALZ41 container (CPU side of ops/csrc/lz4.hip).
//
// The CPU fallback previously ran through a pure-Python codec
// (utils/lz4ref.py, ~tens of MB/s); GPU-less consumers decompressing .z CAS
// entries and image/mount materialization now run at native speed. The
// container format is identical to the HIP kernels': 4 KiB segments, each a
// standard LZ4 block (or stored raw, comp_len 0). Segments compress and
// decompress in parallel across a small thread pool; the GIL is released
// for the whole pass.

static constexpr int kSegSize = 4096;
static constexpr int kMinMatch = 4;
static constexpr int kMFLimit = 12;
static constexpr int kLastLiterals = 5;

static inline uint32_t lz4_hash(uint32_t v) { return (v * 2654435761u) >> (32 - 12); }

// compress one block; returns compressed size or 0 when dst_cap exceeded
static size_t lz4_compress_one(const uint8_t* src, size_t n, uint8_t* dst, size_t dst_cap) {
  uint16_t table[1 << 12];
  std::memset(table, 0, sizeof(table));
  size_t op = 0, ip = 0, anchor = 0;
  auto emit = [&](size_t lit_start, size_t lit_len, int match_len) -> bool {
    size_t need = 1 + lit_len + (lit_len >= 15 ? 1 + (lit_len - 15) / 255 : 0) +
                  (match_len >= 0 ? 2 + 1 + 16 : 0);
    if (op + need + 8 > dst_cap) return false;
    int ml = match_len >= 0 ? match_len - kMinMatch : 0;
    uint8_t token = (uint8_t)((lit_len < 15 ? lit_len : 15) << 4);
    if (match_len >= 0) token |= (uint8_t)(ml < 15 ? ml : 15);
    dst[op++] = token;
    if (lit_len >= 15) {
      size_t rest = lit_len - 15;
      while (rest >= 255) { dst[op++] = 255; rest -= 255; }
      dst[op++] = (uint8_t)rest;
    }
    std::memcpy(dst + op, src + lit_start, lit_len);
    op += lit_len;
    return true;
  };
  if (n >= (size_t)(kMinMatch + kLastLiterals)) {
    size_t mflimit = n >= kMFLimit ? n - kMFLimit : 0;
    while (ip < mflimit) {
      uint32_t seq;
      std::memcpy(&seq, src + ip, 4);
      uint32_t h = lz4_hash(seq);
      size_t cand = table[h];
      table[h] = (uint16_t)ip;  // n <= 4096 so 16 bits suffice
      uint32_t cseq;
      bool ok = cand < ip && ip - cand <= 0xFFFF;
      if (ok) { std::memcpy(&cseq, src + cand, 4); ok = cseq == seq; }
      if (ok) {
        size_t mlen = kMinMatch;
        size_t maxm = n - kLastLiterals - ip;
        while (mlen < maxm && src[cand + mlen] == src[ip + mlen]) mlen++;
        if (!emit(anchor, ip - anchor, (int)mlen)) return 0;
        size_t off = ip - cand;
        dst[op++] = (uint8_t)(off & 0xFF);
        dst[op++] = (uint8_t)(off >> 8);
        int ml = (int)mlen - kMinMatch;
        if (ml >= 15) {
          int rest = ml - 15;
          while (rest >= 255) { dst[op++] = 255; rest -= 255; }
          dst[op++] = (uint8_t)rest;
        }
        ip += mlen;
        anchor = ip;
      } else {
        ip++;
      }
    }
  }
  if (!emit(anchor, n - anchor, -1)) return 0;
  return op;
}

static bool lz4_decompress_one(const uint8_t* src, size_t comp_len,
                               uint8_t* dst, size_t raw_len) {
  size_t ip = 0, op = 0;
  while (ip < comp_len) {
    uint8_t token = src[ip++];
    size_t lit = token >> 4;
    if (lit == 15) {
      uint8_t b;
      do { if (ip >= comp_len) return false; b = src[ip++]; lit += b; } while (b == 255);
    }
    if (ip + lit > comp_len || op + lit > raw_len) return false;
    std::memcpy(dst + op, src + ip, lit);
    ip += lit; op += lit;
    if (ip >= comp_len) break;  // last literals
    if (ip + 2 > comp_len) return false;
    size_t off = src[ip] | ((size_t)src[ip + 1] << 8);
    ip += 2;
    if (off == 0 || off > op) return false;
    size_t mlen = (token & 0xF) + kMinMatch;
    if ((token & 0xF) == 15) {
      uint8_t b;
      do { if (ip >= comp_len) return false; b = src[ip++]; mlen += b; } while (b == 255);
    }
    if (op + mlen > raw_len) return false;
    // overlapping copy: byte-by-byte when ranges overlap
    if (off >= mlen) {
      std::memcpy(dst + op, dst + op - off, mlen);
    } else {
      for (size_t k = 0; k < mlen; k++) dst[op + k] = dst[op - off + k];
    }
    op += mlen;
  }
  return op == raw_len;
}

static void parallel_for(size_t n, const std::function<void(size_t, size_t)>& body) {
  unsigned hw = std::thread::hardware_concurrency();
  size_t n_threads = std::min<size_t>(std::max(1u, hw / 2), 8);
  if (n < 8 || n_threads <= 1) { body(0, n); return; }
  std::vector<std::thread> threads;
  size_t chunk = (n + n_threads - 1) / n_threads;
  for (size_t t = 0; t < n_threads; t++) {
    size_t lo = t * chunk, hi = std::min(n, lo + chunk);
    if (lo >= hi) break;
    threads.emplace_back([&body, lo, hi] { body(lo, hi); });
  }
  for (auto& th : threads) th.join();
}

// MALZ41 container compress; returns None when incompressible (>= min_gain)
py::object malz_compress(py::bytes data, double min_gain) {
  char* buf; ssize_t n_s;
  if (PyBytes_AsStringAndSize(data.ptr(), &buf, &n_s) != 0) throw py::error_already_set();
  size_t n = (size_t)n_s;
  size_t n_seg = (n + kSegSize - 1) / kSegSize;
  if (n_seg == 0) return py::none();
  std::vector<uint32_t> comp_lens(n_seg, 0);
  std::vector<std::vector<uint8_t>> outs(n_seg);
  const uint8_t* src = (const uint8_t*)buf;
  {
    py::gil_scoped_release release;
    parallel_for(n_seg, [&](size_t lo, size_t hi) {
      for (size_t i = lo; i < hi; i++) {
        size_t seg_off = i * kSegSize;
        size_t seg_len = std::min((size_t)kSegSize, n - seg_off);
        outs[i].resize(kSegSize);  // only keep if it SAVES space
        size_t c = lz4_compress_one(src + seg_off, seg_len, outs[i].data(), seg_len > 1 ? seg_len - 1 : 0);
        if (c > 0 && c < seg_len) {
          comp_lens[i] = (uint32_t)c;
          outs[i].resize(c);
        } else {
          comp_lens[i] = 0;
          outs[i].clear();
        }
      }
    });
  }
  size_t total = 0;
  for (size_t i = 0; i < n_seg; i++) {
    size_t seg_len = std::min((size_t)kSegSize, n - i * kSegSize);
    total += comp_lens[i] ? comp_lens[i] : seg_len;
  }
  if ((double)total >= (double)n * min_gain) return py::none();
  size_t header = 6 + 12 + 4 * n_seg;
  py::bytes out_obj(nullptr, (ssize_t)(header + total));
  uint8_t* out = (uint8_t*)PyBytes_AsString(out_obj.ptr());
  std::memcpy(out, "MALZ41", 6);
  uint64_t n64 = n;
  uint32_t ns32 = (uint32_t)n_seg;
  std::memcpy(out + 6, &n64, 8);
  std::memcpy(out + 14, &ns32, 4);
  std::memcpy(out + 18, comp_lens.data(), 4 * n_seg);
  size_t op = header;
  for (size_t i = 0; i < n_seg; i++) {
    size_t seg_off = i * kSegSize;
    size_t seg_len = std::min((size_t)kSegSize, n - seg_off);
    if (comp_lens[i]) {
      std::memcpy(out + op, outs[i].data(), comp_lens[i]);
      op += comp_lens[i];
    } else {
      std::memcpy(out + op, src + seg_off, seg_len);
      op += seg_len;
    }
  }
  return out_obj;
}

py::bytes malz_decompress(py::bytes blob) {
  char* buf; ssize_t bn;
  if (PyBytes_AsStringAndSize(blob.ptr(), &buf, &bn) != 0) throw py::error_already_set();
  if (bn < 18 || std::memcmp(buf, "MALZ41", 6) != 0)
    throw std::runtime_error("not a MALZ41 container");
  uint64_t raw_len;
  uint32_t n_seg;
  std::memcpy(&raw_len, buf + 6, 8);
  std::memcpy(&n_seg, buf + 14, 4);
  size_t header = 6 + 12 + 4 * (size_t)n_seg;
  if ((size_t)bn < header) throw std::runtime_error("truncated container header");
  std::vector<uint32_t> comp_lens(n_seg);
  std::memcpy(comp_lens.data(), buf + 18, 4 * (size_t)n_seg);
  // per-segment payload offsets
  std::vector<size_t> seg_payload_off(n_seg);
  size_t pos = header;
  for (size_t i = 0; i < n_seg; i++) {
    seg_payload_off[i] = pos;
    size_t seg_len = std::min((size_t)kSegSize, (size_t)raw_len - i * kSegSize);
    pos += comp_lens[i] ? comp_lens[i] : seg_len;
  }
  if (pos > (size_t)bn) throw std::runtime_error("truncated container payload");
  py::bytes out_obj(nullptr, (ssize_t)raw_len);
  uint8_t* out = (uint8_t*)PyBytes_AsString(out_obj.ptr());
  const uint8_t* src = (const uint8_t*)buf;
  bool ok = true;
  {
    py::gil_scoped_release release;
    parallel_for(n_seg, [&](size_t lo, size_t hi) {
      for (size_t i = lo; i < hi; i++) {
        size_t seg_off = i * kSegSize;
        size_t seg_len = std::min((size_t)kSegSize, (size_t)raw_len - seg_off);
        if (comp_lens[i] == 0) {
          std::memcpy(out + seg_off, src + seg_payload_off[i], seg_len);
        } else if (!lz4_decompress_one(src + seg_payload_off[i], comp_lens[i],
                                       out + seg_off, seg_len)) {
          ok = false;
        }
      }
    });
  }
  if (!ok) throw std::runtime_error("corrupt LZ4 segment");
  return out_obj;
}




// ---------------------------------------------------------------------------
// Sandbox supervisor: clone3(CLONE_INTO_CGROUP) + execve.
//
// The Python road (asyncio preexec_fn) runs Python between fork and exec in
// a threaded process — both unsafe and racy for cgroup attachment (the child
// runs before the parent can write cgroup.procs). clone3 places the child in
// the target cgroup v2 ATOMICALLY at creation. Parity target: the
// reference's Go sandbox supervisor, done the Linux-native
// way.
// ---------------------------------------------------------------------------
#include <sys/syscall.h>
#include <sys/resource.h>
#include <sched.h>
#include <signal.h>

namespace {

struct clone_args_compat {
  uint64_t flags;
  uint64_t pidfd;
  uint64_t child_tid;
  uint64_t parent_tid;
  uint64_t exit_signal;
  uint64_t stack;
  uint64_t stack_size;
  uint64_t tls;
  uint64_t set_tid;
  uint64_t set_tid_size;
  uint64_t cgroup;
};

#ifndef CLONE_INTO_CGROUP
#define CLONE_INTO_CGROUP 0x200000000ULL
#endif

}  // namespace

// Returns the child pid. Throws on failure. stdio fds are dup2'd onto
// 0/1/2 in the child (pass -1 to inherit). The child setsid()s so the whole
// tree is killable as one process group.
static int64_t spawn_supervised(const std::vector<std::string>& argv,
                                const std::string& cwd,
                                const std::vector<std::string>& env,
                                const std::string& cgroup_dir,
                                int64_t rlimit_as_mib,
                                int stdin_fd, int stdout_fd, int stderr_fd) {
  if (argv.empty()) throw std::runtime_error("empty argv");
  int cg_fd = -1;
  if (!cgroup_dir.empty()) {
    cg_fd = ::open(cgroup_dir.c_str(), O_DIRECTORY | O_RDONLY | O_CLOEXEC);
    if (cg_fd < 0)
      throw std::runtime_error("open cgroup dir: " + std::string(strerror(errno)));
  }

  std::vector<char*> cargv;
  cargv.reserve(argv.size() + 1);
  for (const auto& a : argv) cargv.push_back(const_cast<char*>(a.c_str()));
  cargv.push_back(nullptr);
  std::vector<char*> cenv;
  cenv.reserve(env.size() + 1);
  for (const auto& e : env) cenv.push_back(const_cast<char*>(e.c_str()));
  cenv.push_back(nullptr);

  clone_args_compat ca{};
  ca.exit_signal = SIGCHLD;
  if (cg_fd >= 0) {
    ca.flags = CLONE_INTO_CGROUP;
    ca.cgroup = (uint64_t)cg_fd;
  }

  long pid = syscall(SYS_clone3, &ca, sizeof(ca));
  if (pid < 0 && cg_fd >= 0 && (errno == EINVAL || errno == ENOSYS || errno == EPERM ||
                                errno == EBADF || errno == EOPNOTSUPP)) {
    // kernel/cgroup too old or not delegated: plain clone3; the caller's
    // Python-side attach (child writes cgroup.procs) remains the fallback
    ::close(cg_fd);
    cg_fd = -1;
    throw std::runtime_error("clone3(CLONE_INTO_CGROUP) unavailable: " +
                             std::string(strerror(errno)));
  }
  if (pid < 0) {
    if (cg_fd >= 0) ::close(cg_fd);
    throw std::runtime_error("clone3: " + std::string(strerror(errno)));
  }
  if (pid == 0) {
    // child: async-signal-safe calls only
    ::setsid();
    if (stdin_fd >= 0) ::dup2(stdin_fd, 0);
    if (stdout_fd >= 0) ::dup2(stdout_fd, 1);
    if (stderr_fd >= 0) ::dup2(stderr_fd, 2);
    for (int fd = 3; fd < 1024; ++fd) ::close(fd);
    if (!cwd.empty() && ::chdir(cwd.c_str()) != 0) _exit(126);
    if (rlimit_as_mib > 0) {
      struct rlimit rl;
      rl.rlim_cur = rl.rlim_max = (rlim_t)rlimit_as_mib << 20;
      ::setrlimit(RLIMIT_AS, &rl);
    }
    ::execve(cargv[0], cargv.data(), cenv.data());
    _exit(127);
  }
  if (cg_fd >= 0) ::close(cg_fd);
  return (int64_t)pid;
}


PYBIND11_MODULE(_core, m) {
  m.doc() = "modal_amd native core: shm ring transport + batch framing";
  py::class_<ShmRing>(m, "ShmRing")
      .def(py::init<const std::string&, uint64_t, bool>(), py::arg("path"),
           py::arg("capacity") = (uint64_t)64 << 20, py::arg("create") = false)
      .def("push", &ShmRing::push)
      .def("pop_all", &ShmRing::pop_all, py::arg("max_frames") = 0)
      .def("pending_bytes", &ShmRing::pending_bytes)
      .def("capacity", &ShmRing::capacity);
  m.def("pack_payloads", &pack_payloads);
  m.def("malz_compress", &malz_compress, py::arg("data"), py::arg("min_gain") = 0.95);
  m.def("malz_decompress", &malz_decompress);
  m.def("unpack_payloads", &unpack_payloads);
  m.def("spawn_supervised", &spawn_supervised, py::arg("argv"), py::arg("cwd"),
        py::arg("env"), py::arg("cgroup_dir") = "", py::arg("rlimit_as_mib") = 0,
        py::arg("stdin_fd") = -1, py::arg("stdout_fd") = -1, py::arg("stderr_fd") = -1,
        "clone3(CLONE_INTO_CGROUP) + execve supervisor (atomic cgroup placement)");
}
