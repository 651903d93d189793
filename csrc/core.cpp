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

#include <atomic>
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
  m.def("unpack_payloads", &unpack_payloads);
}
