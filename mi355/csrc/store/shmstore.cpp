// Native shared-memory object store core (SURVEY.md C4 — the plasma-store
// analog). POSIX shm segments with an 8-byte seal header written with
// release semantics; readers attach by name and acquire-load the length.
// Objects are sealed-then-immutable (race-freedom discipline, SURVEY.md
// §5.2). Python keeps serialization (cloudpickle-5); this core owns the
// segment lifecycle and zero-copy mappings.
//
// Built standalone (no torch dependency) as mi355/_store.so by
// tools/build_ext.py; mi355/runtime/core.py uses it when importable and
// falls back to multiprocessing.shared_memory otherwise.
#include <pybind11/pybind11.h>

#include <atomic>
#include <cerrno>
#include <cstring>
#include <fcntl.h>
#include <string>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <unordered_map>

namespace py = pybind11;

namespace {

constexpr size_t HEADER = 8;

struct Mapping {
  void* addr = nullptr;
  size_t size = 0;
};

// process-local cache of attached mappings (mappings live until process exit
// or explicit close; deserialized buffers may alias them)
std::unordered_map<std::string, Mapping>& registry() {
  static std::unordered_map<std::string, Mapping> r;
  return r;
}

Mapping map_fd(int fd, size_t size, bool writable) {
  void* addr = mmap(nullptr, size, writable ? (PROT_READ | PROT_WRITE) : PROT_READ,
                    MAP_SHARED, fd, 0);
  if (addr == MAP_FAILED) throw std::runtime_error(std::string("mmap: ") + strerror(errno));
  return Mapping{addr, size};
}

std::atomic<uint64_t>* seal_word(const Mapping& m) {
  return reinterpret_cast<std::atomic<uint64_t>*>(m.addr);
}

}  // namespace

// Create a segment sized for `payload` bytes; returns a writable memoryview
// of the payload region. Not visible to readers until seal().
py::memoryview create(const std::string& name, size_t payload_bytes) {
  int fd = shm_open(("/" + name).c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
  if (fd < 0) throw std::runtime_error("shm_open(create " + name + "): " + strerror(errno));
  size_t total = HEADER + payload_bytes;
  if (ftruncate(fd, (off_t)total) != 0) {
    close(fd);
    shm_unlink(("/" + name).c_str());
    throw std::runtime_error(std::string("ftruncate: ") + strerror(errno));
  }
  Mapping m = map_fd(fd, total, true);
  close(fd);
  seal_word(m)->store(0, std::memory_order_relaxed);
  registry()[name] = m;
  return py::memoryview::from_memory(static_cast<char*>(m.addr) + HEADER,
                                     payload_bytes);
}

// Publish: release-store the payload length. After this the object is
// immutable by convention.
void seal(const std::string& name, size_t payload_bytes) {
  auto it = registry().find(name);
  if (it == registry().end()) throw std::runtime_error("seal: unknown segment " + name);
  seal_word(it->second)->store((uint64_t)payload_bytes, std::memory_order_release);
}

// Attach (or use cached mapping); returns readonly memoryview of the sealed
// payload, or None if the segment doesn't exist / isn't sealed yet.
py::object attach(const std::string& name) {
  auto& reg = registry();
  auto it = reg.find(name);
  if (it == reg.end()) {
    int fd = shm_open(("/" + name).c_str(), O_RDONLY, 0);
    if (fd < 0) return py::none();
    struct stat st;
    if (fstat(fd, &st) != 0 || (size_t)st.st_size < HEADER) {
      close(fd);
      return py::none();
    }
    Mapping m = map_fd(fd, (size_t)st.st_size, false);
    close(fd);
    it = reg.emplace(name, m).first;
  }
  uint64_t n = seal_word(it->second)->load(std::memory_order_acquire);
  if (n == 0) return py::none();
  return py::memoryview::from_memory(
      static_cast<const char*>(it->second.addr) + HEADER, (size_t)n);
}

bool ready(const std::string& name) {
  return !attach(name).is_none();
}

void unlink_seg(const std::string& name) {
  shm_unlink(("/" + name).c_str());
}

void close_seg(const std::string& name) {
  auto& reg = registry();
  auto it = reg.find(name);
  if (it != reg.end()) {
    munmap(it->second.addr, it->second.size);
    reg.erase(it);
  }
}

size_t n_attached() { return registry().size(); }

PYBIND11_MODULE(_store, m) {
  m.doc() = "mi355 native shm object store core";
  m.def("create", &create, py::arg("name"), py::arg("payload_bytes"));
  m.def("seal", &seal);
  m.def("attach", &attach);
  m.def("ready", &ready);
  m.def("unlink", &unlink_seg);
  m.def("close", &close_seg);
  m.def("n_attached", &n_attached);
}
