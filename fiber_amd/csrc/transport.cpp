// fiber_amd._transport — single-node shared-memory message transport.
//
// Role: replaces the reference's nanomsg TCP data plane (uber/fiber
// fiber/socket.py:297-425 NanomsgContext/NanomsgDevice) with an MI355X-node
// native engine: variable-size message rings in POSIX shared memory guarded
// by process-shared ROBUST pthread mutexes + condvars.  A ring is MPMC:
// any number of producers and consumers in any process on the node.  A
// consumer that dies while holding the lock does not deadlock the ring
// (EOWNERDEAD -> pthread_mutex_consistent), which is what makes the
// resilient pool's worker-kill recovery safe at the transport level.
//
// Message payloads are opaque bytes (pickled host metadata; device tensors
// ride as ~100-byte HIP IPC handles produced by fiber_amd.serialization,
// so the ring never carries tensor data).
//
// Layout of a segment (/dev/shm/<name>):
//   [Header | data area of `capacity` bytes]
// Records in the data area: [u32 len][payload][pad to 8B].  A WRAP marker
// (len == 0xFFFFFFFF) means "skip to offset 0".

#include <pybind11/pybind11.h>

#include <atomic>
#include <cerrno>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <string>

#include <fcntl.h>
#include <pthread.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <signal.h>
#include <time.h>
#include <unistd.h>

namespace py = pybind11;

namespace {

constexpr uint32_t kMagic = 0xFA3B71A6u;
constexpr uint32_t kWrapMarker = 0xFFFFFFFFu;
constexpr size_t kAlign = 8;

// Record layout: [u32 len][u32 state][u32 writer_pid][u32 _pad][payload..]
// state: 0 = reserved (payload being copied outside the lock),
//        1 = committed.  A reserved record whose writer died is reclaimed
// by the next reader (crash safety for mid-copy kills).
constexpr uint32_t kStReserved = 0u;
constexpr uint32_t kStCommitted = 1u;
constexpr size_t kRecHdr = 16;

struct Header {
  uint32_t magic;
  uint32_t ready;  // set to 1 once the creator finished initialization
  pthread_mutex_t mu;
  pthread_cond_t not_empty;
  pthread_cond_t not_full;
  uint64_t capacity;   // bytes in data area
  uint64_t head;       // consumer offset
  uint64_t tail;       // producer offset
  uint64_t used;       // bytes currently occupied (records incl. padding)
  uint64_t msg_count;  // messages currently in the ring
  uint64_t total_in;   // lifetime enqueued messages
  uint64_t total_out;  // lifetime dequeued messages
  uint32_t closed;
};

inline size_t record_bytes(size_t len) {
  return (kRecHdr + len + kAlign - 1) & ~(kAlign - 1);
}

inline bool process_alive(uint32_t pid) {
  if (pid == 0) return false;
  if (kill((pid_t)pid, 0) != 0 && errno == ESRCH) return false;
  // kill(0) reports zombies as alive; a zombie writer will never commit,
  // so check the /proc state byte (field after the comm parens).
  char path[64];
  snprintf(path, sizeof(path), "/proc/%u/stat", pid);
  FILE* f = fopen(path, "r");
  if (!f) return false;
  char buf[512];
  size_t n = fread(buf, 1, sizeof(buf) - 1, f);
  fclose(f);
  buf[n] = 0;
  const char* p = strrchr(buf, ')');
  if (p && p[1] == ' ' && (p[2] == 'Z' || p[2] == 'X')) return false;
  return true;
}

inline void make_deadline(double timeout_s, struct timespec* ts) {
  clock_gettime(CLOCK_MONOTONIC, ts);
  time_t sec = static_cast<time_t>(timeout_s);
  long nsec = static_cast<long>((timeout_s - (double)sec) * 1e9);
  ts->tv_sec += sec;
  ts->tv_nsec += nsec;
  if (ts->tv_nsec >= 1000000000L) {
    ts->tv_sec += 1;
    ts->tv_nsec -= 1000000000L;
  }
}

// cond waits re-acquire the mutex internally and can therefore return
// EOWNERDEAD too (previous owner died while holding).  Ignoring that and
// re-waiting on an inconsistent mutex is UB (observed as a permanent
// wedge under worker-kill chaos) — every wait must run
// pthread_mutex_consistent before continuing.
inline int cond_wait_robust(pthread_cond_t* c, pthread_mutex_t* m) {
  int rc = pthread_cond_wait(c, m);
  if (rc == EOWNERDEAD) {
    pthread_mutex_consistent(m);
    rc = 0;
  }
  return rc;
}

inline int cond_timedwait_robust(pthread_cond_t* c, pthread_mutex_t* m,
                                 const struct timespec* ts) {
  int rc = pthread_cond_timedwait(c, m, ts);
  if (rc == EOWNERDEAD) {
    pthread_mutex_consistent(m);
    rc = 0;
  }
  return rc;
}

class RobustLock {
 public:
  explicit RobustLock(pthread_mutex_t* mu) : mu_(mu) {
    int rc = pthread_mutex_lock(mu_);
    if (rc == EOWNERDEAD) {
      // Previous owner died mid-critical-section.  Ring mutations are
      // ordered so that head/tail/used advance only after payload writes,
      // so the state is consistent enough to continue.
      pthread_mutex_consistent(mu_);
    } else if (rc != 0) {
      throw std::runtime_error("mutex lock failed: " + std::to_string(rc));
    }
  }
  ~RobustLock() { pthread_mutex_unlock(mu_); }

 private:
  pthread_mutex_t* mu_;
};

class ShmRing {
 public:
  ShmRing(const std::string& name, bool create, size_t capacity,
          double open_timeout)
      : name_(name), owner_(create) {
    size_t total = sizeof(Header) + capacity;
    int fd = -1;
    if (create) {
      shm_unlink(name.c_str());  // stale segment from a crashed run
      fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
      if (fd < 0) throw std::runtime_error("shm_open create failed: " + name);
      if (ftruncate(fd, (off_t)total) != 0) {
        close(fd);
        shm_unlink(name.c_str());
        throw std::runtime_error("ftruncate failed: " + name);
      }
    } else {
      // The binder may not have created the segment yet; retry briefly.
      struct timespec start;
      clock_gettime(CLOCK_MONOTONIC, &start);
      for (;;) {
        fd = shm_open(name.c_str(), O_RDWR, 0600);
        if (fd >= 0) break;
        struct timespec now;
        clock_gettime(CLOCK_MONOTONIC, &now);
        double waited = (now.tv_sec - start.tv_sec) +
                        (now.tv_nsec - start.tv_nsec) * 1e-9;
        if (waited > open_timeout)
          throw std::runtime_error("shm ring not found: " + name);
        usleep(2000);
      }
      struct stat st;
      if (fstat(fd, &st) != 0) {
        close(fd);
        throw std::runtime_error("fstat failed: " + name);
      }
      total = (size_t)st.st_size;
      capacity = total - sizeof(Header);
    }

    void* mem =
        mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    close(fd);
    if (mem == MAP_FAILED)
      throw std::runtime_error("mmap failed: " + name);
    hdr_ = reinterpret_cast<Header*>(mem);
    data_ = reinterpret_cast<char*>(mem) + sizeof(Header);
    map_len_ = total;

    if (create) {
      std::memset(hdr_, 0, sizeof(Header));
      hdr_->capacity = capacity;

      pthread_mutexattr_t ma;
      pthread_mutexattr_init(&ma);
      pthread_mutexattr_setpshared(&ma, PTHREAD_PROCESS_SHARED);
      pthread_mutexattr_setrobust(&ma, PTHREAD_MUTEX_ROBUST);
      pthread_mutex_init(&hdr_->mu, &ma);
      pthread_mutexattr_destroy(&ma);

      pthread_condattr_t ca;
      pthread_condattr_init(&ca);
      pthread_condattr_setpshared(&ca, PTHREAD_PROCESS_SHARED);
      pthread_condattr_setclock(&ca, CLOCK_MONOTONIC);
      pthread_cond_init(&hdr_->not_empty, &ca);
      pthread_cond_init(&hdr_->not_full, &ca);
      pthread_condattr_destroy(&ca);

      hdr_->magic = kMagic;
      std::atomic_thread_fence(std::memory_order_release);
      hdr_->ready = 1;
    } else {
      // Wait until creator finished header init.
      struct timespec start;
      clock_gettime(CLOCK_MONOTONIC, &start);
      while (hdr_->ready != 1) {
        struct timespec now;
        clock_gettime(CLOCK_MONOTONIC, &now);
        double waited = (now.tv_sec - start.tv_sec) +
                        (now.tv_nsec - start.tv_nsec) * 1e-9;
        if (waited > open_timeout) {
          munmap(mem, total);
          throw std::runtime_error("shm ring never became ready: " + name);
        }
        usleep(1000);
      }
      if (hdr_->magic != kMagic) {
        munmap(mem, total);
        throw std::runtime_error("bad shm ring magic: " + name);
      }
    }
  }

  ~ShmRing() { detach(); }

  void detach() {
    if (hdr_) {
      munmap(reinterpret_cast<void*>(hdr_), map_len_);
      hdr_ = nullptr;
      data_ = nullptr;
    }
  }

  // timeout < 0: block forever; timeout == 0: non-blocking.
  // Returns false on timeout; throws if the ring is closed.
  // The payload memcpy happens OUTSIDE the lock (reserve -> copy ->
  // commit), so concurrent producers/consumers overlap their copies.
  bool send(const char* buf, size_t len, double timeout) {
    size_t need = record_bytes(len);
    if (need + kRecHdr + kAlign >= hdr_->capacity)
      throw std::runtime_error("message larger than ring capacity");
    struct timespec deadline;
    if (timeout > 0) make_deadline(timeout, &deadline);

    uint64_t rec;
    {
      RobustLock lock(&hdr_->mu);
      for (;;) {
        if (hdr_->closed) throw std::runtime_error("ring closed");
        // Worst case we also need a wrap marker record.
        if (hdr_->capacity - hdr_->used >= need + kRecHdr + kAlign) break;
        if (timeout == 0) return false;
        int rc;
        if (timeout < 0) {
          rc = cond_wait_robust(&hdr_->not_full, &hdr_->mu);
        } else {
          rc = cond_timedwait_robust(&hdr_->not_full, &hdr_->mu,
                                     &deadline);
          if (rc == ETIMEDOUT) return false;
        }
        (void)rc;
      }

      uint64_t cap = hdr_->capacity;
      uint64_t tail = hdr_->tail;
      if (tail + need > cap) {
        // Not enough contiguous space: write wrap marker, jump to 0.
        uint32_t marker = kWrapMarker;
        std::memcpy(data_ + tail, &marker, 4);
        hdr_->used += cap - tail;
        tail = 0;
      }
      rec = tail;
      uint32_t len32 = (uint32_t)len;
      uint32_t pid = (uint32_t)getpid();
      std::memcpy(data_ + rec, &len32, 4);
      reinterpret_cast<std::atomic<uint32_t>*>(data_ + rec + 4)
          ->store(kStReserved, std::memory_order_relaxed);
      std::memcpy(data_ + rec + 8, &pid, 4);
      hdr_->tail = (tail + need) % cap;
      hdr_->used += need;
      hdr_->msg_count += 1;
      hdr_->total_in += 1;
    }

    if (len) std::memcpy(data_ + rec + kRecHdr, buf, len);
    reinterpret_cast<std::atomic<uint32_t>*>(data_ + rec + 4)
        ->store(kStCommitted, std::memory_order_release);
    {
      // Lock-protected signal so a reader between predicate-check and
      // cond_wait cannot miss the wakeup.
      RobustLock lock(&hdr_->mu);
      pthread_cond_signal(&hdr_->not_empty);
    }
    return true;
  }

  // Returns (found, payload).  found=false on timeout.  Throws when the
  // ring is closed AND drained.
  bool recv(std::string* out, double timeout) {
    struct timespec deadline;
    if (timeout > 0) make_deadline(timeout, &deadline);
    // Bounded wait even for infinite timeouts so dead-writer reclaim
    // gets a chance to run.
    struct timespec tick;

    RobustLock lock(&hdr_->mu);
    int stuck_polls = 0;
    for (;;) {
      if (hdr_->msg_count > 0) {
        // resolve head (following a wrap marker) and check commit state
        uint64_t cap = hdr_->capacity;
        uint64_t head = hdr_->head;
        uint32_t len32;
        std::memcpy(&len32, data_ + head, 4);
        if (len32 == kWrapMarker) {
          hdr_->used -= cap - head;
          hdr_->head = head = 0;
          std::memcpy(&len32, data_ + head, 4);
        }
        uint32_t state =
            reinterpret_cast<std::atomic<uint32_t>*>(data_ + head + 4)
                ->load(std::memory_order_acquire);
        if (state == kStCommitted) {
          out->assign(data_ + head + kRecHdr, len32);
          size_t need = record_bytes(len32);
          hdr_->head = (head + need) % cap;
          hdr_->used -= need;
          hdr_->msg_count -= 1;
          hdr_->total_out += 1;
          pthread_cond_signal(&hdr_->not_full);
          return true;
        }
        // Reserved record: writer is copying.  If the writer died
        // mid-copy, reclaim the record so the ring cannot wedge.
        if (stuck_polls > 20) {  // ~2s of 100ms polls
          uint32_t pid;
          std::memcpy(&pid, data_ + head + 8, 4);
          if (!process_alive(pid)) {
            size_t need = record_bytes(len32);
            hdr_->head = (head + need) % cap;
            hdr_->used -= need;
            hdr_->msg_count -= 1;
            pthread_cond_signal(&hdr_->not_full);
            stuck_polls = 0;
            continue;
          }
          stuck_polls = 0;
        }
        // short poll wait for the commit
        make_deadline(0.1, &tick);
        int rc = cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &tick);
        if (rc == ETIMEDOUT) {
          ++stuck_polls;
          if (timeout > 0) {
            struct timespec now;
            clock_gettime(CLOCK_MONOTONIC, &now);
            if (now.tv_sec > deadline.tv_sec ||
                (now.tv_sec == deadline.tv_sec &&
                 now.tv_nsec >= deadline.tv_nsec))
              return false;
          }
          if (timeout == 0) return false;
        }
        continue;
      }
      if (hdr_->closed) throw std::runtime_error("ring closed");
      if (timeout == 0) return false;
      int rc;
      if (timeout < 0) {
        rc = cond_wait_robust(&hdr_->not_empty, &hdr_->mu);
      } else {
        rc = cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &deadline);
        if (rc == ETIMEDOUT) return false;
      }
      (void)rc;
    }
  }

  // Size of the next committed message, or -1 on timeout.  Does not
  // consume.  (Used with recv_into for single-copy receives.)
  int64_t peek_size(double timeout) {
    struct timespec deadline;
    if (timeout > 0) make_deadline(timeout, &deadline);
    struct timespec tick;
    RobustLock lock(&hdr_->mu);
    for (;;) {
      if (hdr_->msg_count > 0) {
        uint64_t cap = hdr_->capacity;
        uint64_t head = hdr_->head;
        uint32_t len32;
        std::memcpy(&len32, data_ + head, 4);
        if (len32 == kWrapMarker) {
          hdr_->used -= cap - head;
          hdr_->head = head = 0;
          std::memcpy(&len32, data_ + head, 4);
        }
        uint32_t state =
            reinterpret_cast<std::atomic<uint32_t>*>(data_ + head + 4)
                ->load(std::memory_order_acquire);
        if (state == kStCommitted) return (int64_t)len32;
        make_deadline(0.1, &tick);
        cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &tick);
        if (timeout == 0) return -1;
        continue;
      }
      if (hdr_->closed) throw std::runtime_error("ring closed");
      if (timeout == 0) return -1;
      int rc;
      if (timeout < 0) {
        rc = cond_wait_robust(&hdr_->not_empty, &hdr_->mu);
      } else {
        rc = cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &deadline);
        if (rc == ETIMEDOUT) return -1;
      }
      (void)rc;
    }
  }

  // Single-copy receive: copies the next committed message into buf.
  // Returns the message length, -1 on timeout, or -(len) - 2 if buf is
  // too small (message left in the ring).
  int64_t recv_into(char* buf, size_t buflen, double timeout) {
    struct timespec deadline;
    if (timeout > 0) make_deadline(timeout, &deadline);
    struct timespec tick;
    RobustLock lock(&hdr_->mu);
    int stuck_polls = 0;
    for (;;) {
      if (hdr_->msg_count > 0) {
        uint64_t cap = hdr_->capacity;
        uint64_t head = hdr_->head;
        uint32_t len32;
        std::memcpy(&len32, data_ + head, 4);
        if (len32 == kWrapMarker) {
          hdr_->used -= cap - head;
          hdr_->head = head = 0;
          std::memcpy(&len32, data_ + head, 4);
        }
        uint32_t state =
            reinterpret_cast<std::atomic<uint32_t>*>(data_ + head + 4)
                ->load(std::memory_order_acquire);
        if (state == kStCommitted) {
          if ((size_t)len32 > buflen) return -((int64_t)len32) - 2;
          std::memcpy(buf, data_ + head + kRecHdr, len32);
          size_t need = record_bytes(len32);
          hdr_->head = (head + need) % cap;
          hdr_->used -= need;
          hdr_->msg_count -= 1;
          hdr_->total_out += 1;
          pthread_cond_signal(&hdr_->not_full);
          return (int64_t)len32;
        }
        if (stuck_polls > 20) {
          uint32_t pid;
          std::memcpy(&pid, data_ + head + 8, 4);
          if (!process_alive(pid)) {
            size_t need = record_bytes(len32);
            hdr_->head = (head + need) % cap;
            hdr_->used -= need;
            hdr_->msg_count -= 1;
            pthread_cond_signal(&hdr_->not_full);
            stuck_polls = 0;
            continue;
          }
          stuck_polls = 0;
        }
        make_deadline(0.1, &tick);
        int rc = cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &tick);
        if (rc == ETIMEDOUT) {
          ++stuck_polls;
          if (timeout == 0) return -1;
          if (timeout > 0) {
            struct timespec now;
            clock_gettime(CLOCK_MONOTONIC, &now);
            if (now.tv_sec > deadline.tv_sec ||
                (now.tv_sec == deadline.tv_sec &&
                 now.tv_nsec >= deadline.tv_nsec))
              return -1;
          }
        }
        continue;
      }
      if (hdr_->closed) throw std::runtime_error("ring closed");
      if (timeout == 0) return -1;
      int rc;
      if (timeout < 0) {
        rc = cond_wait_robust(&hdr_->not_empty, &hdr_->mu);
      } else {
        rc = cond_timedwait_robust(&hdr_->not_empty, &hdr_->mu, &deadline);
        if (rc == ETIMEDOUT) return -1;
      }
      (void)rc;
    }
  }

  void close_ring() {
    if (!hdr_) return;
    // Time-boxed: teardown must never wedge even if the mutex is stuck
    // (e.g. an undiagnosed robust-mutex corner under SIGKILL chaos).
    // The closed flag is a u32 checked inside every wait loop, so a
    // lock-free store + broadcast still wakes and drains all waiters.
    struct timespec deadline;
    clock_gettime(CLOCK_REALTIME, &deadline);
    deadline.tv_sec += 5;
    int rc = pthread_mutex_timedlock(&hdr_->mu, &deadline);
    if (rc == EOWNERDEAD) {
      pthread_mutex_consistent(&hdr_->mu);
      rc = 0;
    }
    hdr_->closed = 1;
    pthread_cond_broadcast(&hdr_->not_empty);
    pthread_cond_broadcast(&hdr_->not_full);
    if (rc == 0) pthread_mutex_unlock(&hdr_->mu);
  }

  void unlink_ring() { shm_unlink(name_.c_str()); }

  uint64_t size() const { return hdr_ ? hdr_->msg_count : 0; }
  uint64_t total_in() const { return hdr_ ? hdr_->total_in : 0; }
  uint64_t total_out() const { return hdr_ ? hdr_->total_out : 0; }
  bool closed() const { return hdr_ ? hdr_->closed != 0 : true; }
  bool is_owner() const { return owner_; }
  const std::string& name() const { return name_; }

 private:
  std::string name_;
  bool owner_;
  Header* hdr_ = nullptr;
  char* data_ = nullptr;
  size_t map_len_ = 0;
};

}  // namespace

PYBIND11_MODULE(_transport, m) {
  m.doc() = "fiber_amd shared-memory message transport (single MI355X node)";

  py::class_<ShmRing>(m, "ShmRing")
      .def(py::init<const std::string&, bool, size_t, double>(),
           py::arg("name"), py::arg("create"),
           py::arg("capacity") = (size_t)(8 << 20),
           py::arg("open_timeout") = 20.0,
           py::call_guard<py::gil_scoped_release>())
      .def(
          "send",
          [](ShmRing& r, py::buffer buf, double timeout) {
            py::buffer_info info = buf.request();
            const char* ptr = static_cast<const char*>(info.ptr);
            size_t len = (size_t)info.size * (size_t)info.itemsize;
            bool ok;
            {
              py::gil_scoped_release release;
              ok = r.send(ptr, len, timeout);
            }
            return ok;
          },
          py::arg("data"), py::arg("timeout") = -1.0)
      .def(
          "recv",
          [](ShmRing& r, double timeout) -> py::object {
            std::string out;
            bool ok;
            {
              py::gil_scoped_release release;
              ok = r.recv(&out, timeout);
            }
            if (!ok) return py::none();
            return py::bytes(out);
          },
          py::arg("timeout") = -1.0)
      .def(
          "peek_size",
          [](ShmRing& r, double timeout) {
            py::gil_scoped_release release;
            return r.peek_size(timeout);
          },
          py::arg("timeout") = -1.0)
      .def(
          "recv_into",
          [](ShmRing& r, py::buffer buf, double timeout) {
            py::buffer_info info = buf.request(true);
            char* ptr = static_cast<char*>(info.ptr);
            size_t cap = (size_t)info.size * (size_t)info.itemsize;
            int64_t n;
            {
              py::gil_scoped_release release;
              n = r.recv_into(ptr, cap, timeout);
            }
            return n;
          },
          py::arg("buf"), py::arg("timeout") = -1.0)
      .def("close", &ShmRing::close_ring,
           py::call_guard<py::gil_scoped_release>())
      .def("unlink", &ShmRing::unlink_ring)
      .def("detach", &ShmRing::detach)
      .def_property_readonly("size", &ShmRing::size)
      .def_property_readonly("total_in", &ShmRing::total_in)
      .def_property_readonly("total_out", &ShmRing::total_out)
      .def_property_readonly("is_closed", &ShmRing::closed)
      .def_property_readonly("is_owner", &ShmRing::is_owner)
      .def_property_readonly("name", &ShmRing::name);
}
