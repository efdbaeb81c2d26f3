// fiber_amd._transport — single-node shared-memory message transport.
//
// Role: replaces the reference's nanomsg TCP data plane (uber/fiber
// fiber/socket.py:297-425 NanomsgContext/NanomsgDevice) with an MI355X-node
// native engine: variable-size message rings in POSIX shared memory.
// A ring is MPMC: any number of producers and consumers in any process on
// the node.
//
// CRASH SAFETY (the whole point of the design):
//   * state mutations are guarded by a process-shared ROBUST pthread
//     mutex; every multi-field mutation is a roll-forward TRANSACTION
//     (Header::txn): post-state journaled + armed before the apply, so
//     the next EOWNERDEAD locker lands the dead owner's mutation
//     exactly — SIGKILL between any two field writes cannot skew
//     used/msg_count/spill accounting (the long chaos soak caught the
//     earlier "ordering makes it consistent-enough" version leaking
//     `used` bytes per in-lock kill until the ring wedged);
//   * blocking uses RAW FUTEX sequence words, NOT pthread condvars:
//     glibc condvars contain an internal NON-robust lock, so a process
//     SIGKILLed inside cond_wait's bookkeeping permanently wedges every
//     later broadcast/wait (observed live under worker-kill chaos).
//     Futex wait/wake are single syscalls with no shared-state locking —
//     kill-safe at every instruction.  All waits are additionally chunked
//     at 1 s so even a missed wake (producer dying between its state
//     update and its FUTEX_WAKE) only costs a bounded stall;
//   * a payload is copied OUTSIDE the lock (reserve -> copy -> commit).
//     A reserved record whose writer died (incl. zombies) is reclaimed
//     by the next reader.
//
// Message payloads are opaque bytes (pickled host metadata; device
// tensors ride as ~100-byte HIP IPC handles produced by
// fiber_amd.serialization, so the ring never carries tensor data).
//
// Layout of a segment (/dev/shm/<name>):
//   [Header | data area of `capacity` bytes]
// Records: [u32 len][u32 state][u32 writer_pid][u32 pad][payload..] padded
// to 8 B.  A WRAP marker (len == 0xFFFFFFFF) means "skip to offset 0".

// FAM_NO_PYBIND: built as a plain C++ TU by the TSAN harness
// (csrc/tsan_harness.cpp) — same engine code, no Python linkage.
#ifndef FAM_NO_PYBIND
#include <pybind11/pybind11.h>
#endif

#include <atomic>
#include <cerrno>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include <dirent.h>
#include <fcntl.h>
#include <linux/futex.h>
#include <pthread.h>
#include <signal.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <time.h>
#include <unistd.h>

#ifndef FAM_NO_PYBIND
namespace py = pybind11;
#endif

namespace {

constexpr uint32_t kMagic = 0xFA3B71AAu;  // bumped: txn journal
// Outstanding spill segments per ring are capped: spilled payloads
// bypass the ring-capacity backpressure, so without this a fast
// producer of huge messages could fill /dev/shm (RAM) unboundedly.
// 16 in flight preserves pipelining; producers block (normal not-full
// wait) once the cap is reached.
constexpr uint64_t kMaxSpillSegs = 16;
constexpr uint32_t kWrapMarker = 0xFFFFFFFFu;
constexpr size_t kAlign = 8;

constexpr uint32_t kStReserved = 0u;
constexpr uint32_t kStCommitted = 1u;
constexpr uint32_t kFlagSpill = 1u;  // record flags word (header byte 12)
constexpr size_t kRecHdr = 16;
constexpr size_t kMaxBatch = 128;  // records per lock hold in batched ops

struct Header {
  uint32_t magic;
  uint32_t ready;  // set to 1 once the creator finished initialization
  pthread_mutex_t mu;
  // futex sequence words: bumped whenever the corresponding condition
  // may have become true, woken with FUTEX_WAKE (skipped when the
  // waiter counter is zero — saves a syscall per message on the fast
  // path)
  uint32_t fut_not_empty;
  uint32_t fut_not_full;
  uint32_t waiters_not_empty;
  uint32_t waiters_not_full;
  uint64_t capacity;   // bytes in data area
  uint64_t head;       // consumer offset
  uint64_t tail;       // producer offset
  uint64_t used;       // bytes currently occupied (records incl. padding)
  uint64_t msg_count;  // messages currently in the ring
  uint64_t total_in;   // lifetime enqueued messages
  uint64_t total_out;  // lifetime dequeued messages
  uint64_t spill_count;  // outstanding spill segments (backpressure)
  uint32_t closed;
  // Crash-atomicity journal.  Every multi-field state mutation first
  // writes its POST-state here and arms txn_stage (release), applies,
  // then disarms (release).  A process SIGKILLed mid-apply leaves
  // stage==1; the next EOWNERDEAD locker rolls the snapshot forward.
  // (The long chaos soak proved "mutation order makes partial death
  // consistent-enough" wrong: a kill between used+= and msg_count+=
  // leaks `used` bytes forever — ~600 kills wedged an 1 MB ring.)
  uint32_t txn_stage;
  uint64_t txn[7];  // head, tail, used, msg_count, total_in, total_out,
                    // spill_count (post-state)
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

inline double monotonic_now() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec + ts.tv_nsec * 1e-9;
}

inline std::atomic<uint32_t>* as_atomic(uint32_t* p) {
  return reinterpret_cast<std::atomic<uint32_t>*>(p);
}

// Shared (cross-process) futex wait: returns after a wake, timeout,
// value mismatch or spurious wakeup — callers always re-check state.
inline void futex_wait(uint32_t* word, uint32_t expected, double max_s) {
  struct timespec rel;
  if (max_s > 1.0) max_s = 1.0;  // bounded: missed-wake insurance
  if (max_s <= 0) return;
  rel.tv_sec = (time_t)max_s;
  rel.tv_nsec = (long)((max_s - (double)rel.tv_sec) * 1e9);
  syscall(SYS_futex, word, FUTEX_WAIT, expected, &rel, nullptr, 0);
}

inline void futex_wake_all(uint32_t* word) {
  syscall(SYS_futex, word, FUTEX_WAKE, INT32_MAX, nullptr, nullptr, 0);
}

// Bump the sequence word and wake everyone — the syscall is skipped
// when no one waits.  A waiter that registers after our check re-reads
// the word before sleeping (snapshot-first ordering), so it cannot
// sleep through this bump.
inline void bump_and_wake(uint32_t* word, uint32_t* waiters) {
  as_atomic(word)->fetch_add(1, std::memory_order_seq_cst);
  if (as_atomic(waiters)->load(std::memory_order_seq_cst) != 0)
    futex_wake_all(word);
}

struct WaiterScope {
  uint32_t* w;
  explicit WaiterScope(uint32_t* waiters) : w(waiters) {
    as_atomic(w)->fetch_add(1, std::memory_order_seq_cst);
  }
  ~WaiterScope() { as_atomic(w)->fetch_sub(1, std::memory_order_seq_cst); }
};

// ---------------------------------------------------------------------------
// Spill segments: a message too big for the ring rides a one-shot shm
// segment; the ring carries a small control record (kFlagSpill) naming it.
// The reader copies the payload out and unlinks the segment; the ring
// owner sweeps stale "<ring>.sp.*" segments on create and unlink (leaks
// only happen when a writer dies between creating the segment and
// committing the control record).
// ---------------------------------------------------------------------------

// Control payload: [u64 real_len][u16 name_len][name bytes]
inline size_t spill_ctrl_encode(char* out, uint64_t real_len,
                                const char* name, size_t name_len) {
  std::memcpy(out, &real_len, 8);
  uint16_t nl = (uint16_t)name_len;
  std::memcpy(out + 8, &nl, 2);
  std::memcpy(out + 10, name, name_len);
  return 10 + name_len;
}

inline bool spill_ctrl_decode(const char* ctrl, size_t ctrl_len,
                              uint64_t* real_len, std::string* name) {
  if (ctrl_len < 10) return false;
  std::memcpy(real_len, ctrl, 8);
  uint16_t nl;
  std::memcpy(&nl, ctrl + 8, 2);
  if (ctrl_len < (size_t)10 + nl) return false;
  name->assign(ctrl + 10, nl);
  return true;
}

// Writes `len` bytes into a fresh exclusive shm segment; returns its name.
inline std::string spill_write(const std::string& ring_name, const char* buf,
                               size_t len) {
  static std::atomic<uint64_t> counter{0};
  char nm[192];
  snprintf(nm, sizeof(nm), "%s.sp.%u.%llu", ring_name.c_str(),
           (unsigned)getpid(),
           (unsigned long long)counter.fetch_add(1));
  int fd = shm_open(nm, O_CREAT | O_EXCL | O_RDWR, 0600);
  if (fd < 0)
    throw std::runtime_error(std::string("spill shm_open failed: ") + nm);
  size_t total = len ? len : 1;
  if (ftruncate(fd, (off_t)total) != 0) {
    close(fd);
    shm_unlink(nm);
    throw std::runtime_error(std::string("spill ftruncate failed: ") + nm);
  }
  void* mem = mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (mem == MAP_FAILED) {
    shm_unlink(nm);
    throw std::runtime_error(std::string("spill mmap failed: ") + nm);
  }
  if (len) std::memcpy(mem, buf, len);
  munmap(mem, total);
  return std::string(nm);
}

// Copies the spill payload into dst and unlinks the segment.
inline void spill_read(const std::string& nm, char* dst, uint64_t len) {
  int fd = shm_open(nm.c_str(), O_RDONLY, 0600);
  if (fd < 0)
    throw std::runtime_error("spill segment missing: " + nm);
  if (len) {
    void* mem = mmap(nullptr, len, PROT_READ, MAP_SHARED, fd, 0);
    close(fd);
    if (mem == MAP_FAILED) {
      shm_unlink(nm.c_str());
      throw std::runtime_error("spill mmap(read) failed: " + nm);
    }
    std::memcpy(dst, mem, len);
    munmap(mem, len);
  } else {
    close(fd);
  }
  shm_unlink(nm.c_str());
}

// Unlink every "<ring>.sp.*" segment (stale leftovers of crashed writers).
inline void spill_sweep(const std::string& ring_name) {
  DIR* d = opendir("/dev/shm");
  if (!d) return;
  std::string prefix = ring_name + ".sp.";
  struct dirent* e;
  std::vector<std::string> victims;
  while ((e = readdir(d)) != nullptr) {
    if (strncmp(e->d_name, prefix.c_str(), prefix.size()) == 0)
      victims.emplace_back(e->d_name);
  }
  closedir(d);
  for (auto& v : victims) shm_unlink(v.c_str());
}

// Journal helpers: arm writes the intended post-state, apply copies it
// into the live fields, disarm closes the transaction.  Stage
// transitions are release stores so a crash at ANY instruction leaves
// either stage==0 (nothing or everything applied) or stage==1 with a
// complete snapshot (roll forward).
inline void txn_arm(Header* h, uint64_t head, uint64_t tail, uint64_t used,
                    uint64_t msg, uint64_t tin, uint64_t tout,
                    uint64_t spill) {
  h->txn[0] = head;
  h->txn[1] = tail;
  h->txn[2] = used;
  h->txn[3] = msg;
  h->txn[4] = tin;
  h->txn[5] = tout;
  h->txn[6] = spill;
  as_atomic(&h->txn_stage)->store(1, std::memory_order_release);
}

inline void txn_apply(Header* h) {
  h->head = h->txn[0];
  h->tail = h->txn[1];
  h->used = h->txn[2];
  h->msg_count = h->txn[3];
  h->total_in = h->txn[4];
  h->total_out = h->txn[5];
  h->spill_count = h->txn[6];
  as_atomic(&h->txn_stage)->store(0, std::memory_order_release);
}

inline void txn_recover(Header* h) {
  if (as_atomic(&h->txn_stage)->load(std::memory_order_acquire) == 1)
    txn_apply(h);  // roll the dead owner's mutation forward
}

class RobustLock {
 public:
  explicit RobustLock(pthread_mutex_t* mu, Header* hdr = nullptr)
      : mu_(mu) {
    int rc = pthread_mutex_lock(mu_);
    if (rc == EOWNERDEAD) {
      // Previous owner died mid-critical-section: roll its armed
      // transaction forward (see Header::txn), then mark consistent.
      if (hdr) txn_recover(hdr);
      pthread_mutex_consistent(mu_);
    } else if (rc != 0) {
      throw std::runtime_error("mutex lock failed: " + std::to_string(rc));
    }
  }
  ~RobustLock() {
    if (mu_) pthread_mutex_unlock(mu_);
  }
  void unlock() {
    pthread_mutex_unlock(mu_);
    mu_ = nullptr;
  }

 private:
  pthread_mutex_t* mu_;
};

class ShmRing {
 public:
  ShmRing(const std::string& name, bool create, size_t capacity,
          double open_timeout)
      : name_(name), owner_(create) {
    // The record format assumes every offset is kAlign-aligned; an
    // unaligned user-configured capacity could leave < 4 bytes for the
    // wrap marker at the end of the data area.  Round down + floor.
    capacity &= ~(kAlign - 1);
    if (capacity < 4096) capacity = 4096;
    size_t total = sizeof(Header) + capacity;
    int fd = -1;
    if (create) {
      shm_unlink(name.c_str());  // stale segment from a crashed run
      spill_sweep(name);         // + stale spill segments
      fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
      if (fd < 0)
        throw std::runtime_error("shm_open create failed: " + name);
      if (ftruncate(fd, (off_t)total) != 0) {
        close(fd);
        shm_unlink(name.c_str());
        throw std::runtime_error("ftruncate failed: " + name);
      }
    } else {
      // The binder may not have created the segment yet; retry briefly.
      double start = monotonic_now();
      for (;;) {
        fd = shm_open(name.c_str(), O_RDWR, 0600);
        if (fd >= 0) break;
        if (monotonic_now() - start > open_timeout)
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
    if (mem == MAP_FAILED) throw std::runtime_error("mmap failed: " + name);
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

      hdr_->magic = kMagic;
      std::atomic_thread_fence(std::memory_order_release);
      hdr_->ready = 1;
    } else {
      double start = monotonic_now();
      while (hdr_->ready != 1) {
        if (monotonic_now() - start > open_timeout) {
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

  // Spill policy: a message that cannot fit, or would occupy more than a
  // quarter of the ring (starving other producers), rides a spill segment.
  bool needs_spill(size_t len) const {
    size_t need = record_bytes(len);
    uint64_t cap = hdr_->capacity;
    return need + kRecHdr + kAlign >= cap || need > cap / 4;
  }

  // timeout < 0: block forever; timeout == 0: non-blocking.
  // Returns false on timeout; throws if the ring is closed.
  bool send(const char* buf, size_t len, double timeout) {
    if (needs_spill(len)) {
      std::string nm = spill_write(name_, buf, len);
      char ctrl[224];
      size_t clen = spill_ctrl_encode(ctrl, len, nm.c_str(), nm.size());
      bool ok = send_record(ctrl, clen, kFlagSpill, timeout);
      if (!ok) shm_unlink(nm.c_str());  // timed out: reclaim the segment
      return ok;
    }
    return send_record(buf, len, 0, timeout);
  }

  bool send_record(const char* buf, size_t len, uint32_t flags,
                   double timeout) {
    size_t need = record_bytes(len);
    if (need + kRecHdr + kAlign >= hdr_->capacity)
      throw std::runtime_error("message larger than ring capacity");
    const double deadline =
        timeout > 0 ? monotonic_now() + timeout : 0.0;

    uint64_t rec;
    for (;;) {
      // snapshot BEFORE the predicate check: any state change after this
      // bumps the word, so our futex_wait cannot sleep through it
      uint32_t snap = as_atomic(&hdr_->fut_not_full)
                          ->load(std::memory_order_acquire);
      {
        RobustLock lock(&hdr_->mu, hdr_);
        if (hdr_->closed) throw std::runtime_error("ring closed");
        // Exact fit check: a record that would straddle the end costs an
        // extra `cap - tail` wasted bytes (wrap marker + dead space).
        // That waste can approach `need`, so it must be priced exactly —
        // a fixed margin lets a wrapping record overrun the reader's
        // head when the ring is nearly full (used > capacity, unsigned
        // free-space underflow, total corruption).
        uint64_t cap = hdr_->capacity;
        // empty ring: rewind to offset 0 so the largest message always
        // has a contiguous region regardless of where tail drifted
        if (hdr_->msg_count == 0 && hdr_->used == 0)
          hdr_->head = hdr_->tail = 0;
        uint64_t tail = hdr_->tail;
        uint64_t waste = (tail + need > cap) ? (cap - tail) : 0;
        const bool spill_ok = !(flags & kFlagSpill) ||
                              hdr_->spill_count < kMaxSpillSegs;
        if (spill_ok && cap - hdr_->used >= need + waste) {
          // DATA writes first (not journaled — invisible until the
          // state transaction lands), then one atomic state txn.
          if (waste) {
            uint32_t marker = kWrapMarker;
            std::memcpy(data_ + tail, &marker, 4);
            tail = 0;
          }
          rec = tail;
          uint32_t len32 = (uint32_t)len;
          uint32_t pid = (uint32_t)getpid();
          std::memcpy(data_ + rec, &len32, 4);
          as_atomic(reinterpret_cast<uint32_t*>(data_ + rec + 4))
              ->store(kStReserved, std::memory_order_relaxed);
          std::memcpy(data_ + rec + 8, &pid, 4);
          std::memcpy(data_ + rec + 12, &flags, 4);
          txn_arm(hdr_, hdr_->head, (tail + need) % cap,
                  hdr_->used + waste + need, hdr_->msg_count + 1,
                  hdr_->total_in + 1, hdr_->total_out,
                  hdr_->spill_count +
                      ((flags & kFlagSpill) ? 1 : 0));
          txn_apply(hdr_);
          break;
        }
        if (timeout == 0) return false;
      }
      double remaining = 1.0;
      if (timeout > 0) {
        remaining = deadline - monotonic_now();
        if (remaining <= 0) return false;
      }
      {
        WaiterScope ws(&hdr_->waiters_not_full);
        if (as_atomic(&hdr_->fut_not_full)
                ->load(std::memory_order_seq_cst) == snap)
          futex_wait(&hdr_->fut_not_full, snap, remaining);
      }
    }

    if (len) std::memcpy(data_ + rec + kRecHdr, buf, len);
    as_atomic(reinterpret_cast<uint32_t*>(data_ + rec + 4))
        ->store(kStCommitted, std::memory_order_release);
    bump_and_wake(&hdr_->fut_not_empty, &hdr_->waiters_not_empty);
    return true;
  }

  // Batched producer: enqueue msgs[done..] with ONE lock hold + ONE wake
  // per burst of records that fit (the mutex/futex round trip is the
  // dominant cost for small messages — see profiles).  Same
  // reserve -> copy-outside-the-lock -> commit protocol and crash-safety
  // story as send(); a record whose writer dies mid-copy is reclaimed by
  // readers exactly as for single sends.  Returns the number of messages
  // enqueued (< n only on timeout / non-blocking backpressure).
  size_t send_many(const char* const* bufs, const size_t* lens, size_t n,
                   double timeout) {
    // Pre-spill oversized messages OUTSIDE the lock: each becomes a small
    // flagged control record; segments of messages that never get sent
    // (timeout) are reclaimed below.
    std::vector<const char*> xbufs(bufs, bufs + n);
    std::vector<size_t> xlens(lens, lens + n);
    std::vector<uint32_t> flags(n, 0);
    std::vector<std::string> ctrl_store(n);
    std::vector<std::string> seg_names(n);
    for (size_t i = 0; i < n; ++i) {
      if (needs_spill(lens[i])) {
        seg_names[i] = spill_write(name_, bufs[i], lens[i]);
        ctrl_store[i].resize(224);
        size_t clen = spill_ctrl_encode(&ctrl_store[i][0], lens[i],
                                        seg_names[i].c_str(),
                                        seg_names[i].size());
        ctrl_store[i].resize(clen);
        xbufs[i] = ctrl_store[i].data();
        xlens[i] = clen;
        flags[i] = kFlagSpill;
      }
    }
    size_t done = send_many_records(xbufs.data(), xlens.data(),
                                    flags.data(), n, timeout);
    for (size_t i = done; i < n; ++i)
      if (flags[i]) shm_unlink(seg_names[i].c_str());
    return done;
  }

  size_t send_many_records(const char* const* bufs, const size_t* lens,
                           const uint32_t* flags, size_t n, double timeout) {
    const double deadline = timeout > 0 ? monotonic_now() + timeout : 0.0;
    size_t done = 0;
    uint64_t recs[kMaxBatch];
    while (done < n) {
      size_t burst = 0;
      for (;;) {
        uint32_t snap = as_atomic(&hdr_->fut_not_full)
                            ->load(std::memory_order_acquire);
        {
          RobustLock lock(&hdr_->mu, hdr_);
          if (hdr_->closed) throw std::runtime_error("ring closed");
          uint64_t cap = hdr_->capacity;
          while (done + burst < n && burst < kMaxBatch) {
            size_t len = lens[done + burst];
            size_t need = record_bytes(len);
            if (need + kRecHdr + kAlign >= cap)
              throw std::runtime_error("message larger than ring capacity");
            if (hdr_->msg_count == 0 && hdr_->used == 0)
              hdr_->head = hdr_->tail = 0;  // see send()
            uint64_t tail = hdr_->tail;
            // exact wrap-waste pricing — see send()
            uint64_t waste = (tail + need > cap) ? (cap - tail) : 0;
            if (cap - hdr_->used < need + waste) break;
            if ((flags[done + burst] & kFlagSpill) &&
                hdr_->spill_count >= kMaxSpillSegs)
              break;  // spill budget backpressure
            if (waste) {
              uint32_t marker = kWrapMarker;
              std::memcpy(data_ + tail, &marker, 4);
              tail = 0;
            }
            uint32_t len32 = (uint32_t)len;
            uint32_t pid = (uint32_t)getpid();
            std::memcpy(data_ + tail, &len32, 4);
            as_atomic(reinterpret_cast<uint32_t*>(data_ + tail + 4))
                ->store(kStReserved, std::memory_order_relaxed);
            std::memcpy(data_ + tail + 8, &pid, 4);
            std::memcpy(data_ + tail + 12, &flags[done + burst], 4);
            recs[burst] = tail;
            txn_arm(hdr_, hdr_->head, (tail + need) % cap,
                    hdr_->used + waste + need, hdr_->msg_count + 1,
                    hdr_->total_in + 1, hdr_->total_out,
                    hdr_->spill_count +
                        ((flags[done + burst] & kFlagSpill) ? 1 : 0));
            txn_apply(hdr_);
            ++burst;
          }
          if (burst > 0) break;
          if (timeout == 0) return done;
        }
        double remaining = 1.0;
        if (timeout > 0) {
          remaining = deadline - monotonic_now();
          if (remaining <= 0) return done;
        }
        {
          WaiterScope ws(&hdr_->waiters_not_full);
          if (as_atomic(&hdr_->fut_not_full)
                  ->load(std::memory_order_seq_cst) == snap)
            futex_wait(&hdr_->fut_not_full, snap, remaining);
        }
      }
      for (size_t i = 0; i < burst; ++i) {
        size_t len = lens[done + i];
        if (len) std::memcpy(data_ + recs[i] + kRecHdr, bufs[done + i], len);
        as_atomic(reinterpret_cast<uint32_t*>(data_ + recs[i] + 4))
            ->store(kStCommitted, std::memory_order_release);
      }
      done += burst;
      bump_and_wake(&hdr_->fut_not_empty, &hdr_->waiters_not_empty);
    }
    return done;
  }

  // Consumer-side helper: with the lock held, resolve the head record.
  // Returns: 0 = a committed record is ready (out params set);
  //          1 = ring empty; 2 = head reserved (writer mid-copy).
  int resolve_head(uint64_t* head_out, uint32_t* len_out,
                   uint32_t* flags_out) {
    if (hdr_->msg_count == 0) return 1;
    uint64_t cap = hdr_->capacity;
    uint64_t head = hdr_->head;
    uint32_t len32;
    std::memcpy(&len32, data_ + head, 4);
    if (len32 == kWrapMarker) {
      txn_arm(hdr_, 0, hdr_->tail, hdr_->used - (cap - head),
              hdr_->msg_count, hdr_->total_in, hdr_->total_out,
              hdr_->spill_count);
      txn_apply(hdr_);
      head = 0;
      std::memcpy(&len32, data_ + head, 4);
    }
    uint32_t state =
        as_atomic(reinterpret_cast<uint32_t*>(data_ + head + 4))
            ->load(std::memory_order_acquire);
    *head_out = head;
    *len_out = len32;
    std::memcpy(flags_out, data_ + head + 12, 4);
    return state == kStCommitted ? 0 : 2;
  }

  // With the lock held: copy a committed spill record's payload into
  // *out and consume it (the control record names the segment).
  void consume_spill(uint64_t head, uint32_t len32, std::string* out) {
    uint64_t real_len;
    std::string nm;
    if (!spill_ctrl_decode(data_ + head + kRecHdr, len32, &real_len, &nm)) {
      consume_record(head, len32, kFlagSpill);
      throw std::runtime_error("corrupt spill control record");
    }
    out->resize(real_len);
    try {
      spill_read(nm, real_len ? &(*out)[0] : nullptr, real_len);
    } catch (...) {
      consume_record(head, len32, kFlagSpill);
      throw;
    }
    consume_record(head, len32, kFlagSpill);
  }

  void consume_record(uint64_t head, uint32_t len32, uint32_t flags = 0) {
    size_t need = record_bytes(len32);
    uint64_t spill = hdr_->spill_count;
    if ((flags & kFlagSpill) && spill) spill -= 1;
    txn_arm(hdr_, (head + need) % hdr_->capacity, hdr_->tail,
            hdr_->used - need, hdr_->msg_count - 1, hdr_->total_in,
            hdr_->total_out + 1, spill);
    txn_apply(hdr_);
  }

  void reclaim_dead_record(uint64_t head, uint32_t len32, uint32_t flags) {
    size_t need = record_bytes(len32);
    // flags were written under the reserve lock, so they are valid even
    // though the payload is not: release the dead writer's spill budget
    uint64_t spill = hdr_->spill_count;
    if ((flags & kFlagSpill) && spill) spill -= 1;
    txn_arm(hdr_, (head + need) % hdr_->capacity, hdr_->tail,
            hdr_->used - need, hdr_->msg_count - 1, hdr_->total_in,
            hdr_->total_out, spill);
    txn_apply(hdr_);
  }

  // Shared blocking structure for recv / recv_into / peek.
  // op(head, len, flags) -> true when it consumed / is satisfied.
  template <typename Op>
  bool recv_loop(double timeout, Op&& op) {
    const double deadline =
        timeout > 0 ? monotonic_now() + timeout : 0.0;
    int reserved_streak = 0;
    int st = 1;
    for (;;) {
      uint32_t snap = as_atomic(&hdr_->fut_not_empty)
                          ->load(std::memory_order_acquire);
      {
        RobustLock lock(&hdr_->mu, hdr_);
        uint64_t head;
        uint32_t len32;
        uint32_t flags;
        st = resolve_head(&head, &len32, &flags);
        if (st == 0) {
          if (op(head, len32, flags)) {
            lock.unlock();
            bump_and_wake(&hdr_->fut_not_full, &hdr_->waiters_not_full);
            return true;
          }
          return false;  // op declined (recv_into: buffer too small)
        }
        if (st == 1 && hdr_->closed)
          throw std::runtime_error("ring closed");
        if (st == 2) {
          // writer mid-copy; reclaim if it died (incl. zombie)
          ++reserved_streak;
          if (reserved_streak > 3) {
            uint32_t pid;
            std::memcpy(&pid, data_ + head + 8, 4);
            if (!process_alive(pid)) {
              reclaim_dead_record(head, len32, flags);
              lock.unlock();
              bump_and_wake(&hdr_->fut_not_full,
                            &hdr_->waiters_not_full);
              reserved_streak = 0;
              continue;
            }
            reserved_streak = 0;
          }
        } else {
          reserved_streak = 0;
        }
        if (timeout == 0) return false;
      }
      // A reserved head means the writer is mid-copy: a LIVE writer's
      // commit bumps the futex and wakes us immediately, so capping this
      // wait at kReservedWait only slows the dead-writer path — it lets
      // reserved_streak reach the liveness check even when the caller
      // uses short per-call timeouts (< the 1 s wait chunk).
      double cap_wait = (st == 2) ? kReservedWait : 1.0;
      double remaining = cap_wait;
      if (timeout > 0) {
        remaining = deadline - monotonic_now();
        if (remaining <= 0) return false;
        if (remaining > cap_wait) remaining = cap_wait;
      }
      {
        WaiterScope ws(&hdr_->waiters_not_empty);
        if (as_atomic(&hdr_->fut_not_empty)
                ->load(std::memory_order_seq_cst) == snap)
          futex_wait(&hdr_->fut_not_empty, snap, remaining);
      }
    }
  }

  // Batched consumer: blocks (per `timeout`) for the FIRST message, then
  // drains up to max_n committed records under the same lock hold with a
  // single not-full wake.  Stops early at a reserved (mid-copy) record —
  // never blocks once something has been drained.
  size_t recv_many(std::vector<std::string>* out, size_t max_n,
                   double timeout) {
    if (max_n == 0) return 0;
    if (max_n > kMaxBatch) max_n = kMaxBatch;
    const double deadline = timeout > 0 ? monotonic_now() + timeout : 0.0;
    int reserved_streak = 0;
    int st = 1;
    for (;;) {
      uint32_t snap = as_atomic(&hdr_->fut_not_empty)
                          ->load(std::memory_order_acquire);
      {
        RobustLock lock(&hdr_->mu, hdr_);
        uint64_t head;
        uint32_t len32;
        uint32_t flags;
        st = resolve_head(&head, &len32, &flags);
        if (st == 0) {
          do {
            if (flags & kFlagSpill) {
              out->emplace_back();
              consume_spill(head, len32, &out->back());
            } else {
              out->emplace_back(data_ + head + kRecHdr, len32);
              consume_record(head, len32);
            }
          } while (out->size() < max_n &&
                   resolve_head(&head, &len32, &flags) == 0);
          lock.unlock();
          bump_and_wake(&hdr_->fut_not_full, &hdr_->waiters_not_full);
          return out->size();
        }
        if (st == 1 && hdr_->closed)
          throw std::runtime_error("ring closed");
        if (st == 2) {
          ++reserved_streak;
          if (reserved_streak > 3) {
            uint32_t pid;
            std::memcpy(&pid, data_ + head + 8, 4);
            if (!process_alive(pid)) {
              reclaim_dead_record(head, len32, flags);
              lock.unlock();
              bump_and_wake(&hdr_->fut_not_full,
                            &hdr_->waiters_not_full);
              reserved_streak = 0;
              continue;
            }
            reserved_streak = 0;
          }
        } else {
          reserved_streak = 0;
        }
        if (timeout == 0) return 0;
      }
      double cap_wait = (st == 2) ? kReservedWait : 1.0;  // see recv_loop
      double remaining = cap_wait;
      if (timeout > 0) {
        remaining = deadline - monotonic_now();
        if (remaining <= 0) return 0;
        if (remaining > cap_wait) remaining = cap_wait;
      }
      {
        WaiterScope ws(&hdr_->waiters_not_empty);
        if (as_atomic(&hdr_->fut_not_empty)
                ->load(std::memory_order_seq_cst) == snap)
          futex_wait(&hdr_->fut_not_empty, snap, remaining);
      }
    }
  }

  bool recv(std::string* out, double timeout) {
    return recv_loop(
        timeout, [&](uint64_t head, uint32_t len32, uint32_t flags) {
          if (flags & kFlagSpill) {
            consume_spill(head, len32, out);
            return true;
          }
          out->assign(data_ + head + kRecHdr, len32);
          consume_record(head, len32);
          return true;
        });
  }

  // Single-copy receive.  Returns len, -1 on timeout, or -(len)-2 if the
  // buffer is too small (message left in place).
  int64_t recv_into(char* buf, size_t buflen, double timeout) {
    int64_t result = -1;
    bool ok = recv_loop(
        timeout, [&](uint64_t head, uint32_t len32, uint32_t flags) {
          if (flags & kFlagSpill) {
            uint64_t real_len;
            std::string nm;
            if (!spill_ctrl_decode(data_ + head + kRecHdr, len32,
                                   &real_len, &nm)) {
              consume_record(head, len32);
              throw std::runtime_error("corrupt spill control record");
            }
            if (real_len > buflen) {
              result = -((int64_t)real_len) - 2;
              return false;  // left in place (incl. the segment)
            }
            try {
              spill_read(nm, buf, real_len);
            } catch (...) {
              consume_record(head, len32, kFlagSpill);
              throw;
            }
            consume_record(head, len32, kFlagSpill);
            result = (int64_t)real_len;
            return true;
          }
          if ((size_t)len32 > buflen) {
            result = -((int64_t)len32) - 2;
            return false;
          }
          std::memcpy(buf, data_ + head + kRecHdr, len32);
          consume_record(head, len32);
          result = (int64_t)len32;
          return true;
        });
    (void)ok;
    return result;
  }

  // Size of the next committed message, or -1 on timeout.  No consume.
  int64_t peek_size(double timeout) {
    int64_t result = -1;
    recv_loop(timeout,
              [&](uint64_t head, uint32_t len32, uint32_t flags) {
                if (flags & kFlagSpill) {
                  uint64_t real_len;
                  std::string nm;
                  if (spill_ctrl_decode(data_ + head + kRecHdr, len32,
                                        &real_len, &nm))
                    result = (int64_t)real_len;
                  else
                    result = (int64_t)len32;
                  return true;
                }
                result = (int64_t)len32;
                return true;
              });
    return result;
  }

  void close_ring() {
    if (!hdr_) return;
    // Time-boxed lock: teardown must never wedge.  The closed flag is
    // checked inside every wait loop, so a lock-free store + wake still
    // drains all waiters even if the mutex is stuck.
    struct timespec deadline;
    clock_gettime(CLOCK_REALTIME, &deadline);
    deadline.tv_sec += 5;
    int rc = pthread_mutex_timedlock(&hdr_->mu, &deadline);
    if (rc == EOWNERDEAD) {
      txn_recover(hdr_);
      pthread_mutex_consistent(&hdr_->mu);
      rc = 0;
    }
    as_atomic(&hdr_->closed)->store(1, std::memory_order_release);
    if (rc == 0) pthread_mutex_unlock(&hdr_->mu);
    as_atomic(&hdr_->fut_not_empty)
        ->fetch_add(1, std::memory_order_seq_cst);
    as_atomic(&hdr_->fut_not_full)->fetch_add(1, std::memory_order_seq_cst);
    futex_wake_all(&hdr_->fut_not_empty);  // unconditional on teardown
    futex_wake_all(&hdr_->fut_not_full);
  }

  void unlink_ring() {
    shm_unlink(name_.c_str());
    spill_sweep(name_);  // unread / leaked spill segments die with the ring
  }

  uint64_t size() const { return hdr_ ? hdr_->msg_count : 0; }
  uint64_t total_in() const { return hdr_ ? hdr_->total_in : 0; }
  uint64_t total_out() const { return hdr_ ? hdr_->total_out : 0; }
  bool closed() const { return hdr_ ? hdr_->closed != 0 : true; }
  bool is_owner() const { return owner_; }
  const std::string& name() const { return name_; }

 private:
  static constexpr double kReservedWait = 0.1;
  std::string name_;
  bool owner_;
  Header* hdr_ = nullptr;
  char* data_ = nullptr;
  size_t map_len_ = 0;
};

}  // namespace

#ifndef FAM_NO_PYBIND
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
          "send_many",
          [](ShmRing& r, py::sequence msgs, double timeout) {
            std::vector<py::buffer_info> infos;  // keeps buffers alive
            std::vector<const char*> ptrs;
            std::vector<size_t> lens;
            size_t n = (size_t)py::len(msgs);
            infos.reserve(n);
            ptrs.reserve(n);
            lens.reserve(n);
            for (auto item : msgs) {
              infos.push_back(py::buffer(item.cast<py::object>()).request());
              ptrs.push_back(static_cast<const char*>(infos.back().ptr));
              lens.push_back((size_t)infos.back().size *
                             (size_t)infos.back().itemsize);
            }
            size_t sent;
            {
              py::gil_scoped_release release;
              sent = r.send_many(ptrs.data(), lens.data(), n, timeout);
            }
            return sent;
          },
          py::arg("msgs"), py::arg("timeout") = -1.0)
      .def(
          "recv_many",
          [](ShmRing& r, size_t max_n, double timeout) {
            std::vector<std::string> out;
            {
              py::gil_scoped_release release;
              r.recv_many(&out, max_n, timeout);
            }
            py::list result;
            for (auto& s : out) result.append(py::bytes(s));
            return result;
          },
          py::arg("max_n") = 64, py::arg("timeout") = -1.0)
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
#endif  // FAM_NO_PYBIND
