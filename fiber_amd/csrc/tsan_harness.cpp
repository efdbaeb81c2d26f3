// ThreadSanitizer harness for the shm-ring transport engine.
//
// Race detection for the C++ data plane (SURVEY §5 "race detection"):
// the cross-PROCESS protocol runs the exact same code paths when driven
// from multiple THREADS of one process, which is where TSAN can see it.
// The harness hammers one ring with concurrent producers (single +
// batched sends, spill-sized payloads included) and consumers (recv,
// recv_many, recv_into, peek) and verifies message integrity by
// checksum.  The synchronization contract TSAN checks:
//   * head/tail/used/msg_count only mutate under the robust mutex;
//   * payload bytes are written OUTSIDE the lock but ordered by the
//     record's atomic state word (release commit / acquire resolve);
//   * futex sequence words + waiter counters are seq_cst atomics.
//
// Build + run (recorded in profiles/r02_tsan.md):
//   g++ -std=c++17 -O1 -g -fsanitize=thread -DFAM_NO_PYBIND \
//       fiber_amd/csrc/tsan_harness.cpp -o /tmp/fam_tsan -lpthread -lrt
//   /tmp/fam_tsan
#define FAM_NO_PYBIND
#include "transport.cpp"

#include <atomic>
#include <cassert>
#include <cstdlib>
#include <thread>
#include <vector>

namespace {

uint64_t fnv1a(const char* p, size_t n) {
  uint64_t h = 1469598103934665603ull;
  for (size_t i = 0; i < n; ++i) {
    h ^= (unsigned char)p[i];
    h *= 1099511628211ull;
  }
  return h;
}

}  // namespace

int main() {
  const char* name = "fam-tsan-ring";
  const int kProducers = 4, kConsumers = 4;
  const int kPerProducer = 1500;
  ShmRing ring(name, true, 64 << 10, 5.0);

  std::atomic<long> consumed{0}, produced{0};
  std::atomic<bool> corrupt{false};

  auto make_msg = [&](unsigned seed, std::string* out) {
    // sizes sweep inline records, wrap geometry and the spill path
    size_t len = (seed * 2654435761u) % 40000;  // up to ~40 KB (> cap/4)
    out->resize(8 + len);
    for (size_t i = 0; i < len; ++i)
      (*out)[8 + i] = (char)((seed + i * 131) & 0xFF);
    uint64_t h = fnv1a(out->data() + 8, len);
    std::memcpy(&(*out)[0], &h, 8);
  };

  std::vector<std::thread> threads;
  for (int p = 0; p < kProducers; ++p) {
    threads.emplace_back([&, p] {
      std::string msg;
      for (int i = 0; i < kPerProducer; ++i) {
        unsigned seed = (unsigned)(p * 100003 + i);
        if (i % 7 == 0) {
          // batched path
          std::string a, b;
          make_msg(seed, &a);
          make_msg(seed + 1, &b);
          const char* bufs[2] = {a.data(), b.data()};
          size_t lens[2] = {a.size(), b.size()};
          // blocking sends: under TSAN's slowdown + spill backpressure a
          // finite timeout could drop messages and wedge the consumers'
          // consumed==total exit condition
          size_t sent = ring.send_many(bufs, lens, 2, -1.0);
          produced += (long)sent;
          ++i;  // consumed two seeds
        } else {
          make_msg(seed, &msg);
          if (ring.send(msg.data(), msg.size(), -1.0)) ++produced;
        }
      }
    });
  }

  const long kTotal = (long)kProducers * kPerProducer;
  for (int c = 0; c < kConsumers; ++c) {
    threads.emplace_back([&, c] {
      std::string out;
      std::vector<std::string> batch;
      std::vector<char> buf(64 << 10);
      while (consumed.load() < kTotal) {
        bool got = false;
        if (c == 0) {
          batch.clear();
          ring.recv_many(&batch, 16, 0.2);
          for (auto& m : batch) {
            uint64_t h;
            std::memcpy(&h, m.data(), 8);
            if (h != fnv1a(m.data() + 8, m.size() - 8)) corrupt = true;
            ++consumed;
          }
          got = !batch.empty();
        } else if (c == 1) {
          int64_t n = ring.recv_into(buf.data(), buf.size(), 0.2);
          if (n >= 0) {
            uint64_t h;
            std::memcpy(&h, buf.data(), 8);
            if (h != fnv1a(buf.data() + 8, (size_t)n - 8)) corrupt = true;
            ++consumed;
            got = true;
          } else if (n < -1) {
            buf.resize((size_t)(-n - 2));
          }
        } else {
          (void)ring.peek_size(0.0);
          if (ring.recv(&out, 0.2)) {
            uint64_t h;
            std::memcpy(&h, out.data(), 8);
            if (h != fnv1a(out.data() + 8, out.size() - 8)) corrupt = true;
            ++consumed;
            got = true;
          }
        }
        (void)got;
      }
    });
  }

  for (auto& t : threads) t.join();
  ring.close_ring();
  ring.unlink_ring();

  std::printf("produced=%ld consumed=%ld corrupt=%d\n", produced.load(),
              consumed.load(), (int)corrupt.load());
  if (corrupt.load() || consumed.load() != kTotal) return 1;
  return 0;
}
