// ThreadSanitizer re-statement of the two shared-memory protocols
// (SURVEY.md §5.2's promised race-detection pass):
//
//   1. the SPSC trajectory ring   — parallel/queue.py  TrajectoryRing
//   2. the seqlock weight blob    — parallel/weights.py WeightPublisher/
//                                                        WeightSubscriber
//
// The Python implementations use PLAIN numpy u64 stores and rely on x86
// TSO for publication ordering (stores retire in program order; loads are
// not reordered with older loads). This file maps that TSO-carried
// ordering onto the equivalent C++ memory_order so ThreadSanitizer can
// check the CLAIM: given publication ordering (free on x86), the
// protocols contain no other data race — no slot-reuse overlap, no
// header/payload aliasing, no torn snapshot accepted by the seqlock
// retry loop.
//
//   Python (TSO)                      C++ model
//   ---------------------------------------------------------------
//   tail store after payload writes   tail.store(release)
//   tail load before payload reads    tail.load(acquire)
//   head store after payload reads    head.store(release)
//   head load before payload writes   head.load(acquire)
//   version odd/even plain stores     fence-based seqlock (seq_cst
//                                     fences; payload elements relaxed
//                                     atomics — the by-design torn
//                                     window is detected and DISCARDED
//                                     by the v0==v1 retry, so those
//                                     value races are not bugs)
//
// Build & run (scripts/tsan_ring.sh):
//   g++ -std=c++17 -O1 -g -fsanitize=thread tests/tsan/ring_tsan.cc \
//       -o /tmp/ring_tsan -lpthread && /tmp/ring_tsan
// TSan exits non-zero on any report; the checksum asserts exit non-zero
// on any protocol (value-consistency) violation.

#include <atomic>
#include <cassert>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <chrono>
#include <thread>
#include <vector>

// ---------------------------------------------------------------- ring --

struct Ring {
  // mirrors TrajectoryRing: capacity slots of fixed layout, u64 tail/head
  // in the header, heartbeat double (producer-only write, consumer read —
  // monotonic timestamp, value races benign but modeled relaxed-atomic)
  static constexpr int CAP = 8;          // queue.py capacity (queue_size)
  static constexpr int SLOT = 256;       // u32 words per slot
  std::atomic<uint64_t> tail{0}, head{0};
  std::atomic<double> heartbeat{0.0};
  uint32_t payload[CAP][SLOT];           // plain memory, like the shm buf

  bool try_push(const uint32_t* src) {
    const uint64_t t = tail.load(std::memory_order_relaxed);  // own field
    const uint64_t h = head.load(std::memory_order_acquire);  // slot free?
    if (t - h >= CAP) return false;
    std::memcpy(payload[t % CAP], src, SLOT * sizeof(uint32_t));
    heartbeat.store(static_cast<double>(t), std::memory_order_relaxed);
    tail.store(t + 1, std::memory_order_release);             // publish
    return true;
  }

  bool try_pop(uint32_t* dst) {
    const uint64_t h = head.load(std::memory_order_relaxed);  // own field
    const uint64_t t = tail.load(std::memory_order_acquire);  // published?
    if (h >= t) return false;
    std::memcpy(dst, payload[h % CAP], SLOT * sizeof(uint32_t));
    head.store(h + 1, std::memory_order_release);             // slot free
    return true;
  }
};

static uint32_t slot_checksum(const uint32_t* w) {
  uint32_t x = 2166136261u;
  for (int i = 0; i + 1 < Ring::SLOT; ++i) x = (x ^ w[i]) * 16777619u;
  return x;
}

static void ring_producer(Ring* r, int unrolls, uint32_t seed) {
  uint32_t slot[Ring::SLOT];
  for (int u = 0; u < unrolls; ++u) {
    for (int i = 0; i + 1 < Ring::SLOT; ++i)
      slot[i] = seed * 2654435761u + u * 97u + i;
    slot[Ring::SLOT - 1] = slot_checksum(slot);
    while (!r->try_push(slot)) std::this_thread::yield();
  }
}

static void ring_consumer(Ring* r, int unrolls, std::atomic<int>* bad) {
  uint32_t slot[Ring::SLOT];
  for (int u = 0; u < unrolls; ++u) {
    while (!r->try_pop(slot)) std::this_thread::yield();
    if (slot[Ring::SLOT - 1] != slot_checksum(slot)) bad->fetch_add(1);
  }
}

// ------------------------------------------------------------- seqlock --

struct Seqlock {
  static constexpr int N = 4096;  // u32 words of weight blob
  std::atomic<uint64_t> version{0};
  std::atomic<uint64_t> global_step{0};
  // numpy's memcpy never tears a 4-byte element at the value level; the
  // relaxed atomic elements model that while keeping TSan focused on the
  // snapshot-acceptance logic rather than the by-design torn window
  std::atomic<uint32_t> payload[N];

  void publish(uint64_t v, uint32_t seed, uint64_t step) {
    version.store(v + 1, std::memory_order_relaxed);          // odd
    std::atomic_thread_fence(std::memory_order_seq_cst);
    global_step.store(step, std::memory_order_relaxed);
    for (int i = 0; i < N; ++i)
      payload[i].store(seed + static_cast<uint32_t>(i),
                       std::memory_order_relaxed);
    std::atomic_thread_fence(std::memory_order_seq_cst);
    version.store(v + 2, std::memory_order_relaxed);          // even
  }

  // returns 0 on no-new/unstable, else the accepted version
  uint64_t pull(uint32_t* out, uint64_t last) {
    const uint64_t v0 = version.load(std::memory_order_relaxed);
    if (v0 == last || v0 == 0 || (v0 & 1)) return 0;
    std::atomic_thread_fence(std::memory_order_seq_cst);
    for (int i = 0; i < N; ++i)
      out[i] = payload[i].load(std::memory_order_relaxed);
    std::atomic_thread_fence(std::memory_order_seq_cst);
    const uint64_t v1 = version.load(std::memory_order_relaxed);
    return (v0 == v1) ? v0 : 0;
  }
};

static void seq_writer(Seqlock* s, int publishes) {
  for (int p = 0; p < publishes; ++p) {
    s->publish(2ull * p, 0x9e3779b9u * (p + 1), p + 1);
    // let readers land inside stable windows (the learner publishes every
    // publish_every steps, not back-to-back) — more accepted snapshots
    // means more write/read interleavings actually exercised
    if ((p & 7) == 0)
      std::this_thread::sleep_for(std::chrono::microseconds(50));
  }
}

static void seq_reader(Seqlock* s, int publishes, std::atomic<int>* bad,
                       std::atomic<int>* accepted) {
  std::vector<uint32_t> snap(Seqlock::N);
  uint64_t last = 0;
  while (last < 2ull * publishes) {
    const uint64_t v = s->pull(snap.data(), last);
    if (!v) { std::this_thread::yield(); continue; }
    // every accepted snapshot must be internally consistent: all elements
    // from the SAME publish (seed + i with seed derived from v)
    const uint32_t seed = 0x9e3779b9u * static_cast<uint32_t>(v / 2);
    for (int i = 0; i < Seqlock::N; ++i)
      if (snap[i] != seed + static_cast<uint32_t>(i)) {
        bad->fetch_add(1);
        break;
      }
    accepted->fetch_add(1);
    last = v;
  }
}

// ---------------------------------------------------------------- main --

int main() {
  constexpr int UNROLLS = 20000, PRODUCERS = 4;
  std::vector<Ring> rings(PRODUCERS);
  std::atomic<int> bad{0};
  {
    // the production topology: one ring per actor (SPSC), one learner
    // draining all of them — modeled as one consumer thread per ring to
    // maximize interleavings (round-robin drain adds no extra sharing)
    std::vector<std::thread> ts;
    for (int p = 0; p < PRODUCERS; ++p) {
      ts.emplace_back(ring_producer, &rings[p], UNROLLS, p + 1);
      ts.emplace_back(ring_consumer, &rings[p], UNROLLS, &bad);
    }
    for (auto& t : ts) t.join();
  }
  if (bad.load()) {
    std::fprintf(stderr, "ring: %d corrupted slots\n", bad.load());
    return 1;
  }

  constexpr int PUBLISHES = 10000, READERS = 3;
  auto* s = new Seqlock();
  std::atomic<int> torn{0}, accepted{0};
  {
    std::vector<std::thread> ts;
    ts.emplace_back(seq_writer, s, PUBLISHES);
    for (int r = 0; r < READERS; ++r)
      ts.emplace_back(seq_reader, s, PUBLISHES, &torn, &accepted);
    for (auto& t : ts) t.join();
  }
  if (torn.load()) {
    std::fprintf(stderr, "seqlock: %d torn snapshots accepted\n",
                 torn.load());
    return 1;
  }
  std::printf("ring_tsan OK: %d slots x %d rings, %d seqlock snapshots "
              "accepted torn-free\n",
              UNROLLS, PRODUCERS, accepted.load());
  delete s;
  return 0;
}
