// Sanitizer harness for the native router core (`make test-native-debug`):
// the production routing logic (router_core.h) under ASan + UBSan —
// deterministic semantic checks plus a randomized churn fuzz (upsert /
// remove / gauge flips / schedule) that would surface OOB reads on the
// shared gauge buffer, use-after-move in the swap-remove path, and UB in
// the band math.  The reference ships no race/sanitizer CI;
// this closes that gap for the C++ side.
//
// Build+run: make test-native-debug
#include <cstdio>
#include <cstdlib>

#include "router_core.h"

#define CHECK(cond)                                                    \
  do {                                                                 \
    if (!(cond)) {                                                     \
      std::fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__,    \
                   __LINE__, #cond);                                   \
      std::exit(1);                                                    \
    }                                                                  \
  } while (0)

using lig::RouterCore;

static void test_semantics() {
  RouterCore r(42);
  r.set_thresholds(0.8, 5, 50);
  // three engines: e0 idle with affinity, e1 idle, e2 overloaded
  r.upsert("e0", "", -1, {"lora-a"}, 4);
  r.upsert("e1", "", -1, {}, 4);
  r.upsert("e2", "", -1, {}, 4);
  r.set_gauges("e0", 0, 0, 0.1, false);
  r.set_gauges("e1", 0, 0, 0.1, false);
  r.set_gauges("e2", 60, 8, 0.95, false);
  // critical + affinity -> e0 only
  auto s = r.survivors("lora-a", true, "");
  CHECK(s.size() == 1 && s[0] == "e0");
  // sheddable with headroom -> routed (non-empty)
  CHECK(!r.schedule("lora-a", false, "").empty());
  // saturate everything -> sheddable drops ("" = ResourceExhausted)
  for (const char* n : {"e0", "e1", "e2"}) r.set_gauges(n, 60, 8, 0.95, false);
  CHECK(r.schedule("x", false, "") == "");
  // critical still routes under saturation
  CHECK(!r.schedule("x", true, "").empty());
  // draining engine leaves the set
  r.set_gauges("e0", 0, 0, 0.1, false);
  r.set_gauges("e1", 0, 0, 0.1, true);
  r.set_gauges("e2", 0, 0, 0.1, false);
  for (int i = 0; i < 50; ++i) CHECK(r.schedule("x", true, "") != "e1");
  // every engine draining -> throws (SchedulingError upstream)
  for (const char* n : {"e0", "e2"}) r.set_gauges(n, 0, 0, 0.1, true);
  bool threw = false;
  try {
    r.schedule("x", true, "");
  } catch (const std::runtime_error&) {
    threw = true;
  }
  CHECK(threw);
}

static void test_shm_gauges() {
  // gauge slots per serving/pool.py layout (9 doubles per engine)
  constexpr int SPE = 9, W = 2, KV = 4, DR = 7;
  std::vector<double> shm(2 * SPE, 0.0);
  RouterCore r(7);
  r.attach_shm(shm.data(), shm.size(), SPE, W, KV, DR);
  r.set_thresholds(0.8, 5, 50);
  r.upsert("g0", "", 0, {}, 4);
  r.upsert("g1", "", SPE, {}, 4);
  shm[0 * SPE + W] = 0;   // g0 idle
  shm[1 * SPE + W] = 40;  // g1 busy
  shm[0 * SPE + KV] = 0.1;
  shm[1 * SPE + KV] = 0.5;
  for (int i = 0; i < 50; ++i) CHECK(r.schedule("x", true, "") == "g0");
  // drain through shm
  shm[0 * SPE + DR] = 1.0;
  for (int i = 0; i < 50; ++i) CHECK(r.schedule("x", true, "") == "g1");
  // slot_base past the buffer end must fall back, not read OOB
  r.upsert("g2", "", 64 * SPE, {}, 4);
  r.set_gauges("g2", 0, 0, 0.0, false);
  for (int i = 0; i < 50; ++i) CHECK(!r.schedule("x", true, "").empty());
}

static void test_churn_fuzz() {
  std::mt19937_64 rng(1234);
  RouterCore r(99);
  r.set_thresholds(0.8, 5, 50);
  constexpr int SPE = 9;
  std::vector<double> shm(64 * SPE, 0.0);
  r.attach_shm(shm.data(), shm.size(), SPE, 2, 4, 7);
  std::vector<std::string> pool_names = {"", "p1", "p2"};
  int added = 0;
  for (int it = 0; it < 20000; ++it) {
    double dice = std::uniform_real_distribution<>(0, 1)(rng);
    std::string name = "e" + std::to_string(rng() % 48);
    if (dice < 0.30) {
      std::vector<std::string> ad;
      for (unsigned a = 0; a < rng() % 4; ++a)
        ad.push_back("lora-" + std::to_string(rng() % 8));
      int slot_base = (rng() % 3 == 0) ? -1 : (int)(rng() % 70) * SPE;
      r.upsert(name, pool_names[rng() % 3], slot_base, ad, 4);
      ++added;
    } else if (dice < 0.45) {
      r.remove(name);
    } else if (dice < 0.70) {
      size_t base = (rng() % 64) * SPE;
      shm[base + 2] = (double)(rng() % 80);
      shm[base + 4] = std::uniform_real_distribution<>(0, 1)(rng);
      shm[base + 7] = (rng() % 8 == 0) ? 1.0 : 0.0;
    } else if (r.size() > 0) {
      try {
        r.schedule("lora-" + std::to_string(rng() % 8), rng() % 2 == 0,
                   pool_names[rng() % 3]);
      } catch (const std::runtime_error&) {
        // empty routable set: fine
      }
      try {
        r.survivors("lora-" + std::to_string(rng() % 8), rng() % 2 == 0, "");
      } catch (const std::runtime_error&) {
      }
    }
  }
  CHECK(added > 1000);
}

int main() {
  test_semantics();
  test_shm_gauges();
  test_churn_fuzz();
  std::printf("router_core sanitizer tests OK\n");
  return 0;
}
