// Core of the native filter-tree router — pure C++ (no Python), so the
// sanitizer harness (router_core_test.cpp, `make test-native-debug`) can
// exercise the exact production logic under ASan/UBSan.  Semantics are the
// reference's scheduling/filter.go tree, kept EXACTLY equal to the Python
// tree in ../filters.py (tests/test_native_router.py proves survivor-set
// equivalence on random fleets); router_native.cpp is the pybind11 wrapper.
#pragma once

#include <algorithm>
#include <cassert>
#include <cstdint>
#include <limits>
#include <random>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace lig {

struct Engine {
  std::string name;
  std::string pool;
  int slot_base = -1;  // >=0: gauges live in the shared buffer
  std::unordered_set<std::string> adapters;
  int max_active = 0;
  // fallback gauges (slot_base < 0)
  double waiting = 0, running = 0, kv = 0;
  bool draining = false;
};

struct Gauges {
  double waiting, kv;
  bool draining;
};

class RouterCore {
 public:
  explicit RouterCore(uint64_t seed) : rng_(seed) {}

  void attach_shm(const double* ptr, size_t len, int slots_per_engine,
                  int slot_waiting, int slot_kv, int slot_draining) {
    shm_ = ptr;
    shm_len_ = len;
    spe_ = slots_per_engine;
    s_wait_ = slot_waiting;
    s_kv_ = slot_kv;
    s_drain_ = slot_draining;
  }

  void set_thresholds(double kv_thresh, int queue_crit, int queue_lora) {
    kv_thresh_ = kv_thresh;
    q_crit_ = queue_crit;
    q_lora_ = queue_lora;
  }

  void upsert(const std::string& name, const std::string& pool, int slot_base,
              const std::vector<std::string>& adapters, int max_active) {
    auto it = index_.find(name);
    Engine* e;
    if (it == index_.end()) {
      engines_.emplace_back();
      index_[name] = engines_.size() - 1;
      e = &engines_.back();
      e->name = name;
    } else {
      e = &engines_[it->second];
    }
    e->pool = pool;
    e->slot_base = slot_base;
    e->adapters =
        std::unordered_set<std::string>(adapters.begin(), adapters.end());
    e->max_active = max_active;
  }

  void set_gauges(const std::string& name, double waiting, double running,
                  double kv, bool draining) {
    Engine& e = engines_.at(index_.at(name));
    e.waiting = waiting;
    e.running = running;
    e.kv = kv;
    e.draining = draining;
  }

  void remove(const std::string& name) {
    auto it = index_.find(name);
    if (it == index_.end()) return;
    size_t idx = it->second;
    index_.erase(it);
    if (idx != engines_.size() - 1) {
      engines_[idx] = std::move(engines_.back());
      index_[engines_[idx].name] = idx;
    }
    engines_.pop_back();
  }

  size_t size() const { return engines_.size(); }

  // survivor names after the full tree (test/debug surface)
  std::vector<std::string> survivors(const std::string& resolved,
                                     bool critical,
                                     const std::string& pool_ref) {
    std::vector<int> out = run_tree(resolved, critical, pool_ref);
    std::vector<std::string> names;
    names.reserve(out.size());
    for (int i : out) names.push_back(engines_[i].name);
    return names;
  }

  // "" == drop (ResourceExhausted); throws std::runtime_error when no engine
  // is routable (SchedulingError upstream)
  std::string schedule(const std::string& resolved, bool critical,
                       const std::string& pool_ref) {
    std::vector<int> out = run_tree(resolved, critical, pool_ref);
    if (dropped_) return "";
    if (out.empty()) throw std::runtime_error("filter produced 0 engines");
    std::uniform_int_distribution<size_t> pick(0, out.size() - 1);
    return engines_[out[pick(rng_)]].name;
  }

 private:
  Gauges gauges(const Engine& e) const {
    if (e.slot_base >= 0 && shm_ != nullptr &&
        static_cast<size_t>(e.slot_base) + spe_ <= shm_len_) {
      return Gauges{shm_[e.slot_base + s_wait_], shm_[e.slot_base + s_kv_],
                    shm_[e.slot_base + s_drain_] > 0.5};
    }
    return Gauges{e.waiting, e.kv, e.draining};
  }

  // --- band filters (filter.go:102-154 semantics, incl. int vs float math)
  std::vector<int> least_queuing(const std::vector<int>& in) const {
    long lo = std::numeric_limits<long>::max(),
         hi = std::numeric_limits<long>::min();
    for (int i : in) {
      long w = static_cast<long>(gauges(engines_[i]).waiting);
      lo = std::min(lo, w);
      hi = std::max(hi, w);
    }
    long band = lo + (hi - lo) / static_cast<long>(in.size());
    std::vector<int> out;
    for (int i : in) {
      long w = static_cast<long>(gauges(engines_[i]).waiting);
      if (w >= lo && w <= band) out.push_back(i);
    }
    return out;
  }

  std::vector<int> least_kv(const std::vector<int>& in) const {
    double lo = std::numeric_limits<double>::infinity(), hi = -lo;
    for (int i : in) {
      double k = gauges(engines_[i]).kv;
      lo = std::min(lo, k);
      hi = std::max(hi, k);
    }
    double band = lo + (hi - lo) / static_cast<double>(in.size());
    std::vector<int> out;
    for (int i : in) {
      double k = gauges(engines_[i]).kv;
      if (k >= lo && k <= band) out.push_back(i);
    }
    return out;
  }

  bool affinity(const Engine& e, const std::string& resolved) const {
    return e.adapters.count(resolved) > 0;
  }
  bool can_accept(const Engine& e) const {
    return static_cast<int>(e.adapters.size()) < e.max_active;
  }
  bool low_lora_cost(const Engine& e, const std::string& resolved) const {
    return affinity(e, resolved) || can_accept(e);
  }

  // queue_lora_kv chain: leastQueuing -> lowLoRACost(either) -> leastKV
  std::vector<int> queue_lora_kv(std::vector<int> in,
                                 const std::string& resolved) const {
    in = least_queuing(in);
    std::vector<int> lc;
    for (int i : in)
      if (low_lora_cost(engines_[i], resolved)) lc.push_back(i);
    // predicate filters pass the ORIGINAL set on failure (filter.go:66-72)
    return least_kv(lc.empty() ? in : lc);
  }

  std::vector<int> run_tree(const std::string& resolved, bool critical,
                            const std::string& pool_ref) {
    dropped_ = false;
    std::vector<int> all;
    all.reserve(engines_.size());
    for (size_t i = 0; i < engines_.size(); ++i) {
      if (gauges(engines_[i]).draining) continue;
      all.push_back(static_cast<int>(i));
    }
    // mixed-pool scoping (router/scheduler.py extension): tagged engines
    // qualify for their pool; untagged always; empty match falls back
    if (!pool_ref.empty()) {
      std::vector<int> scoped;
      for (int i : all) {
        const std::string& p = engines_[i].pool;
        if (p.empty() || p == pool_ref) scoped.push_back(i);
      }
      if (!scoped.empty()) all = std::move(scoped);
    }
    if (all.empty()) return all;

    if (critical) {
      std::vector<int> lowq;
      for (int i : all)
        if (static_cast<long>(gauges(engines_[i]).waiting) < q_lora_)
          lowq.push_back(i);
      if (!lowq.empty()) {
        std::vector<int> aff;
        for (int i : lowq)
          if (affinity(engines_[i], resolved)) aff.push_back(i);
        if (!aff.empty()) return least_kv(least_queuing(aff));
        std::vector<int> acc;
        for (int i : lowq)
          if (can_accept(engines_[i])) acc.push_back(i);
        // canAcceptNewLora: either-branch (original set on failure)
        return least_kv(least_queuing(acc.empty() ? lowq : acc));
      }
      return queue_lora_kv(all, resolved);
    }
    // sheddable: the headroom predicate's SUCCESS branch receives the
    // FILTERED set (engines with queue <= q_crit and kv <= threshold);
    // empty -> drop (scheduler.go:74-90, filter.go success semantics)
    std::vector<int> headroom;
    for (int i : all) {
      Gauges g = gauges(engines_[i]);
      if (static_cast<long>(g.waiting) <= q_crit_ && g.kv <= kv_thresh_)
        headroom.push_back(i);
    }
    if (headroom.empty()) {
      dropped_ = true;
      return {};
    }
    return queue_lora_kv(headroom, resolved);
  }

  std::vector<Engine> engines_;
  std::unordered_map<std::string, size_t> index_;
  const double* shm_ = nullptr;
  size_t shm_len_ = 0;
  int spe_ = 0, s_wait_ = 2, s_kv_ = 4, s_drain_ = 7;
  double kv_thresh_ = 0.8;
  int q_crit_ = 5, q_lora_ = 50;
  bool dropped_ = false;
  std::mt19937_64 rng_;
};

}  // namespace lig
