// GPU embedding-cache index: LRU / LFU / LFUOpt slot management.
//
// Re-creation of the HET (VLDB'22) client-side embedding caches the
// reference ships in hetu/v1/src/hetu_cache/include/{cache.h,lru_cache.h,
// lfu_cache.h}: a fixed-capacity cache of embedding rows held on the GPU,
// with the *index* (id -> slot, recency/frequency bookkeeping, versions for
// bounded-staleness sync) maintained host-side in C++.  The row storage
// itself is a plain [capacity, dim] torch tensor owned by Python
// (hetu_amd/ps/cache.py) so lookups become a single GPU gather.
//
// Policies:
//   lru    — evict least-recently-used slot
//   lfu    — evict least-frequently-used (ties: oldest)
//   lfuopt — LFU with admission control: a new id is admitted only if its
//            running frequency beats the cache's current minimum (the
//            "opt" variant of HET); rejected ids get slot -1 and the
//            caller serves them straight from the pulled rows.
//
// All batch ops take/return int64 torch CPU tensors so the Python side can
// move them to GPU once per step.  Complexity: O(log n) per touched id via
// an ordered eviction set.
#include <torch/extension.h>

#include <cstdint>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

namespace {

struct EmbedCache {
  enum Policy { LRU, LFU, LFUOPT };

  explicit EmbedCache(int64_t capacity, const std::string& policy)
      : capacity_(capacity) {
    if (policy == "lru") policy_ = LRU;
    else if (policy == "lfu") policy_ = LFU;
    else if (policy == "lfuopt") policy_ = LFUOPT;
    else TORCH_CHECK(false, "policy must be lru|lfu|lfuopt, got ", policy);
    slot_id_.assign(capacity_, -1);
    slot_key_.assign(capacity_, 0);
    slot_freq_.assign(capacity_, 0);
    slot_version_.assign(capacity_, 0);
    slot_batch_.assign(capacity_, -1);
    for (int64_t s = capacity_ - 1; s >= 0; --s) free_.push_back(s);
  }

  // priority key for the eviction order of a slot
  int64_t prio(int64_t slot) const {
    return policy_ == LRU ? slot_key_[slot] : slot_freq_[slot];
  }

  void touch(int64_t slot) {
    order_.erase({prio(slot), slot});
    ++tick_;
    if (policy_ == LRU) slot_key_[slot] = tick_;
    else ++slot_freq_[slot];
    order_.insert({prio(slot), slot});
  }

  // query(ids) -> (slots, hit_mask): slot of each id or -1; hits are
  // touched (recency/frequency update).  ids should be unique.  Starts a
  // new batch: every slot returned by this query or the following admit()
  // is pinned (not evictable) until the next query — otherwise a batch
  // whose unique-id count exceeds capacity would alias slots it just
  // handed out.
  std::pair<torch::Tensor, torch::Tensor> query(torch::Tensor ids) {
    ++batch_;
    TORCH_CHECK(ids.device().is_cpu() && ids.dtype() == torch::kInt64);
    auto idsc = ids.contiguous();
    int64_t n = idsc.numel();
    auto slots = torch::empty({n}, torch::kInt64);
    auto hit = torch::empty({n}, torch::kBool);
    const int64_t* ip = idsc.data_ptr<int64_t>();
    int64_t* sp = slots.data_ptr<int64_t>();
    bool* hp = hit.data_ptr<bool>();
    for (int64_t i = 0; i < n; ++i) {
      auto it = map_.find(ip[i]);
      if (it == map_.end()) {
        sp[i] = -1;
        hp[i] = false;
        if (policy_ == LFUOPT) ++ghost_freq_[ip[i]];  // admission stats
      } else {
        sp[i] = it->second;
        hp[i] = true;
        touch(it->second);
        slot_batch_[it->second] = batch_;
        ++hits_;
      }
      ++lookups_;
    }
    return {slots, hit};
  }

  // admit(ids) -> (slots, evicted_ids, evicted_slots): give each missed id
  // a slot, evicting per policy.  LFUOpt may refuse (slot -1).  The caller
  // must write the pulled rows into storage[slots] and flush/refresh the
  // evicted slots as needed.
  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> admit(
      torch::Tensor ids) {
    TORCH_CHECK(ids.device().is_cpu() && ids.dtype() == torch::kInt64);
    auto idsc = ids.contiguous();
    int64_t n = idsc.numel();
    auto slots = torch::empty({n}, torch::kInt64);
    const int64_t* ip = idsc.data_ptr<int64_t>();
    int64_t* sp = slots.data_ptr<int64_t>();
    std::vector<int64_t> ev_ids, ev_slots;
    for (int64_t i = 0; i < n; ++i) {
      auto it = map_.find(ip[i]);
      if (it != map_.end()) {  // admitted earlier in this batch / racing
        sp[i] = it->second;
        continue;
      }
      int64_t slot = -1;
      if (!free_.empty()) {
        slot = free_.back();
        free_.pop_back();
      } else {
        // first eviction candidate NOT pinned by the current batch
        auto low = order_.begin();
        while (low != order_.end() && slot_batch_[low->second] == batch_)
          ++low;
        if (low == order_.end()) {  // whole cache pinned: serve direct
          sp[i] = -1;
          continue;
        }
        if (policy_ == LFUOPT) {
          int64_t cand_freq = ghost_freq_[ip[i]];
          if (cand_freq <= low->first) {  // does not beat cache minimum
            sp[i] = -1;
            continue;
          }
        }
        slot = low->second;
        order_.erase(low);
        map_.erase(slot_id_[slot]);
        ev_ids.push_back(slot_id_[slot]);
        ev_slots.push_back(slot);
      }
      slot_id_[slot] = ip[i];
      slot_freq_[slot] =
          policy_ == LFUOPT ? ghost_freq_[ip[i]] : 1;
      ++tick_;
      slot_key_[slot] = tick_;
      slot_version_[slot] = version_;
      slot_batch_[slot] = batch_;
      order_.insert({prio(slot), slot});
      map_[ip[i]] = slot;
      sp[i] = slot;
    }
    auto evi = torch::from_blob(ev_ids.data(), {(int64_t)ev_ids.size()},
                                torch::kInt64).clone();
    auto evs = torch::from_blob(ev_slots.data(), {(int64_t)ev_slots.size()},
                                torch::kInt64).clone();
    return {slots, evi, evs};
  }

  // Versioned sync (bounded staleness): bump_version() after each global
  // parameter update; stale(slots-of-interest is the whole cache) returns
  // (ids, slots) with version older than current - bound, and marks them
  // current (the caller re-pulls those rows).
  void bump_version() { ++version_; }

  std::pair<torch::Tensor, torch::Tensor> stale(int64_t bound) {
    std::vector<int64_t> ids, slots;
    for (auto& kv : map_) {
      int64_t s = kv.second;
      if (version_ - slot_version_[s] > bound) {
        ids.push_back(kv.first);
        slots.push_back(s);
        slot_version_[s] = version_;
      }
    }
    auto t1 = torch::from_blob(ids.data(), {(int64_t)ids.size()},
                               torch::kInt64).clone();
    auto t2 = torch::from_blob(slots.data(), {(int64_t)slots.size()},
                               torch::kInt64).clone();
    return {t1, t2};
  }

  // mark rows (by slot) as refreshed at the current version
  void refresh(torch::Tensor slots) {
    auto sc = slots.contiguous();
    const int64_t* sp = sc.data_ptr<int64_t>();
    for (int64_t i = 0; i < sc.numel(); ++i)
      if (sp[i] >= 0) slot_version_[sp[i]] = version_;
  }

  int64_t size() const { return (int64_t)map_.size(); }
  int64_t capacity() const { return capacity_; }
  double hit_rate() const {
    return lookups_ ? (double)hits_ / (double)lookups_ : 0.0;
  }

  int64_t capacity_;
  Policy policy_;
  std::unordered_map<int64_t, int64_t> map_;       // id -> slot
  std::unordered_map<int64_t, int64_t> ghost_freq_;  // LFUOpt admission
  std::vector<int64_t> slot_id_, slot_key_, slot_freq_, slot_version_,
      slot_batch_;
  std::set<std::pair<int64_t, int64_t>> order_;    // (prio, slot)
  std::vector<int64_t> free_;
  int64_t tick_ = 0, version_ = 0, hits_ = 0, lookups_ = 0, batch_ = 0;
};

}  // namespace

void register_embed_cache(pybind11::module& m) {
  pybind11::class_<EmbedCache>(m, "EmbedCache")
      .def(pybind11::init<int64_t, const std::string&>())
      .def("query", &EmbedCache::query)
      .def("admit", &EmbedCache::admit)
      .def("bump_version", &EmbedCache::bump_version)
      .def("stale", &EmbedCache::stale)
      .def("refresh", &EmbedCache::refresh)
      .def("size", &EmbedCache::size)
      .def("capacity", &EmbedCache::capacity)
      .def("hit_rate", &EmbedCache::hit_rate);
}
