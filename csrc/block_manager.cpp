// Native paged-KV block manager.
//
// The C++ twin of engine/kv.py::PyBlockManager (tests assert identical
// behavior): free-list allocation, per-sequence block tables, slot-mapping
// computation.  Lives in C++ so scheduling thousands of concurrent
// sequences costs no Python-object churn on the hot step path; calls
// release the GIL at the binding layer.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <stdexcept>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

class BlockManager {
 public:
  BlockManager(int64_t num_blocks, int64_t block_size)
      : num_blocks_(num_blocks), block_size_(block_size) {
    free_.reserve(num_blocks);
    for (int64_t b = num_blocks - 1; b >= 0; --b) free_.push_back(b);
  }

  void add_seq(int64_t seq_id) {
    if (tables_.count(seq_id))
      throw py::key_error("seq " + std::to_string(seq_id) + " already exists");
    tables_[seq_id] = {};
    lens_[seq_id] = 0;
  }

  void free_seq(int64_t seq_id) {
    auto it = tables_.find(seq_id);
    if (it == tables_.end()) return;
    if (!it->second.empty()) ++table_epoch_;
    for (auto rit = it->second.rbegin(); rit != it->second.rend(); ++rit)
      release_block(*rit);
    tables_.erase(it);
    lens_.erase(seq_id);
  }

  bool has_seq(int64_t seq_id) const { return tables_.count(seq_id) != 0; }

  bool can_append(int64_t seq_id, int64_t n_tokens) const {
    return blocks_needed(seq_id, n_tokens) <= (int64_t)free_.size();
  }

  std::vector<int64_t> append_tokens(int64_t seq_id, int64_t n_tokens) {
    const int64_t need = blocks_needed(seq_id, n_tokens);
    if (need > (int64_t)free_.size())
      throw std::runtime_error("seq " + std::to_string(seq_id) + ": need " +
                               std::to_string(need) + " blocks, " +
                               std::to_string(free_.size()) + " free");
    auto& table = tables_.at(seq_id);
    if (need > 0) ++table_epoch_;
    for (int64_t i = 0; i < need; ++i) {
      table.push_back(free_.back());
      free_.pop_back();
    }
    std::vector<int64_t> slots;
    slots.reserve(n_tokens);
    const int64_t start = lens_.at(seq_id);
    for (int64_t i = 0; i < n_tokens; ++i) {
      const int64_t pos = start + i;
      slots.push_back(table[pos / block_size_] * block_size_ + pos % block_size_);
    }
    lens_[seq_id] = start + n_tokens;
    return slots;
  }

  void pop_last_token(int64_t seq_id) {
    auto lit = lens_.find(seq_id);
    if (lit == lens_.end() || lit->second <= 0)
      throw std::runtime_error("pop_last_token: empty seq");
    const int64_t cur = lit->second;
    lit->second = cur - 1;
    auto& table = tables_.at(seq_id);
    if ((cur - 1) % block_size_ == 0 &&
        (int64_t)table.size() * block_size_ >= cur) {
      release_block(table.back());
      table.pop_back();
      ++table_epoch_;
    }
  }

  void adopt_prefix(int64_t new_seq_id, int64_t old_seq_id, int64_t n_blocks,
                    int64_t n_tokens) {
    if (tables_.count(new_seq_id))
      throw py::key_error("seq " + std::to_string(new_seq_id) + " already exists");
    auto it = tables_.find(old_seq_id);
    if (it == tables_.end())
      throw py::key_error("seq " + std::to_string(old_seq_id) + " not found");
    auto old = std::move(it->second);
    tables_.erase(it);
    lens_.erase(old_seq_id);
    if (n_blocks > (int64_t)old.size() || n_tokens > n_blocks * block_size_)
      throw std::runtime_error("adopt_prefix: bad prefix bounds");
    tables_[new_seq_id] =
        std::vector<int64_t>(old.begin(), old.begin() + n_blocks);
    lens_[new_seq_id] = n_tokens;
    for (auto rit = old.rbegin(); rit != old.rend() - n_blocks; ++rit)
      release_block(*rit);
    ++table_epoch_;
  }

  void share_prefix(int64_t new_seq_id, int64_t donor_seq_id, int64_t n_blocks,
                    int64_t n_tokens) {
    if (tables_.count(new_seq_id))
      throw py::key_error("seq " + std::to_string(new_seq_id) + " already exists");
    auto it = tables_.find(donor_seq_id);
    if (it == tables_.end())
      throw py::key_error("seq " + std::to_string(donor_seq_id) + " not found");
    if (n_blocks > (int64_t)it->second.size() || n_tokens != n_blocks * block_size_)
      throw std::runtime_error("share_prefix: full-block prefixes only");
    std::vector<int64_t> shared(it->second.begin(), it->second.begin() + n_blocks);
    for (int64_t b : shared) {
      auto r = refs_.find(b);
      refs_[b] = (r == refs_.end() ? 1 : r->second) + 1;
    }
    tables_[new_seq_id] = std::move(shared);
    lens_[new_seq_id] = n_tokens;
    ++table_epoch_;
  }

  int64_t ref_count(int64_t block) const {
    auto r = refs_.find(block);
    return r == refs_.end() ? 1 : r->second;
  }

  std::vector<int64_t> block_table(int64_t seq_id) const {
    return tables_.at(seq_id);
  }

  int64_t seq_len(int64_t seq_id) const { return lens_.at(seq_id); }

  int64_t free_blocks() const { return (int64_t)free_.size(); }
  int64_t table_epoch() const { return table_epoch_; }
  int64_t used_blocks() const { return num_blocks_ - (int64_t)free_.size(); }
  int64_t num_blocks() const { return num_blocks_; }
  int64_t block_size() const { return block_size_; }
  double occupancy() const {
    return num_blocks_ ? (double)used_blocks() / (double)num_blocks_ : 0.0;
  }

 private:
  void release_block(int64_t b) {
    auto r = refs_.find(b);
    if (r == refs_.end() || r->second <= 1) {
      if (r != refs_.end()) refs_.erase(r);
      free_.push_back(b);
    } else {
      --r->second;
    }
  }

  int64_t blocks_needed(int64_t seq_id, int64_t n_tokens) const {
    const int64_t cur = lens_.at(seq_id);
    const int64_t have = (int64_t)tables_.at(seq_id).size();
    const int64_t total = (cur + n_tokens + block_size_ - 1) / block_size_;
    return std::max<int64_t>(0, total - have);
  }

  int64_t num_blocks_, block_size_;
  int64_t table_epoch_ = 0;
  std::vector<int64_t> free_;
  std::unordered_map<int64_t, std::vector<int64_t>> tables_;
  std::unordered_map<int64_t, int64_t> lens_;
  std::unordered_map<int64_t, int64_t> refs_;
};

void register_block_manager(py::module_& m) {
  py::class_<BlockManager>(m, "BlockManager")
      .def(py::init<int64_t, int64_t>(), py::arg("num_blocks"),
           py::arg("block_size"))
      .def("add_seq", &BlockManager::add_seq)
      .def("free_seq", &BlockManager::free_seq)
      .def("has_seq", &BlockManager::has_seq)
      .def("can_append", &BlockManager::can_append)
      .def("append_tokens", &BlockManager::append_tokens)
      .def("adopt_prefix", &BlockManager::adopt_prefix)
      .def("pop_last_token", &BlockManager::pop_last_token)
      .def("share_prefix", &BlockManager::share_prefix)
      .def("ref_count", &BlockManager::ref_count)
      .def("block_table", &BlockManager::block_table)
      .def("seq_len", &BlockManager::seq_len)
      .def("occupancy", &BlockManager::occupancy)
      .def_property_readonly("free_blocks", &BlockManager::free_blocks)
      .def_property_readonly("table_epoch", &BlockManager::table_epoch)
      .def_property_readonly("used_blocks", &BlockManager::used_blocks)
      .def_property_readonly("num_blocks", &BlockManager::num_blocks)
      .def_property_readonly("block_size", &BlockManager::block_size);
}
