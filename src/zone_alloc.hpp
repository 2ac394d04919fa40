// Offset-based best-fit slab allocator with neighbor coalescing.
//
// Reference parity: utils/zone_malloc.c:1-398 (segment/fit allocator with an
// RB-tree of free chunks over one big slab, used for the GPU memory pool).
// Here the RB-tree becomes two ordered maps — free blocks keyed by offset
// (for O(log n) neighbor merging on free) and by size (for best-fit) — which
// fixes the round-1 fragmentation risk: size-class freelists that never
// coalesce can strand capacity under mixed tile sizes (main tiles + subtiles
// + QR workspaces).
//
// Pure host-side bookkeeping (no HIP): unit-testable on CPU.
#pragma once

#include <cstddef>
#include <map>

namespace pa {

class ZoneAlloc {
 public:
  static constexpr size_t NPOS = ~size_t(0);

  void init(size_t bytes) {
    bytes_ = bytes;
    reset();
  }

  void reset() {
    by_off_.clear();
    by_size_.clear();
    in_use_ = 0;
    if (bytes_) insert_free(0, bytes_);
  }

  // Best-fit allocation; returns the offset or NPOS when no free block fits.
  size_t alloc(size_t sz) {
    if (sz == 0) sz = 1;
    auto it = by_size_.lower_bound(sz);
    if (it == by_size_.end()) return NPOS;
    size_t bsz = it->first, off = it->second;
    by_size_.erase(it);
    by_off_.erase(off);
    if (bsz > sz) insert_free(off + sz, bsz - sz);
    in_use_ += sz;
    return off;
  }

  // Free [off, off+sz), merging with adjacent free blocks.
  void free(size_t off, size_t sz) {
    in_use_ -= sz;
    auto next = by_off_.lower_bound(off);
    if (next != by_off_.begin()) {
      auto prev = std::prev(next);
      if (prev->first + prev->second == off) {  // merge left
        off = prev->first;
        sz += prev->second;
        erase_free(prev);
      }
    }
    if (next != by_off_.end() && off + sz == next->first) {  // merge right
      sz += next->second;
      erase_free(next);
    }
    insert_free(off, sz);
  }

  size_t in_use() const { return in_use_; }
  size_t capacity() const { return bytes_; }
  size_t largest_free() const {
    return by_size_.empty() ? 0 : by_size_.rbegin()->first;
  }
  size_t free_blocks() const { return by_off_.size(); }

 private:
  void insert_free(size_t off, size_t sz) {
    by_off_[off] = sz;
    by_size_.emplace(sz, off);
  }
  void erase_free(std::map<size_t, size_t>::iterator it) {
    auto range = by_size_.equal_range(it->second);
    for (auto s = range.first; s != range.second; ++s)
      if (s->second == it->first) {
        by_size_.erase(s);
        break;
      }
    by_off_.erase(it);
  }

  size_t bytes_ = 0;
  size_t in_use_ = 0;
  std::map<size_t, size_t> by_off_;        // offset -> size
  std::multimap<size_t, size_t> by_size_;  // size -> offset
};

}  // namespace pa
