// freelist.h — first-fit free-list allocator over a fixed arena, used by
// the pinned staging ring (staging.cc).  Header-only so it can be
// unit-tested natively (csrc/tests/test_freelist.cc) — fragmentation or
// coalescing bugs here would otherwise only surface as GPU soak
// corruption.  Not thread-safe; callers hold the pool mutex.

#pragma once

#include <algorithm>
#include <cstddef>
#include <cstdint>
#include <vector>

namespace baguanet {

class FreeList {
 public:
  explicit FreeList(size_t arena_bytes) {
    free_.push_back({0, arena_bytes});
  }

  static size_t round_up(size_t sz) {
    size_t need = (sz + 255) & ~size_t(255);
    return need ? need : 256;
  }

  // Returns the offset of a block of round_up(sz) bytes, or SIZE_MAX.
  size_t alloc(size_t sz) {
    size_t need = round_up(sz);
    for (auto it = free_.begin(); it != free_.end(); ++it) {
      if (it->len >= need) {
        size_t off = it->off;
        it->off += need;
        it->len -= need;
        if (it->len == 0) free_.erase(it);
        return off;
      }
    }
    return SIZE_MAX;
  }

  // Releases a block previously returned by alloc(sz) (same sz).
  void free(size_t off, size_t sz) {
    size_t need = round_up(sz);
    Range r{off, need};
    auto it = std::lower_bound(
        free_.begin(), free_.end(), r,
        [](const Range& a, const Range& b) { return a.off < b.off; });
    it = free_.insert(it, r);
    if (it + 1 != free_.end() && it->off + it->len == (it + 1)->off) {
      it->len += (it + 1)->len;
      free_.erase(it + 1);
    }
    if (it != free_.begin() && (it - 1)->off + (it - 1)->len == it->off) {
      (it - 1)->len += it->len;
      free_.erase(it);
    }
  }

  size_t free_bytes() const {
    size_t t = 0;
    for (auto& r : free_) t += r.len;
    return t;
  }
  size_t fragments() const { return free_.size(); }

 private:
  struct Range {
    size_t off, len;
  };
  std::vector<Range> free_;
};

}  // namespace baguanet
