// Native unit tests for the staging arena allocator.  Exits non-zero on
// the first failure; run by `make test` / tests/test_native_units.py.

#include <cstdio>
#include <cstdlib>
#include <random>
#include <vector>

#include "../net/freelist.h"

using baguanet::FreeList;

#define REQUIRE(cond)                                                   \
  do {                                                                  \
    if (!(cond)) {                                                      \
      fprintf(stderr, "FAILED %s:%d: %s\n", __FILE__, __LINE__, #cond); \
      exit(1);                                                          \
    }                                                                   \
  } while (0)

int main() {
  {  // basic alloc/free/coalesce
    FreeList f(1 << 20);
    size_t a = f.alloc(1000), b = f.alloc(1), c = f.alloc(256 * 3);
    REQUIRE(a != SIZE_MAX && b != SIZE_MAX && c != SIZE_MAX);
    REQUIRE(a != b && b != c);
    f.free(b, 1);
    f.free(a, 1000);
    f.free(c, 256 * 3);
    REQUIRE(f.fragments() == 1);           // fully coalesced
    REQUIRE(f.free_bytes() == (1 << 20));  // nothing leaked
  }
  {  // exhaustion + exact reuse
    FreeList f(1024);
    size_t a = f.alloc(512), b = f.alloc(512);
    REQUIRE(a != SIZE_MAX && b != SIZE_MAX);
    REQUIRE(f.alloc(1) == SIZE_MAX);
    f.free(a, 512);
    REQUIRE(f.alloc(512) == a);
    f.free(b, 512);
  }
  {  // randomized invariants vs a shadow model
    std::mt19937 rng(7);
    const size_t arena = 4 << 20;
    FreeList f(arena);
    struct Live {
      size_t off, sz;
    };
    std::vector<Live> live;
    size_t live_bytes = 0;
    for (int step = 0; step < 200000; step++) {
      bool do_alloc = live.empty() || (rng() % 2 && live.size() < 64);
      if (do_alloc) {
        size_t sz = 1 + rng() % (256 * 1024);
        size_t off = f.alloc(sz);
        if (off != SIZE_MAX) {
          size_t need = FreeList::round_up(sz);
          // no overlap with any live block, inside the arena
          for (auto& l : live) {
            size_t ln = FreeList::round_up(l.sz);
            REQUIRE(off + need <= l.off || l.off + ln <= off);
          }
          REQUIRE(off + need <= arena);
          live.push_back({off, sz});
          live_bytes += need;
        }
        // on SIZE_MAX: first-fit may fail under fragmentation even with
        // enough total free bytes — nothing to assert
      } else {
        size_t i = rng() % live.size();
        f.free(live[i].off, live[i].sz);
        live_bytes -= FreeList::round_up(live[i].sz);
        live[i] = live.back();
        live.pop_back();
      }
      REQUIRE(f.free_bytes() == arena - live_bytes);
    }
    for (auto& l : live) f.free(l.off, l.sz);
    REQUIRE(f.fragments() == 1 && f.free_bytes() == arena);
  }
  printf("freelist tests ok\n");
  return 0;
}
