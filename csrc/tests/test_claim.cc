// Replay of the soak-stall state against claim_chunk (stall triage).
#include <cstdio>
#include <cstdlib>

#include "../net/transport.h"

using namespace baguanet;

int main() {
  SendComm c;
  for (int i = 0; i < 4; i++) {
    auto* s = new TcpSock();
    s->idx = i;
    c.socks.push_back(s);
  }
  // state from the dump: oldest=20172, evens 20172..20202 ACTIVE with
  // avail==total, odds completed (FREE with their seq)
  c.oldest.store(20172);
  c.seq_next = 20204;
  for (uint32_t q = 20172; q < 20204; q++) {
    SendRequest* r = &c.reqs[q % NCCL_NET_MAX_REQUESTS];
    if (q % 2 == 0) {
      r->total = (q % 4 == 0) ? 5000 : 64;
      r->chunk = 32768;
      r->cursor.store(pack_cur(q, 0));
      r->avail.store(r->total.load());
      r->sent.store(0);
      r->state_seq.store(pack_ss(q, REQ_ACTIVE));
    } else {
      r->state_seq.store(pack_ss(q, REQ_FREE));
    }
  }
  uint32_t off = 1, len = 1;
  SendRequest* r0 = claim_chunk(&c, 0, &off, &len);
  printf("sock0 claim: %s off=%u len=%u\n", r0 ? "HIT" : "NULL", off, len);
  SendRequest* r1 = claim_chunk(&c, 1, &off, &len);
  printf("sock1 claim: %s\n", r1 ? "HIT" : "NULL");
  if (!r0) {
    fprintf(stderr, "BUG REPRODUCED: claimable even request not claimed\n");
    return 1;
  }
  return 0;
}
