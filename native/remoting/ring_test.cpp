// tf_ring_test — CPU unit test of the SPSC ring + futex doorbells
// (tests/test_remoting_cpu.py runs it; no GPU needed).
#include <pthread.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>

#include "protocol.h"
#include "ring.h"

using namespace tfrpc;

struct Ctx {
  Header* hdr;
  long n;
};

static void* consumer(void* arg) {
  Ctx* ctx = (Ctx*)arg;
  RingView cmd(&ctx->hdr->cmd, cmd_buf(ctx->hdr), CMD_RING_BYTES);
  long seen = 0;
  uint64_t sum = 0;
  while (seen < ctx->n) {
    size_t len;
    uint8_t* p = cmd.try_next(&len);
    if (!p) {
      cmd.wait_nonempty();
      continue;
    }
    auto* h = (CmdHdr*)p;
    if (h->seq != (uint64_t)seen + 1) {
      fprintf(stderr, "order violation: got %lu want %ld\n",
              (unsigned long)h->seq, seen + 1);
      exit(1);
    }
    uint8_t* body = p + sizeof(CmdHdr);
    for (uint32_t i = 0; i < h->body_len; ++i) {
      if (body[i] != (uint8_t)(h->seq + i)) {
        fprintf(stderr, "payload corruption at seq %lu\n",
                (unsigned long)h->seq);
        exit(1);
      }
      sum += body[i];
    }
    cmd.pop();
    ++seen;
  }
  return (void*)sum;
}

int main(int argc, char** argv) {
  long n = argc > 1 ? atol(argv[1]) : 200000;
  void* seg = mmap(nullptr, HDR_BYTES + CMD_RING_BYTES, PROT_READ | PROT_WRITE,
                   MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
  auto* hdr = (Header*)seg;
  memset(hdr, 0, HDR_BYTES);
  pthread_t t;
  Ctx ctx{hdr, n};
  pthread_create(&t, nullptr, consumer, &ctx);

  RingView cmd(&hdr->cmd, cmd_buf(hdr), CMD_RING_BYTES);
  uint64_t sum = 0;
  srand(7);
  for (long i = 1; i <= n; ++i) {
    uint32_t blen = rand() % 300;  // varied sizes force wraps
    uint8_t body[300];
    for (uint32_t j = 0; j < blen; ++j) {
      body[j] = (uint8_t)(i + j);
      sum += body[j];
    }
    uint8_t* p;
    while (!(p = cmd.try_reserve(sizeof(CmdHdr) + blen))) sched_yield();
    auto* h = (CmdHdr*)p;
    h->op = OP_NOP;
    h->flags = 0;
    h->seq = i;
    h->body_len = blen;
    memcpy(p + sizeof(CmdHdr), body, blen);
    cmd.commit();
    if ((i & 1023) == 0) cmd.wake_consumer();
  }
  cmd.wake_consumer();
  void* out;
  pthread_join(t, &out);
  if ((uint64_t)out != sum) {
    fprintf(stderr, "sum mismatch\n");
    return 1;
  }
  printf("{\"records\": %ld, \"ok\": true}\n", n);
  return 0;
}
