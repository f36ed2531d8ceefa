// Intra-node shared-memory allreduce for CPU tensors (reference:
// csrc/cpu/comm/shm.cpp, 692 LoC — used for CPU-side inference TP and
// offloaded collectives that should not bounce through a NIC loopback).
//
// Design (host C++, vendor-neutral): one POSIX shm segment per
// (name, world) group holding a header of atomics (arrival / generation
// counters for a sense-reversing barrier) plus world slots of payload.
// Every rank copies in, barriers, rank 0 reduces all slots into slot 0,
// barriers, everyone copies out. fp32 reduction; bf16/fp16 callers
// convert at the torch layer. Segments are unlinked by rank 0 on close.

#include <atomic>
#include <cstdio>
#include <cstring>
#include <fcntl.h>
#include <string>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

namespace {

struct ShmHeader {
  std::atomic<int> arrived;
  std::atomic<int> generation;
  std::atomic<int> initialized;
};

struct ShmGroup {
  ShmHeader* hdr;
  float* slots;        // [world][max_elems]
  long long max_elems;
  int rank, world;
  std::string name;
  size_t bytes;
};

void barrier_wait(ShmHeader* h, int world) {
  const int gen = h->generation.load(std::memory_order_acquire);
  if (h->arrived.fetch_add(1, std::memory_order_acq_rel) == world - 1) {
    h->arrived.store(0, std::memory_order_release);
    h->generation.fetch_add(1, std::memory_order_acq_rel);
  } else {
    while (h->generation.load(std::memory_order_acquire) == gen) {
      // spin; intra-node latencies are ~us
    }
  }
}

}  // namespace

extern "C" void* ds_shm_open(const char* name, int rank, int world,
                             long long max_elems) {
  std::string path = std::string("/ds_amd_") + name;
  size_t bytes = sizeof(ShmHeader) +
                 (size_t)world * max_elems * sizeof(float);
  int fd = shm_open(path.c_str(), O_CREAT | O_RDWR, 0600);
  if (fd < 0) return nullptr;
  if (ftruncate(fd, (off_t)bytes) != 0) {
    close(fd);
    return nullptr;
  }
  void* base = mmap(nullptr, bytes, PROT_READ | PROT_WRITE, MAP_SHARED, fd,
                    0);
  close(fd);
  if (base == MAP_FAILED) return nullptr;
  auto* g = new ShmGroup();
  g->hdr = reinterpret_cast<ShmHeader*>(base);
  g->slots = reinterpret_cast<float*>(
      reinterpret_cast<char*>(base) + sizeof(ShmHeader));
  g->max_elems = max_elems;
  g->rank = rank;
  g->world = world;
  g->name = path;
  g->bytes = bytes;
  if (rank == 0) {
    g->hdr->arrived.store(0);
    g->hdr->generation.store(0);
    g->hdr->initialized.store(1, std::memory_order_release);
  } else {
    while (g->hdr->initialized.load(std::memory_order_acquire) != 1) {
    }
  }
  return g;
}

extern "C" int ds_shm_allreduce(void* handle, float* data, long long n) {
  auto* g = reinterpret_cast<ShmGroup*>(handle);
  if (g == nullptr || n > g->max_elems) return -1;
  std::memcpy(g->slots + (size_t)g->rank * g->max_elems, data,
              n * sizeof(float));
  barrier_wait(g->hdr, g->world);
  if (g->rank == 0) {
    for (int r = 1; r < g->world; ++r) {
      const float* src = g->slots + (size_t)r * g->max_elems;
      for (long long i = 0; i < n; ++i) g->slots[i] += src[i];
    }
  }
  barrier_wait(g->hdr, g->world);
  std::memcpy(data, g->slots, n * sizeof(float));
  barrier_wait(g->hdr, g->world);  // slot 0 free for the next call
  return 0;
}

extern "C" void ds_shm_close(void* handle) {
  auto* g = reinterpret_cast<ShmGroup*>(handle);
  if (g == nullptr) return;
  const bool owner = g->rank == 0;
  std::string path = g->name;
  munmap(reinterpret_cast<void*>(g->hdr), g->bytes);
  if (owner) shm_unlink(path.c_str());
  delete g;
}
