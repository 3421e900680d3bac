// Shared-memory object store for ant_ray_amd.
//
// Role parity: the reference's plasma store (reference:
// src/ray/object_manager/plasma/store.h:55, dlmalloc arena over mmap'd tmpfs at
// plasma/dlmalloc.cc:63-170, fd-passing protocol in plasma/fling.cc). This is a
// from-scratch design, not a translation: instead of a store *server* process
// that clients talk to over a unix socket (create/seal/get RPCs + fd passing),
// every client maps the same shm segment and operates on it directly:
//   * create/seal/get/release are in-process calls under a robust
//     process-shared pthread mutex living inside the segment,
//   * sealed-object notification is a futex on a global seal-sequence word
//     (same family of mechanism the reference uses for mutable objects,
//     src/ray/core_worker/experimental_mutable_object_manager.h:44),
//   * reads are zero-copy memoryviews over the mapped arena, pinned by a
//     per-object refcount, LRU-evicted only at refcount==0.
// This removes two IPC round-trips and one memcpy from every put/get compared
// to the plasma protocol, which is what lets the single-client put path run
// at memory bandwidth.
//
// Layout of the segment:
//   [ShmHeader | slot table (num_slots * Slot) | data arena ]
// Allocator: address-ordered first-fit free list with coalescing; blocks carry
// an 8-byte size header so free() is O(1) lookup + O(list) insert.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <cerrno>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include <fcntl.h>
#include <linux/futex.h>
#include <pthread.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/time.h>
#include <time.h>
#include <unistd.h>

namespace py = pybind11;

namespace antray {

static constexpr uint64_t kMagic = 0xA17BA1D0C0FFEE01ULL;
static constexpr uint32_t kIdLen = 20;
static constexpr uint64_t kAlign = 64;

// Slot states.
enum : uint32_t {
  SLOT_EMPTY = 0,
  SLOT_CREATED = 1,
  SLOT_SEALED = 2,
  SLOT_TOMBSTONE = 3,  // deleted; slot reusable but keeps probe chains intact
};

struct Slot {
  uint8_t oid[kIdLen];
  uint32_t state;
  std::atomic<uint32_t> refcount;
  uint64_t offset;     // absolute offset of payload (meta+data) in the segment
  uint64_t data_size;  // user data bytes
  uint32_t meta_size;  // metadata bytes (stored before data)
  uint32_t flags;      // bit 0: pending delete
  uint64_t last_access;
};
static_assert(sizeof(Slot) == 64, "slot should be cacheline sized");

struct FreeBlock {
  uint64_t size;  // total bytes of this free block (incl. this header)
  uint64_t next;  // absolute offset of next free block, 0 = end
};

struct ShmHeader {
  uint64_t magic;
  uint64_t segment_size;
  uint64_t arena_offset;
  uint64_t arena_size;
  uint32_t num_slots;
  uint32_t _pad0;
  pthread_mutex_t mutex;  // robust, process-shared
  uint64_t free_head;     // absolute offset of first free block, 0 = none
  uint64_t used_bytes;
  std::atomic<uint64_t> clock;     // LRU tick
  std::atomic<uint32_t> seal_seq;  // futex word: bumped+woken on every seal
  uint32_t _pad1;
  // stats
  uint64_t num_objects;
  uint64_t total_created;
  uint64_t total_evicted;
  uint64_t total_sealed_bytes;
};

static inline int futex_wait(std::atomic<uint32_t>* addr, uint32_t expected,
                             const struct timespec* timeout) {
  return syscall(SYS_futex, reinterpret_cast<uint32_t*>(addr), FUTEX_WAIT,
                 expected, timeout, nullptr, 0);
}
static inline int futex_wake_all(std::atomic<uint32_t>* addr) {
  return syscall(SYS_futex, reinterpret_cast<uint32_t*>(addr), FUTEX_WAKE,
                 INT32_MAX, nullptr, nullptr, 0);
}

static inline uint64_t align_up(uint64_t v, uint64_t a) {
  return (v + a - 1) & ~(a - 1);
}

class ShmStore : public std::enable_shared_from_this<ShmStore> {
 public:
  // Create a new segment backing file (head node does this once per node).
  static std::shared_ptr<ShmStore> Create(const std::string& path,
                                          uint64_t capacity,
                                          uint32_t num_slots) {
    int fd = ::open(path.c_str(), O_RDWR | O_CREAT | O_EXCL, 0600);
    if (fd < 0) throw std::runtime_error("shm create failed: " + path + ": " + strerror(errno));
    uint64_t slots_bytes = align_up((uint64_t)num_slots * sizeof(Slot), kAlign);
    uint64_t header_bytes = align_up(sizeof(ShmHeader), kAlign);
    uint64_t total = align_up(header_bytes + slots_bytes + capacity, 4096);
    if (ftruncate(fd, (off_t)total) != 0) {
      ::close(fd);
      throw std::runtime_error("shm ftruncate failed");
    }
    void* base = mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (base == MAP_FAILED) throw std::runtime_error("shm mmap failed");

    auto* h = reinterpret_cast<ShmHeader*>(base);
    std::memset(h, 0, sizeof(ShmHeader));
    h->segment_size = total;
    h->arena_offset = header_bytes + slots_bytes;
    h->arena_size = total - h->arena_offset;
    h->num_slots = num_slots;

    pthread_mutexattr_t attr;
    pthread_mutexattr_init(&attr);
    pthread_mutexattr_setpshared(&attr, PTHREAD_PROCESS_SHARED);
    pthread_mutexattr_setrobust(&attr, PTHREAD_MUTEX_ROBUST);
    pthread_mutex_init(&h->mutex, &attr);
    pthread_mutexattr_destroy(&attr);

    std::memset(reinterpret_cast<char*>(base) + header_bytes, 0, slots_bytes);

    // One big free block spanning the arena.
    auto* fb = reinterpret_cast<FreeBlock*>(reinterpret_cast<char*>(base) + h->arena_offset);
    fb->size = h->arena_size;
    fb->next = 0;
    h->free_head = h->arena_offset;

    std::atomic_thread_fence(std::memory_order_seq_cst);
    h->magic = kMagic;  // publish
    return std::shared_ptr<ShmStore>(new ShmStore(base, total, path));
  }

  // Open an existing segment (workers).
  static std::shared_ptr<ShmStore> Open(const std::string& path,
                                        double timeout_s) {
    double waited = 0;
    int fd = -1;
    while (true) {
      fd = ::open(path.c_str(), O_RDWR);
      if (fd >= 0) break;
      if (waited > timeout_s) throw std::runtime_error("shm open timed out: " + path);
      usleep(10000);
      waited += 0.01;
    }
    struct stat st;
    while (true) {
      if (fstat(fd, &st) != 0) { ::close(fd); throw std::runtime_error("shm fstat failed"); }
      if (st.st_size > (off_t)sizeof(ShmHeader)) break;
      if (waited > timeout_s) { ::close(fd); throw std::runtime_error("shm init timed out"); }
      usleep(10000);
      waited += 0.01;
    }
    uint64_t total = (uint64_t)st.st_size;
    void* base = mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (base == MAP_FAILED) throw std::runtime_error("shm mmap failed");
    auto* h = reinterpret_cast<ShmHeader*>(base);
    while (h->magic != kMagic) {
      if (waited > timeout_s) { munmap(base, total); throw std::runtime_error("shm magic timed out"); }
      usleep(10000);
      waited += 0.01;
    }
    return std::shared_ptr<ShmStore>(new ShmStore(base, total, path));
  }

  ~ShmStore() {
    if (base_) munmap(base_, size_);
  }

  // --- public ops -----------------------------------------------------------

  // Create an unsealed object; returns absolute payload data offset or throws.
  // The object is pinned (refcount 1) for the writer.
  uint64_t CreateObject(const std::string& oid, uint64_t data_size,
                        const std::string& meta) {
    CheckId(oid);
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    if (slot) throw py::value_error("object already exists");
    uint64_t payload = align_up(meta.size(), 8) + data_size;
    uint64_t off = Alloc(payload);
    if (!off) {
      EvictUntil(payload);
      off = Alloc(payload);
      if (!off) throw std::runtime_error("object store out of memory");
    }
    slot = InsertSlot(oid.data());
    if (!slot) { Free(off); throw std::runtime_error("object store slot table full"); }
    slot->state = SLOT_CREATED;
    slot->refcount.store(1, std::memory_order_relaxed);
    slot->offset = off;
    slot->data_size = data_size;
    slot->meta_size = (uint32_t)meta.size();
    slot->flags = 0;
    slot->last_access = header()->clock.fetch_add(1) + 1;
    if (!meta.empty()) std::memcpy(Ptr(off), meta.data(), meta.size());
    header()->num_objects++;
    header()->total_created++;
    return off + align_up(meta.size(), 8);
  }

  void Seal(const std::string& oid) {
    {
      Guard g(this);
      Slot* slot = FindSlot(oid.data());
      if (!slot || slot->state == SLOT_TOMBSTONE) throw py::key_error("no such object");
      slot->state = SLOT_SEALED;
      header()->total_sealed_bytes += slot->data_size;
    }
    header()->seal_seq.fetch_add(1, std::memory_order_release);
    futex_wake_all(&header()->seal_seq);
  }

  // Abort an unsealed create (failure path): free without sealing.
  void Abort(const std::string& oid) {
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    if (!slot) return;
    slot->refcount.store(0, std::memory_order_relaxed);
    FreeSlotLocked(slot);
  }

  // Blocking get: waits for seal. Returns (data_offset, data_size, meta) and
  // pins the object; caller must Release. Returns data_offset=UINT64_MAX on
  // timeout.
  std::tuple<uint64_t, uint64_t, py::bytes> GetObject(const std::string& oid,
                                                      double timeout_s) {
    CheckId(oid);
    struct timespec deadline;
    clock_gettime(CLOCK_MONOTONIC, &deadline);
    int64_t ns = (int64_t)(timeout_s * 1e9);
    deadline.tv_sec += ns / 1000000000;
    deadline.tv_nsec += ns % 1000000000;
    if (deadline.tv_nsec >= 1000000000) { deadline.tv_sec++; deadline.tv_nsec -= 1000000000; }

    while (true) {
      uint32_t seq = header()->seal_seq.load(std::memory_order_acquire);
      {
        Guard g(this);
        Slot* slot = FindSlot(oid.data());
        if (slot && slot->state == SLOT_SEALED) {
          slot->refcount.fetch_add(1, std::memory_order_relaxed);
          slot->last_access = header()->clock.fetch_add(1) + 1;
          uint64_t doff = slot->offset + align_up(slot->meta_size, 8);
          py::bytes meta;
          {
            // build meta bytes while holding pin (safe: we hold the lock)
            meta = py::bytes(reinterpret_cast<char*>(Ptr(slot->offset)), slot->meta_size);
          }
          return {doff, slot->data_size, meta};
        }
      }
      if (timeout_s == 0) return {UINT64_MAX, 0, py::bytes()};
      // Wait for the next seal anywhere, bounded by remaining time.
      struct timespec now;
      clock_gettime(CLOCK_MONOTONIC, &now);
      int64_t rem_ns = (deadline.tv_sec - now.tv_sec) * 1000000000LL +
                       (deadline.tv_nsec - now.tv_nsec);
      if (timeout_s >= 0 && rem_ns <= 0) return {UINT64_MAX, 0, py::bytes()};
      struct timespec rel;
      // cap waits at 50ms so cross-process wake misses can't hang us
      int64_t wait_ns = (timeout_s < 0) ? 50000000LL : std::min<int64_t>(rem_ns, 50000000LL);
      rel.tv_sec = wait_ns / 1000000000;
      rel.tv_nsec = wait_ns % 1000000000;
      py::gil_scoped_release nogil;
      futex_wait(&header()->seal_seq, seq, &rel);
    }
  }

  // Blocking get returning a PinnedBuffer (pin ownership moves to the buffer)
  // plus the metadata bytes. Returns (None, None) on timeout.
  // Defined after PinnedBuffer below.
  py::tuple GetBuffer(const std::string& oid, double timeout_s);

  bool Contains(const std::string& oid) {
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    return slot && slot->state == SLOT_SEALED;
  }

  void Release(const std::string& oid) {
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    if (!slot || slot->state == SLOT_TOMBSTONE) return;
    uint32_t prev = slot->refcount.fetch_sub(1, std::memory_order_acq_rel);
    if (prev == 1 && (slot->flags & 1)) FreeSlotLocked(slot);
  }

  void AddRef(const std::string& oid) {
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    if (slot && slot->state != SLOT_TOMBSTONE)
      slot->refcount.fetch_add(1, std::memory_order_relaxed);
  }

  // Owner-driven delete (refcount-zero objects freed now, pinned ones marked).
  void Delete(const std::string& oid) {
    Guard g(this);
    Slot* slot = FindSlot(oid.data());
    if (!slot || slot->state == SLOT_TOMBSTONE) return;
    if (slot->refcount.load(std::memory_order_acquire) == 0) {
      FreeSlotLocked(slot);
    } else {
      slot->flags |= 1;
    }
  }

  py::dict Stats() {
    Guard g(this);
    auto* h = header();
    py::dict d;
    d["segment_size"] = h->segment_size;
    d["arena_size"] = h->arena_size;
    d["used_bytes"] = h->used_bytes;
    d["num_objects"] = h->num_objects;
    d["total_created"] = h->total_created;
    d["total_evicted"] = h->total_evicted;
    d["total_sealed_bytes"] = h->total_sealed_bytes;
    return d;
  }

  // Zero-copy views -----------------------------------------------------------
  py::memoryview ViewAt(uint64_t offset, uint64_t size, bool writable) {
    if (offset + size > size_) throw std::runtime_error("view out of range");
    return py::memoryview::from_memory(Ptr(offset), (ssize_t)size, !writable);
  }

  uintptr_t AddressAt(uint64_t offset) { return (uintptr_t)Ptr(offset); }

  const std::string& path() const { return path_; }

  // --- mutable channels -------------------------------------------------------
  // A channel is a pinned, sealed store object whose payload is
  // [ChanHeader | capacity bytes]. Single writer, num_readers readers.
  // Writer blocks until every reader consumed the previous message
  // (reference parity: experimental/channel shared-memory channels used by
  // compiled graphs — python/ray/experimental/channel/shared_memory_channel.py);
  // readers block on a version futex, so a hop costs a memcpy + futex wake
  // instead of an RPC round trip.
  struct ChanHeader {
    std::atomic<uint32_t> version;     // futex: bumped on every write/close
    std::atomic<uint32_t> reads_done;  // futex: readers finished this version
    uint32_t num_readers;
    std::atomic<uint32_t> closed;
    uint64_t msg_size;  // UINT64_MAX = closed sentinel
    uint8_t _pad[40];
  };
  static_assert(sizeof(ChanHeader) == 64, "chan header one cacheline");

  uint64_t ChannelCreate(const std::string& oid, uint64_t capacity,
                         uint32_t num_readers) {
    uint64_t doff = CreateObject(oid, sizeof(ChanHeader) + capacity, "chan");
    auto* h = reinterpret_cast<ChanHeader*>(Ptr(doff));
    h->version.store(0, std::memory_order_relaxed);
    h->reads_done.store(num_readers, std::memory_order_relaxed);  // "consumed"
    h->num_readers = num_readers;
    h->closed.store(0, std::memory_order_relaxed);
    h->msg_size = 0;
    Seal(oid);  // creator's pin (refcount 1) keeps it resident
    return capacity;
  }

  void ChannelWrite(const std::string& oid, py::bytes data, double timeout_s) {
    uint64_t cap;
    ChanHeader* h = ChanPtr(oid, &cap);
    char* buf;
    ssize_t n;
    if (PYBIND11_BYTES_AS_STRING_AND_SIZE(data.ptr(), &buf, &n) != 0)
      throw py::value_error("bad bytes");
    if ((uint64_t)n > cap) throw py::value_error("message exceeds channel capacity");
    {
      py::gil_scoped_release nogil;
      struct timespec slice = {0, 50000000};
      double waited = 0;
      while (true) {
        if (h->closed.load(std::memory_order_acquire))
          throw std::runtime_error("channel closed");
        uint32_t rd = h->reads_done.load(std::memory_order_acquire);
        if (rd >= h->num_readers) break;
        if (timeout_s >= 0 && waited >= timeout_s)
          throw std::runtime_error("channel write timed out");
        futex_wait(&h->reads_done, rd, &slice);
        waited += 0.05;
      }
      std::memcpy(reinterpret_cast<char*>(h) + sizeof(ChanHeader), buf, n);
      h->msg_size = (uint64_t)n;
      h->reads_done.store(0, std::memory_order_release);
      h->version.fetch_add(1, std::memory_order_release);
      futex_wake_all(&h->version);
    }
  }

  // Returns (payload_bytes, new_version); raises on close/timeout.
  py::tuple ChannelRead(const std::string& oid, uint32_t last_version,
                        double timeout_s) {
    uint64_t cap;
    ChanHeader* h = ChanPtr(oid, &cap);
    std::string out;
    uint32_t v;
    {
      py::gil_scoped_release nogil;
      struct timespec slice = {0, 50000000};
      double waited = 0;
      while (true) {
        v = h->version.load(std::memory_order_acquire);
        if (v != last_version) break;
        if (h->closed.load(std::memory_order_acquire))
          throw std::runtime_error("channel closed");
        if (timeout_s >= 0 && waited >= timeout_s)
          throw std::runtime_error("channel read timed out");
        futex_wait(&h->version, v, &slice);
        waited += 0.05;
      }
      uint64_t n = h->msg_size;
      if (n == UINT64_MAX) throw std::runtime_error("channel closed");
      out.assign(reinterpret_cast<char*>(h) + sizeof(ChanHeader), n);
      h->reads_done.fetch_add(1, std::memory_order_acq_rel);
      futex_wake_all(&h->reads_done);
    }
    return py::make_tuple(py::bytes(out), v);
  }

  void ChannelClose(const std::string& oid) {
    uint64_t cap;
    ChanHeader* h = ChanPtr(oid, &cap);
    h->closed.store(1, std::memory_order_release);
    h->msg_size = UINT64_MAX;
    h->version.fetch_add(1, std::memory_order_release);
    futex_wake_all(&h->version);
    futex_wake_all(&h->reads_done);
  }

 private:
  ShmStore(void* base, uint64_t size, std::string path)
      : base_(base), size_(size), path_(std::move(path)) {}

  ShmHeader* header() { return reinterpret_cast<ShmHeader*>(base_); }
  char* Ptr(uint64_t off) { return reinterpret_cast<char*>(base_) + off; }

  static void CheckId(const std::string& oid) {
    if (oid.size() != kIdLen) throw py::value_error("object id must be 20 bytes");
  }

  // Resolve a channel's header pointer (pinned objects never move).
  ChanHeader* ChanPtr(const std::string& oid, uint64_t* cap_out) {
    CheckId(oid);
    Guard g(this);
    Slot* s = FindSlot(oid.data());
    if (!s || s->state != SLOT_SEALED) throw py::key_error("no such channel");
    if (s->data_size < sizeof(ChanHeader)) throw py::value_error("not a channel");
    *cap_out = s->data_size - sizeof(ChanHeader);
    return reinterpret_cast<ChanHeader*>(Ptr(s->offset + align_up(s->meta_size, 8)));
  }

  struct Guard {
    ShmStore* s;
    explicit Guard(ShmStore* store) : s(store) {
      int rc = pthread_mutex_lock(&s->header()->mutex);
      if (rc == EOWNERDEAD) {
        // A client died holding the lock. State under the lock is guarded by
        // short critical sections; mark consistent and continue.
        pthread_mutex_consistent(&s->header()->mutex);
      } else if (rc != 0) {
        throw std::runtime_error("shm mutex lock failed");
      }
    }
    ~Guard() { pthread_mutex_unlock(&s->header()->mutex); }
  };

  // --- slot table (open addressing, linear probing) -------------------------
  uint64_t HashId(const char* oid) {
    uint64_t h;
    std::memcpy(&h, oid, 8);
    h ^= h >> 33;
    h *= 0xff51afd7ed558ccdULL;
    h ^= h >> 33;
    return h;
  }

  Slot* SlotAt(uint32_t i) {
    return reinterpret_cast<Slot*>(Ptr(align_up(sizeof(ShmHeader), kAlign))) + i;
  }

  Slot* FindSlot(const char* oid) {
    uint32_t n = header()->num_slots;
    uint32_t i = (uint32_t)(HashId(oid) % n);
    for (uint32_t probe = 0; probe < n; ++probe, i = (i + 1) % n) {
      Slot* s = SlotAt(i);
      if (s->state == SLOT_EMPTY) return nullptr;
      if (s->state != SLOT_TOMBSTONE && std::memcmp(s->oid, oid, kIdLen) == 0)
        return s;
    }
    return nullptr;
  }

  Slot* InsertSlot(const char* oid) {
    uint32_t n = header()->num_slots;
    uint32_t i = (uint32_t)(HashId(oid) % n);
    for (uint32_t probe = 0; probe < n; ++probe, i = (i + 1) % n) {
      Slot* s = SlotAt(i);
      if (s->state == SLOT_EMPTY || s->state == SLOT_TOMBSTONE) {
        std::memcpy(s->oid, oid, kIdLen);
        return s;
      }
    }
    return nullptr;
  }

  // --- allocator (address-ordered first fit w/ coalescing) ------------------
  // Allocated blocks: [8B size][payload...]; returned offset points at payload.
  uint64_t Alloc(uint64_t payload) {
    uint64_t need = align_up(payload + 8, kAlign);
    auto* h = header();
    uint64_t prev = 0, cur = h->free_head;
    while (cur) {
      auto* fb = reinterpret_cast<FreeBlock*>(Ptr(cur));
      if (fb->size >= need) {
        uint64_t remain = fb->size - need;
        if (remain >= kAlign * 2) {
          // split: keep the tail free
          uint64_t tail = cur + need;
          auto* tb = reinterpret_cast<FreeBlock*>(Ptr(tail));
          tb->size = remain;
          tb->next = fb->next;
          if (prev) reinterpret_cast<FreeBlock*>(Ptr(prev))->next = tail;
          else h->free_head = tail;
          fb->size = need;
        } else {
          need = fb->size;
          if (prev) reinterpret_cast<FreeBlock*>(Ptr(prev))->next = fb->next;
          else h->free_head = fb->next;
        }
        *reinterpret_cast<uint64_t*>(Ptr(cur)) = need;
        h->used_bytes += need;
        return cur + 8;
      }
      prev = cur;
      cur = fb->next;
    }
    return 0;
  }

  void Free(uint64_t payload_off) {
    uint64_t blk = payload_off - 8;
    uint64_t bsize = *reinterpret_cast<uint64_t*>(Ptr(blk));
    auto* h = header();
    h->used_bytes -= bsize;
    // address-ordered insert with coalescing
    uint64_t prev = 0, cur = h->free_head;
    while (cur && cur < blk) {
      prev = cur;
      cur = reinterpret_cast<FreeBlock*>(Ptr(cur))->next;
    }
    auto* nb = reinterpret_cast<FreeBlock*>(Ptr(blk));
    nb->size = bsize;
    nb->next = cur;
    if (prev) reinterpret_cast<FreeBlock*>(Ptr(prev))->next = blk;
    else h->free_head = blk;
    // coalesce with next
    if (cur && blk + nb->size == cur) {
      auto* cb = reinterpret_cast<FreeBlock*>(Ptr(cur));
      nb->size += cb->size;
      nb->next = cb->next;
    }
    // coalesce with prev
    if (prev) {
      auto* pb = reinterpret_cast<FreeBlock*>(Ptr(prev));
      if (prev + pb->size == blk) {
        pb->size += nb->size;
        pb->next = nb->next;
      }
    }
  }

  void FreeSlotLocked(Slot* slot) {
    Free(slot->offset);
    slot->state = SLOT_TOMBSTONE;
    header()->num_objects--;
  }

  void EvictUntil(uint64_t payload) {
    // Evict LRU sealed refcount-0 objects until a block of `payload` fits.
    uint64_t need = align_up(payload + 8, kAlign);
    auto* h = header();
    while (true) {
      // quick check: any free block big enough?
      uint64_t cur = h->free_head;
      while (cur) {
        auto* fb = reinterpret_cast<FreeBlock*>(Ptr(cur));
        if (fb->size >= need) return;
        cur = fb->next;
      }
      // find LRU victim
      Slot* victim = nullptr;
      for (uint32_t i = 0; i < h->num_slots; ++i) {
        Slot* s = SlotAt(i);
        if (s->state == SLOT_SEALED &&
            s->refcount.load(std::memory_order_acquire) == 0) {
          if (!victim || s->last_access < victim->last_access) victim = s;
        }
      }
      if (!victim) return;  // nothing evictable; caller will fail alloc
      FreeSlotLocked(victim);
      h->total_evicted++;
    }
  }

  void* base_;
  uint64_t size_;
  std::string path_;

  friend class PinnedBuffer;
};

// A read-only buffer over a sealed object, holding its refcount pin.
// Objects deserialized zero-copy (numpy arrays over shm) keep this alive via
// the buffer protocol (`.base` chains), so eviction cannot invalidate them.
class PinnedBuffer {
 public:
  PinnedBuffer(std::shared_ptr<ShmStore> store, std::string oid, uint64_t off,
               uint64_t size)
      : store_(std::move(store)), oid_(std::move(oid)), off_(off), size_(size) {}
  // copying takes its OWN pin; moving transfers it. (The default copy ctor
  // previously let the pybind return-tuple's temporary release the pin on
  // destruction, leaving live zero-copy readers unprotected and corrupting
  // the refcount when callers released explicitly.)
  PinnedBuffer(const PinnedBuffer& o)
      : store_(o.store_), oid_(o.oid_), off_(o.off_), size_(o.size_) {
    if (store_) store_->AddRef(oid_);
  }
  PinnedBuffer(PinnedBuffer&& o) noexcept
      : store_(std::move(o.store_)), oid_(std::move(o.oid_)), off_(o.off_),
        size_(o.size_) {
    o.store_.reset();
  }
  PinnedBuffer& operator=(const PinnedBuffer&) = delete;
  PinnedBuffer& operator=(PinnedBuffer&&) = delete;
  ~PinnedBuffer() {
    if (!store_) return;
    try {
      store_->Release(oid_);
    } catch (...) {
    }
  }
  char* data() { return store_->Ptr(off_); }
  uint64_t size() const { return size_; }
  const std::string& oid() const { return oid_; }

 private:
  std::shared_ptr<ShmStore> store_;
  std::string oid_;
  uint64_t off_;
  uint64_t size_;
};

py::tuple ShmStore::GetBuffer(const std::string& oid, double timeout_s) {
  auto [doff, size, meta] = GetObject(oid, timeout_s);
  if (doff == UINT64_MAX) return py::make_tuple(py::none(), py::none());
  return py::make_tuple(PinnedBuffer(shared_from_this(), oid, doff, size), meta);
}

}  // namespace antray

PYBIND11_MODULE(_shm_store, m) {
  using antray::PinnedBuffer;
  using antray::ShmStore;

  py::class_<PinnedBuffer>(m, "PinnedBuffer", py::buffer_protocol())
      .def_buffer([](PinnedBuffer& b) -> py::buffer_info {
        return py::buffer_info(b.data(), 1, py::format_descriptor<uint8_t>::format(),
                               1, {(ssize_t)b.size()}, {(ssize_t)1}, /*readonly=*/true);
      })
      .def_property_readonly("nbytes", &PinnedBuffer::size)
      .def_property_readonly("oid", [](PinnedBuffer& b) { return py::bytes(b.oid()); });

  py::class_<ShmStore, std::shared_ptr<ShmStore>>(m, "ShmStore")
      .def_static("create", &ShmStore::Create, py::arg("path"),
                  py::arg("capacity"), py::arg("num_slots") = 65536)
      .def_static("open", &ShmStore::Open, py::arg("path"),
                  py::arg("timeout_s") = 30.0)
      .def("get_buffer", &ShmStore::GetBuffer, py::arg("oid"),
           py::arg("timeout_s") = -1.0)
      .def("create_object", &ShmStore::CreateObject)
      .def("seal", &ShmStore::Seal)
      .def("abort", &ShmStore::Abort)
      .def("get_object", &ShmStore::GetObject, py::arg("oid"),
           py::arg("timeout_s") = -1.0)
      .def("contains", &ShmStore::Contains)
      .def("release", &ShmStore::Release)
      .def("delete", &ShmStore::Delete)
      .def("stats", &ShmStore::Stats)
      .def("channel_create", &ShmStore::ChannelCreate)
      .def("channel_write", &ShmStore::ChannelWrite, py::arg("oid"),
           py::arg("data"), py::arg("timeout_s") = -1.0)
      .def("channel_read", &ShmStore::ChannelRead, py::arg("oid"),
           py::arg("last_version"), py::arg("timeout_s") = -1.0)
      .def("channel_close", &ShmStore::ChannelClose)
      .def("view_at", &ShmStore::ViewAt, py::arg("offset"), py::arg("size"),
           py::arg("writable") = false)
      .def("address_at", &ShmStore::AddressAt)
      .def_property_readonly("path", &ShmStore::path);
}
