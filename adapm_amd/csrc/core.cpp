// adapm_amd C++ core: HBM-slab parameter store + per-rank server with
// intent-driven replication/relocation, remote ops riding per-channel
// all-to-all-v sync rounds.
//
// Capability parity with the reference AdaPM (cited file:line into
// /root/reference/):
//  - store + per-key metadata + lock striping: coloc_kv_server_handle.h
//  - worker API (Pull/Push/Set/Intent/clock/Wait): coloc_kv_worker.h
//  - ownership directory (manager = key % world): addressbook.h:110
//  - replication/relocation protocol: sync_manager.h — re-designed: the
//    reference's per-message Van/Customer request machinery is collapsed
//    into two-phase batched all-to-all-v rounds run by one Python thread
//    per channel over torch.distributed (RCCL P2P over xGMI on GPU, gloo
//    on CPU). Remote Pull/Push requests, replica deltas, refreshes and
//    relocations all ride the same rounds.
//
// Concurrency model (differs from the reference's 16k host mutexes +
// single-receiver-thread design, because values live in HBM and value ops
// are async kernels):
//  - striped host mutexes guard per-key METADATA only (lock order:
//    stripe(k) may be taken alone or before a channel mutex; never the
//    reverse),
//  - all value kernels run on the rank's current stream, so slab-slot
//    reuse is ordered after prior reads/writes by stream order,
//  - an "inflight" counter closes the window between a worker's metadata
//    pass and its kernel launch: the sync thread quiesces it before
//    structural metadata changes (Server::quiesce),
//  - each channel's sync_* methods are called by exactly ONE thread (the
//    channel's sync loop); per-channel maps that only that thread touches
//    (reloc counters) need no locks,
//  - the CPU backend serializes value ops with one mutex (cpu_val_mu_) —
//    it is the test tier, not the perf path.
#include <torch/extension.h>
#include <ATen/Parallel.h>

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <cstring>
#include <limits>
#include <deque>
#include <map>
#include <queue>
#include <tuple>
#include <memory>
#include <mutex>
#include <random>
#include <thread>
#include <chrono>
#include <unordered_map>
#include <unordered_set>
#include <cstdio>
#include <vector>

#include "common.h"
#include "kernels.h"
#include "ops.h"

#ifdef ADAPM_WITH_HIP
#include <c10/hip/HIPStream.h>
#endif

extern "C" {
int adapm_host_arena_alloc(long long floats, int want_device_visible, void** host_ptr,
                           void** dev_ptr);
void adapm_host_arena_free(void* host_ptr, int was_device_visible);
}

namespace adapm {

namespace py = pybind11;

static void* current_stream(const torch::Device& dev) {
#ifdef ADAPM_WITH_HIP
  if (dev.is_cuda()) return (void*)c10::hip::getCurrentHIPStream(dev.index()).stream();
#endif
  return nullptr;
}

// ---------------------------------------------------------------- slab

// Flat float32 value arena with size-class free lists. Slot sizes are
// padded to a multiple of 4 floats so kernels can use float4.
struct Slab {
  torch::Tensor buf;
  float* data = nullptr;
  int64_t capacity = 0;
  int64_t bump = 0;
  std::unordered_map<int32_t, std::vector<int64_t>> freelists;
  std::mutex mu;
  std::atomic<int64_t> in_use{0};
  // host-spill arena: pinned host memory, device-visible (zero-copy over
  // PCIe). Used when the device arena is exhausted — "HBM as a cache over
  // pinned host memory" (BASELINE config 5). Offsets carry SPILL_BIT.
  bool is_cuda = false;
  void* host_raw = nullptr;          // host pointer (hipHostMalloc / malloc)
  float* host_host = nullptr;        // host-side view
  float* host_dev = nullptr;         // device-visible pointer
  int64_t host_capacity = 0;
  int64_t host_bump = 0;
  std::unordered_map<int32_t, std::vector<int64_t>> host_freelists;
  std::atomic<int64_t> host_in_use{0};

  static int32_t padded(int32_t len) { return (len + 3) & ~3; }

  void init(int64_t cap, const torch::Device& dev) {
    capacity = cap;
    is_cuda = dev.is_cuda();
    buf = torch::zeros({cap}, torch::TensorOptions().dtype(torch::kFloat32).device(dev));
    data = buf.data_ptr<float>();
  }

  void init_host(int64_t cap) {
    void *hp = nullptr, *dp = nullptr;
    int rc = adapm_host_arena_alloc(cap, is_cuda ? 1 : 0, &hp, &dp);
    if (rc != 0) throw std::runtime_error("host-spill arena allocation failed");
    host_raw = hp;
    host_host = (float*)hp;
    host_dev = (float*)dp;
    host_capacity = cap;
  }
  ~Slab() { adapm_host_arena_free(host_raw, is_cuda ? 1 : 0); }

  SlabBases bases() const { return SlabBases{data, is_cuda ? host_dev : host_host}; }

  // device-arena-only allocation attempt (spill-tier promotion): -1 if full
  int64_t try_alloc_device(int32_t len) {
    int32_t p = padded(len);
    std::lock_guard<std::mutex> g(mu);
    auto it = freelists.find(p);
    if (it != freelists.end() && !it->second.empty()) {
      int64_t off = it->second.back();
      it->second.pop_back();
      in_use += p;
      return off;
    }
    if (bump + p <= capacity) {
      int64_t off = bump;
      bump += p;
      in_use += p;
      return off;
    }
    return -1;
  }

  int64_t alloc(int32_t len) {
    int32_t p = padded(len);
    std::lock_guard<std::mutex> g(mu);
    auto it = freelists.find(p);
    if (it != freelists.end() && !it->second.empty()) {
      int64_t off = it->second.back();
      it->second.pop_back();
      in_use += p;
      return off;
    }
    if (bump + p <= capacity) {
      int64_t off = bump;
      bump += p;
      in_use += p;
      return off;
    }
    // device arena exhausted: spill to the host arena
    if (host_capacity > 0) {
      auto hit = host_freelists.find(p);
      if (hit != host_freelists.end() && !hit->second.empty()) {
        int64_t off = hit->second.back();
        hit->second.pop_back();
        host_in_use += p;
        return off | SPILL_BIT;
      }
      if (host_bump + p <= host_capacity) {
        int64_t off = host_bump;
        host_bump += p;
        host_in_use += p;
        return off | SPILL_BIT;
      }
    }
    throw std::runtime_error("adapm slab out of capacity (device and host-spill)");
  }

  bool poison = getenv("ADAPM_DEBUG_POISON") != nullptr;

  void free_(int64_t off, int32_t len) {
    int32_t p = padded(len);
    std::lock_guard<std::mutex> g(mu);
    if (off & SPILL_BIT) {
      host_freelists[p].push_back(off & ~SPILL_BIT);
      host_in_use -= p;
    } else {
      if (poison) {
        // debug mode: NaN-fill freed slots so use-after-free reads are
        // loud (stream-ordered, so this poisons only after prior readers)
        buf.narrow(0, off, p).fill_(std::numeric_limits<float>::quiet_NaN());
      }
      freelists[p].push_back(off);
      in_use -= p;
    }
  }
};

// ---------------------------------------------------------------- tickets

// Async-op completion tracking (replaces the reference Customer,
// customer.cc:31-77). A ticket completes when `received == expected`.
struct Ticket {
  int expected = 0;
  int received = 0;
  torch::Tensor out;             // device-side output (pull)
  torch::Tensor caller_out;      // caller's tensor if different device
  std::vector<int64_t> out_off;  // flat offsets per original key index
  std::vector<int32_t> out_len;
  std::string fail;              // non-empty: some records were NACKed; wait() throws
};

// pending outgoing record (remote op or forward), per channel
struct OutRec {
  int dest;
  int64_t code, key, f0, f1, f2;
  torch::Tensor payload;       // may be undefined
  std::vector<Key> keys;       // bulk records: keys (key field unused)
  std::vector<int64_t> aux;    // bulk pulls: out indices
};

// phase-B pending response with deferred payload gather from the slab
struct RespRec {
  int dest = 0;
  int64_t code = 0, key = 0, f0 = 0, f1 = 0, f2 = 0;
  int64_t slab_off = -1;    // gather source (refresh / pull resp / relocate)
  int32_t len = 0;
  bool free_after = false;  // relocation: free the slot after gathering
  torch::Tensor payload;    // alternative payload source
  std::vector<Key> keys;    // bulk responses
  std::vector<int64_t> aux;       // bulk pulls: out indices / bulk refresh: versions
  std::vector<int64_t> aux2;      // bulk refresh: (reloc_ctr<<8)|rflags per key
  std::vector<int64_t> slab_offs; // bulk responses: per-key gather sources
};

struct IntentReq {
  int wid;
  Clock start, end;
  std::vector<Key> keys;
};

struct ChannelState {
  std::mutex mu;
  std::deque<IntentReq> intent_queue;                                   // worker -> sync
  std::vector<IntentReq> future_intents;                                // not yet due
  // active local intents live in Server::intent_cnt_ (flat per-key
  // atomic counters — a key belongs to exactly one channel); the heap
  // below expires them against worker clocks
  // expiry min-heap: ONE entry per intent REQUEST (not per key — a
  // request covers thousands of keys; per-key heap pushes were a
  // measurable share of the round at churn). Popping an entry
  // decrements every key's counter.
  struct IntentExpiry {
    Clock end;
    int wid;
    std::shared_ptr<std::vector<Key>> keys;
    bool operator>(const IntentExpiry& o) const { return end > o.end; }
  };
  std::priority_queue<IntentExpiry, std::vector<IntentExpiry>, std::greater<>> intent_expiry;
  std::unordered_set<Key> replicas;                                     // local replica/stub keys
  std::unordered_map<Key, uint64_t> holders;    // owner side: ranks holding replicas
  std::deque<OutRec> out_queue;                 // pending remote ops + forwards
  std::vector<RespRec> responses;               // built in process, sent in respond
  std::unordered_map<Key, uint32_t> reloc_ctr;  // per-key relocation counter, travels
                                                // with ownership (sync thread only)
  std::unordered_map<Key, int64_t> reloc_round;  // round a key relocated IN (sync thread
                                                 // only): relocation cooldown, see below
  std::atomic<int64_t> rounds{0};
  // strong-WaitSync bookkeeping: a round is "globally idle" when NO rank
  // sent anything on this channel. idle2_events counts rounds that were
  // the >=2nd consecutive globally-idle round (sync thread only writes).
  int64_t idle_streak = 0;
  std::atomic<int64_t> idle2_events{0};
};

// ---------------------------------------------------------------- server

class Server {
 public:
  Server(int64_t num_keys, torch::Tensor lens, int rank, int world, int num_channels,
         int num_workers, std::string device, double capacity_factor, int techniques,
         bool location_caches, int64_t device_cap_floats = 0, int64_t host_spill_floats = 0,
         double sync_threshold = 0.0)
      : num_keys_(num_keys),
        rank_(rank),
        world_(world),
        nch_(num_channels),
        techniques_(techniques),
        use_loc_cache_(location_caches),
        sync_threshold_(sync_threshold),
        dev_(device) {
    TORCH_CHECK((nch_ & (nch_ - 1)) == 0, "num_channels must be a power of 2");
    {
      // metadata-pass pool: share the node's cores across the co-located
      // ranks (the driver runs world_ ranks per node), leave headroom
      // for the sync threads and torch's own pools
      int hw = (int)std::thread::hardware_concurrency();
      int nt = std::max(1, std::min(hw / std::max(1, world_) - 2, 31));
      if (const char* e = getenv("ADAPM_PASS_THREADS")) nt = std::max(1, atoi(e));
      pass_pool_.want_threads = nt;
    }
    log2ch_ = 0;
    while ((1 << log2ch_) < nch_) log2ch_++;

    lens = lens.to(torch::kInt32).contiguous();
    if (lens.numel() == 1) {
      uniform_len_ = lens.item<int32_t>();
    } else {
      TORCH_CHECK(lens.numel() == num_keys_, "value_lengths must have num_keys entries");
      uniform_len_ = -1;
      lens_.assign(lens.data_ptr<int32_t>(), lens.data_ptr<int32_t>() + num_keys_);
    }

    meta_ = std::vector<std::atomic<int64_t>>(num_keys_);
    version_ = std::vector<std::atomic<uint32_t>>(num_keys_);
    for (int64_t i = 0; i < num_keys_; ++i) {
      meta_[i].store(0, std::memory_order_relaxed);
      version_[i].store(0, std::memory_order_relaxed);
    }
    sync_loc_.assign(num_keys_, -1);
    if (use_loc_cache_) loc_cache_.assign(num_keys_, -1);
    int64_t n_managed = (num_keys_ + world_ - 1) / world_;
    owner_of_.assign(n_managed, rank_);  // initially every key lives at its manager
    mgr_reloc_ctr_.assign(n_managed, 0);

    // initial allocation: insert every key managed here
    // (reference coloc_kv_server.h:85-90)
    int64_t owned_floats = 0;
    for (Key k = rank_; k < num_keys_; k += world_) owned_floats += Slab::padded(len_of(k));
    int64_t cap = (int64_t)((double)owned_floats * capacity_factor) + (1 << 20);
    if (device_cap_floats > 0) cap = device_cap_floats;  // explicit HBM budget
    slab_.init(cap, dev_);
    if (host_spill_floats > 0) {
      slab_.init_host(host_spill_floats);
      // tiered store: track per-key access heat so rebalance_spill can
      // keep the hot set HBM-resident (HBM as a cache over pinned host)
      heat_.reset(new std::atomic<uint32_t>[num_keys_]);
      for (Key k = 0; k < num_keys_; ++k) heat_[k].store(0, std::memory_order_relaxed);
    }
    // if the initial allocation will not fit the device arena, identity
    // breaks immediately (spilled slots are not at identity offsets)
    if (owned_floats > cap) layout_identity_.store(false);
    for (Key k = rank_; k < num_keys_; k += world_) {
      int32_t l = len_of(k);
      meta_[k] = mpack(slab_.alloc(l), F_PRESENT | F_OWNER);
    }
    // slab is zero-initialized by torch::zeros; bump-fresh slots stay zero

    // flat per-key intent counters: has-intent checks are lock-free
    // array reads instead of probes into a multi-million-entry hash map
    // (the round's hottest lookup at high relocation churn)
    intent_cnt_.reset(new std::atomic<uint16_t>[num_keys_]);
    for (Key k = 0; k < num_keys_; ++k) intent_cnt_[k].store(0, std::memory_order_relaxed);
    channels_ = std::vector<ChannelState>(nch_);
    clocks_ = std::vector<std::atomic<Clock>>(std::max(1, num_workers));
    for (auto& c : clocks_) c = 0;
    locks_ = std::make_unique<std::mutex[]>(N_STRIPES);
  }

  // ------------------------------------------------ helpers

  // ---- packed per-key metadata: one atomic int64 = flag bits (0-3),
  // spill bit (4), has-replica bit (5), slab offset (bits 6+). A
  // flags+loc pair updates ATOMICALLY, which retires round-1's
  // loc-before-flags write-ordering rules; readers pay ONE cache miss
  // per key. loc values keep their external encoding (SPILL_BIT = bit
  // 62) at the pack/unpack boundary.
  static constexpr int64_t MSPILL = 16;
  static constexpr int64_t MFLAGS = F_PRESENT | F_OWNER | F_STUB | F_UPDATED | F_HASREP;
  static inline int64_t mpack(int64_t loc, int64_t flags) {
    if (loc < 0) return flags;
    return ((loc & ~SPILL_BIT) << 6) | ((loc & SPILL_BIT) ? MSPILL : 0) | flags;
  }
  static inline uint8_t mflags(int64_t m) { return (uint8_t)(m & MFLAGS); }
  static inline int64_t mloc(int64_t m) {
    return (int64_t)((uint64_t)m >> 6) | ((m & MSPILL) ? SPILL_BIT : 0);
  }

  inline bool on_default_stream() const {
#ifdef ADAPM_WITH_HIP
    if (dev_.is_cuda())
      return c10::hip::getCurrentHIPStream(dev_.index()) ==
             c10::hip::getDefaultHIPStream(dev_.index());
#endif
    return true;
  }

  inline int32_t len_of(Key k) const { return uniform_len_ >= 0 ? uniform_len_ : lens_[k]; }
  inline int manager_of(Key k) const { return (int)(k % world_); }
  inline int channel_of(Key k) const {
    if (nch_ == 1) return 0;
    return (int)(((uint32_t)((uint64_t)k * 2654435769ULL)) >> (32 - log2ch_));
  }
  inline std::mutex& stripe(Key k) { return locks_[(size_t)k % N_STRIPES]; }

  // believed current location of a key (reference addressbook.h:50-70)
  int directions(Key k) {
    if (meta_[k].load(std::memory_order_acquire) & F_OWNER) return rank_;
    if (manager_of(k) == rank_) return owner_of_[k / world_];
    if (use_loc_cache_) {
      int c = loc_cache_[k];
      if (c >= 0) return c;
    }
    return manager_of(k);
  }

  struct InflightGuard {
    Server* s;
    explicit InflightGuard(Server* sv) : s(sv) {
      // increment-then-check: if a spill rebalance is in progress, back
      // out and wait (it quiesces after raising migrating_, so a guard
      // acquired before the raise is waited for; one after waits here).
      for (;;) {
        s->inflight_.fetch_add(1, std::memory_order_acquire);
        if (!s->migrating_.load(std::memory_order_acquire)) break;
        s->inflight_.fetch_sub(1, std::memory_order_release);
        std::this_thread::yield();
      }
    }
    ~InflightGuard() { s->inflight_.fetch_sub(1, std::memory_order_release); }
  };

  // wait until no worker op is between its metadata pass and kernel launch
  void quiesce() {
    while (inflight_.load(std::memory_order_acquire) != 0) std::this_thread::yield();
  }

  // ------------------------------------------------ batched local ops

  struct HostBatch {
    std::vector<int64_t> src, dst;
    std::vector<int32_t> len;
    void add(int64_t s, int64_t d, int32_t l) {
      src.push_back(s);
      dst.push_back(d);
      len.push_back(l);
    }
    size_t size() const { return src.size(); }
  };

  // Persistent thread pool for the per-key metadata passes. ATen's
  // at::parallel_for is a header template: compiled WITHOUT -fopenmp
  // (as this extension is) its omp pragma is a no-op and the "parallel"
  // pass runs serial; compiling WITH -fopenmp loads ROCm's libomp next
  // to torch's libgomp and everything slows 2-4x (two spinning
  // runtimes). A tiny dedicated pool avoids both. The pass is memory-
  // latency-bound (~85 ns/key serial: two random cache misses over
  // GB-scale metadata arrays), so threads scale it well.
  struct PassPool {
    struct Task {
      const std::function<void(int64_t, int64_t)>* fn = nullptr;
      int64_t n = 0;
      std::atomic<int64_t>* next = nullptr;
      std::atomic<int>* done = nullptr;
      uint64_t gen = 0;
    };
    std::vector<std::thread> threads;
    std::mutex run_mu_;  // one run() at a time: task is a single slot
    std::mutex mu;
    std::condition_variable cv;
    Task task;
    uint64_t gen = 0;
    bool stop = false;

    ~PassPool() {
      {
        std::lock_guard<std::mutex> g(mu);
        stop = true;
      }
      cv.notify_all();
      for (auto& t : threads) t.join();
    }

    int want_threads = 0;  // set by Server ctor (cores / world, capped)

    void ensure_started() {
      if (!threads.empty()) return;
      int nt = want_threads;
      if (nt <= 0) {
        unsigned hw = std::thread::hardware_concurrency();
        nt = (int)std::min<unsigned>(hw > 2 ? hw - 2 : 1, 15);
      }
      for (int i = 0; i < nt; ++i)
        threads.emplace_back([this] {
          uint64_t seen = 0;
          for (;;) {
            Task t;
            {
              std::unique_lock<std::mutex> lk(mu);
              cv.wait(lk, [&] { return stop || gen != seen; });
              if (stop) return;
              seen = gen;
              t = task;
            }
            try {
              work(t);
            } catch (...) {
              // pass bodies only allocate; an OOM here surfaces on the
              // caller side as missing work is impossible (chunks the
              // worker claimed are lost) — treat as fatal store failure
              fprintf(stderr, "adapm: metadata-pass worker exception\n");
            }
          }
        });
    }

    static void work(const Task& t) {
      constexpr int64_t STEP = 4;
      try {
        for (;;) {
          int64_t b = t.next->fetch_add(STEP, std::memory_order_relaxed);
          if (b >= t.n) break;
          (*t.fn)(b, std::min(t.n, b + STEP));
        }
      } catch (...) {
        t.done->fetch_add(1, std::memory_order_acq_rel);
        throw;  // caller rethrows its own; worker threads must not leak a hang
      }
      t.done->fetch_add(1, std::memory_order_acq_rel);
    }

    // run fn over [0, n) chunks; caller participates. fn must not throw
    // on worker threads — callers validate inputs before entering.
    void run(int64_t n, const std::function<void(int64_t, int64_t)>& fn) {
      if (n <= 1) {
        if (n == 1) fn(0, 1);
        return;
      }
      // concurrent worker threads each bring their own pass; serialize
      // (they would otherwise overwrite the single task slot and hang)
      std::lock_guard<std::mutex> rg(run_mu_);
      ensure_started();
      std::atomic<int64_t> next{0};
      std::atomic<int> done{0};
      {
        std::lock_guard<std::mutex> g(mu);
        gen++;
        task = Task{&fn, n, &next, &done, gen};
      }
      cv.notify_all();
      int want = (int)threads.size() + 1;
      std::exception_ptr err;
      try {
        work(task);  // caller participates
      } catch (...) {
        err = std::current_exception();  // work() already counted this participant
      }
      // workers reference next/done on this stack frame — always drain
      while (done.load(std::memory_order_acquire) < want) std::this_thread::yield();
      if (err) std::rethrow_exception(err);
    }
  };

  struct DevBatch {
    torch::Tensor src_t;  // the packed staging blob (device or pinned host)
    const int64_t* aux_dev = nullptr;
    OpsBatch b;
  };

  // Stage a host batch for the device kernels with ONE allocation + ONE
  // H2D copy: [src i64 | dst i64 | aux i64 | len i32] packed into a
  // pinned blob (torch's caching host allocator recycles it), instead of
  // 3-4 separate pageable clones + copies. This was the largest C++ term
  // on the CTR spill config (t_todev, docs/NOTES_NEXT_ROUND.md lever c).
  DevBatch to_dev(const HostBatch& hb, const std::vector<int64_t>* aux = nullptr) {
    int64_t t0 = cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0;
    DevBatch d;
    const int64_t n = (int64_t)hb.size();
    const int64_t na = aux ? (int64_t)aux->size() : 0;
    const int64_t bytes = n * 8 * 2 + na * 8 + n * 4;
    auto blob = torch::empty({bytes}, torch::TensorOptions()
                                          .dtype(torch::kUInt8)
                                          .pinned_memory(dev_.is_cuda()));
    uint8_t* p = blob.data_ptr<uint8_t>();
    std::memcpy(p, hb.src.data(), n * 8);
    std::memcpy(p + n * 8, hb.dst.data(), n * 8);
    if (na) std::memcpy(p + n * 16, aux->data(), na * 8);
    std::memcpy(p + n * 16 + na * 8, hb.len.data(), n * 4);
    torch::Tensor dblob = dev_.is_cuda() ? blob.to(dev_, /*non_blocking=*/true) : blob;
    uint8_t* dp = dblob.data_ptr<uint8_t>();
    d.src_t = dblob;  // keeps the device blob alive
    d.b.src_off = (const int64_t*)dp;
    d.b.dst_off = (const int64_t*)(dp + n * 8);
    if (na) d.aux_dev = (const int64_t*)(dp + n * 16);
    d.b.lens = (const int32_t*)(dp + n * 16 + na * 8);
    d.b.n = (int)n;
    if (cpp_timing_)
      t_todev_ += std::chrono::steady_clock::now().time_since_epoch().count() - t0;
    return d;
  }

  void run_gather(const HostBatch& hb, torch::Tensor out) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb);
    if (dev_.is_cuda()) {
      ops_gather_gpu(slab_.bases(), d.b, out.data_ptr<float>(), current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_gather_cpu(slab_.bases(), d.b, out.data_ptr<float>());
    }
  }
  void run_scatter(const HostBatch& hb, torch::Tensor in, bool set) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb);
    auto in_c = in.is_contiguous() ? in : in.contiguous();
    if (dev_.is_cuda()) {
      ops_scatter_gpu(slab_.bases(), d.b, in_c.data_ptr<float>(), set, current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_scatter_cpu(slab_.bases(), d.b, in_c.data_ptr<float>(), set);
    }
  }
  void run_scatter_rmw(const HostBatch& hb, torch::Tensor in) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb);
    auto in_c = in.is_contiguous() ? in : in.contiguous();
    ops_scatter_rmw_gpu(slab_.bases(), d.b, in_c.data_ptr<float>(), current_stream(dev_));
  }
  void run_delta_sqnorm(const HostBatch& hb, const std::vector<int64_t>& sync_off,
                        torch::Tensor out) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb, &sync_off);
    if (dev_.is_cuda()) {
      ops_delta_sqnorm_gpu(slab_.bases(), d.b, d.aux_dev,
                           out.data_ptr<float>(), current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_delta_sqnorm_cpu(slab_.bases(), d.b, sync_off.data(), out.data_ptr<float>());
    }
  }
  void run_extract(const HostBatch& hb, const std::vector<int64_t>& sync_off, torch::Tensor out) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb, &sync_off);
    if (dev_.is_cuda()) {
      ops_extract_gpu(slab_.bases(), d.b, d.aux_dev, out.data_ptr<float>(),
                      current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_extract_cpu(slab_.bases(), d.b, sync_off.data(), out.data_ptr<float>());
    }
  }
  void run_refresh(const HostBatch& hb, const std::vector<int64_t>& sync_off, torch::Tensor in) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb, &sync_off);
    if (dev_.is_cuda()) {
      ops_refresh_gpu(slab_.bases(), d.b, d.aux_dev, in.data_ptr<float>(),
                      current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_refresh_cpu(slab_.bases(), d.b, sync_off.data(), in.data_ptr<float>());
    }
  }
  void run_zero(const HostBatch& hb) {
    if (hb.size() == 0) return;
    auto d = to_dev(hb);
    if (dev_.is_cuda()) {
      ops_zero_gpu(slab_.bases(), d.b, current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_zero_cpu(slab_.bases(), d.b);
    }
  }

  // identity-layout direct ops: offsets derived from keys on the device
  KeyBatch key_batch(const torch::Tensor& keys_dev) {
    KeyBatch b;
    b.keys = keys_dev.data_ptr<int64_t>();
    b.n = (int)keys_dev.numel();
    b.len = uniform_len_;
    b.plen = Slab::padded(uniform_len_);
    b.world = world_;
    b.rank = rank_;
    return b;
  }
  void run_gather_keys(const torch::Tensor& keys_cpu, torch::Tensor out) {
    if (dev_.is_cuda()) {
      auto kd = keys_cpu.to(dev_, /*non_blocking=*/true);
      ops_gather_keys_gpu(slab_.bases(), key_batch(kd), out.data_ptr<float>(), current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_gather_keys_cpu(slab_.bases(), key_batch(keys_cpu), out.data_ptr<float>());
    }
  }
  void run_scatter_keys(const torch::Tensor& keys_cpu, const torch::Tensor& in, bool set) {
    if (dev_.is_cuda()) {
      auto kd = keys_cpu.to(dev_, /*non_blocking=*/true);
      // NOTE: a sorted chunked scatter (ops_scatter_sorted_gpu) was
      // measured here and LOST to plain HW atomics on every workload
      // (the device sort + permuted reads cost more than hot-row atomic
      // contention saves) — keep the straightforward atomic path.
      ops_scatter_keys_gpu(slab_.bases(), key_batch(kd), in.data_ptr<float>(), set,
                           current_stream(dev_));
    } else {
      std::lock_guard<std::mutex> g(cpu_val_mu_);
      ops_scatter_keys_cpu(slab_.bases(), key_batch(keys_cpu), in.data_ptr<float>(), set);
    }
  }

  // spill-tier promotion candidates: remember which spilled keys were
  // touched (bounded; rebalance_spill consumes the list — this is what
  // makes rebalance O(touched), not O(num_keys))
  template <class Parts>
  void record_spill_touches(const Parts& parts) {
    if (!heat_) return;
    size_t add = 0;
    for (auto& P : parts) add += P.spilled.size();
    if (add == 0) return;
    std::lock_guard<std::mutex> g(spill_mu_);
    constexpr size_t CAP = 1 << 22;
    for (auto& P : parts) {
      if (spill_touched_.size() >= CAP) break;
      spill_touched_.insert(spill_touched_.end(), P.spilled.begin(), P.spilled.end());
    }
  }

  // ------------------------------------------------ worker API

  // Pull: local fast path returns -1 with the gather already enqueued on
  // the caller's stream (reference coloc_kv_worker.h:253-318 contract).
  int64_t pull(int wid, torch::Tensor keys, torch::Tensor vals) {
    (void)wid;
    check_not_failed();
    check_keys(keys);
    TORCH_CHECK(vals.is_contiguous() && vals.scalar_type() == torch::kFloat32,
                "vals must be contiguous float32");
    check_val_size(keys, vals.numel(), "pull");
    torch::Tensor vals_dev =
        vals.device() == dev_ ? vals
                              : torch::empty({vals.numel()},
                                             torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
    int64_t n = keys.numel();
    const int64_t* kp = keys.data_ptr<int64_t>();

    HostBatch local;
    struct Remote {
      Key k;
      int64_t out_index;
    };
    std::vector<Remote> remote;
    std::vector<int64_t> out_off(n);
    std::vector<int32_t> out_len(n);
    int64_t cum = 0;
    auto tick = [&]() { return cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0; };
    int64_t tp0 = tick();
    {
      InflightGuard g(this);
      if (layout_identity_.load(std::memory_order_acquire) && uniform_len_ >= 0) {
        // fast path: zero host per-key work — the device kernel derives
        // every offset from the key (see layout_identity_). The host only
        // scans for non-owned keys (pure arithmetic, nothing when world=1).
        const int32_t l = uniform_len_;
        check_key_range(keys);
        for (int64_t i = 0; i < n; ++i) {
          if (world_ > 1 && kp[i] % world_ != rank_) remote.push_back({kp[i], i});
        }
        if (!remote.empty()) {
          for (int64_t i = 0; i < n; ++i) {
            out_off[i] = i * (int64_t)l;
            out_len[i] = l;
          }
        }
        if (cpp_timing_) { t_pass_ += tick() - tp0; tp0 = tick(); }
        run_gather_keys(keys, vals_dev);
        if (cpp_timing_) { t_launch_ += tick() - tp0; t_calls_++; }
        stat_pull_local_ += n - (int64_t)remote.size();
        stat_pull_keys_ += n;
      } else {
        // lock-free parallel metadata pass (atomic reads; see meta rules)
        if (uniform_len_ >= 0) {
          const int32_t l = uniform_len_;
          for (int64_t i = 0; i < n; ++i) {
            TORCH_CHECK((uint64_t)kp[i] < (uint64_t)num_keys_, "key out of range: ", kp[i]);
            out_off[i] = i * (int64_t)l;
            out_len[i] = l;
          }
        } else {
          for (int64_t i = 0; i < n; ++i) {
            TORCH_CHECK((uint64_t)kp[i] < (uint64_t)num_keys_, "key out of range: ", kp[i]);
            int32_t l = len_of(kp[i]);
            out_off[i] = cum;
            out_len[i] = l;
            cum += l;
          }
        }
        constexpr int64_t G = 2048;
        int64_t nchunks = (n + G - 1) / G;
        struct Part {
          HostBatch local;
          std::vector<Remote> remote;
          std::vector<Key> spilled;
          int64_t n_repl = 0;
        };
        std::vector<Part> parts(nchunks);
        pass_pool_.run(nchunks, [&](int64_t c0, int64_t c1) {
          for (int64_t c = c0; c < c1; ++c) {
            Part& P = parts[c];
            int64_t e = std::min(n, (c + 1) * G);
            for (int64_t i = c * G; i < e; ++i) {
              Key k = kp[i];
              int64_t m = meta_[k].load(std::memory_order_acquire);  // ONE miss: flags+loc
              uint8_t f = mflags(m);
              if ((f & F_PRESENT) && !(f & F_STUB)) {
                int64_t off = mloc(m);
                P.local.add(off, out_off[i], out_len[i]);
                if (!(f & F_OWNER)) P.n_repl++;
                if (heat_) {
                  heat_[k].fetch_add(1, std::memory_order_relaxed);
                  if (off & SPILL_BIT) P.spilled.push_back(k);
                }
                if (locality_stats_) {
                  key_accesses_[k].fetch_add(1, std::memory_order_relaxed);
                  key_local_[k].fetch_add(1, std::memory_order_relaxed);
                }
              } else {
                P.remote.push_back({k, i});
                if (locality_stats_) key_accesses_[k].fetch_add(1, std::memory_order_relaxed);
              }
            }
          }
        });
        for (auto& P : parts) {
          local.src.insert(local.src.end(), P.local.src.begin(), P.local.src.end());
          local.dst.insert(local.dst.end(), P.local.dst.begin(), P.local.dst.end());
          local.len.insert(local.len.end(), P.local.len.begin(), P.local.len.end());
          remote.insert(remote.end(), P.remote.begin(), P.remote.end());
          stat_pull_replica_ += P.n_repl;
        }
        record_spill_touches(parts);
        stat_pull_local_ += n - (int64_t)remote.size();
        stat_pull_keys_ += n;
        if (cpp_timing_) { t_pass_ += tick() - tp0; tp0 = tick(); }
        run_gather(local, vals_dev);
        if (cpp_timing_) { t_launch_ += tick() - tp0; t_calls_++; }
      }
    }
    stat_pulls_ += 1;

    if (remote.empty()) {
      if (vals_dev.data_ptr() != vals.data_ptr()) vals.view({-1}).copy_(vals_dev);
      return -1;
    }
    int64_t ts;
    {
      std::lock_guard<std::mutex> g(tickets_mu_);
      ts = next_ts_++;
      auto t = std::make_unique<Ticket>();
      t->expected = (int)remote.size();
      t->out = vals_dev;
      if (vals_dev.data_ptr() != vals.data_ptr()) t->caller_out = vals;
      t->out_off = std::move(out_off);
      t->out_len = std::move(out_len);
      tickets_[ts] = std::move(t);
    }
    if (uniform_len_ >= 0) {
      // bulk requests: one record per (channel, destination)
      std::map<std::pair<int, int>, std::pair<std::vector<Key>, std::vector<int64_t>>> groups;
      for (auto& r : remote) {
        auto& g = groups[{channel_of(r.k), directions(r.k)}];
        g.first.push_back(r.k);
        g.second.push_back(r.out_index);
      }
      for (auto& [cd, g] : groups) {
        OutRec rec{cd.second, M_PULL_REQ_BULK, 0, rank_, ts, 0, {}};
        rec.keys = std::move(g.first);
        rec.aux = std::move(g.second);
        enqueue_out(cd.first, std::move(rec));
      }
    } else {
      for (auto& r : remote) {
        enqueue_out(channel_of(r.k),
                    OutRec{directions(r.k), M_PULL_REQ, r.k, rank_, ts, r.out_index, {}});
      }
    }
    return ts;
  }

  // Push (additive) / Set (overwrite). Local fast path: merge into owned
  // value or replica (marks it updated); remote keys become requests.
  // Set on a replica is routed to the owner (the replica refreshes later),
  // preserving delta semantics.
  int64_t push(int wid, torch::Tensor keys, torch::Tensor vals, bool set_mode) {
    (void)wid;
    check_not_failed();
    check_keys(keys);
    TORCH_CHECK(vals.scalar_type() == torch::kFloat32, "vals must be float32");
    check_val_size(keys, vals.numel(), "push");
    auto tick = [&]() { return cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0; };
    int64_t tp0 = tick();
    torch::Tensor vals_dev = vals.device() == dev_ ? vals : vals.to(dev_);
    if (!vals_dev.is_contiguous()) vals_dev = vals_dev.contiguous();
    torch::Tensor flat = vals_dev.view({-1});
    int64_t n = keys.numel();
    const int64_t* kp = keys.data_ptr<int64_t>();
    if (cpp_timing_) { t_misc_ += tick() - tp0; }

    HostBatch merge, assign;
    struct Remote {
      Key k;
      int64_t off;
      int32_t len;
    };
    std::vector<Remote> remote;
    int64_t cum = 0;
    {
      InflightGuard g(this);
      if (layout_identity_.load(std::memory_order_acquire) && uniform_len_ >= 0) {
        // fast path: device kernel derives offsets from keys; host only
        // scans for non-owned keys (pure arithmetic, nothing at world=1).
        // Version bumps are skipped here: handle_owner_delta clears
        // layout_identity_ when the first replica of one of our keys is
        // granted, so while the flag holds versions are unobserved.
        const int32_t l = uniform_len_;
        check_key_range(keys);
        for (int64_t i = 0; i < n; ++i) {
          if (world_ > 1 && kp[i] % world_ != rank_) remote.push_back({kp[i], i * (int64_t)l, l});
        }
        if (cpp_timing_) { int64_t t1 = tick(); t_pass_ += t1 - tp0; tp0 = t1; }
        run_scatter_keys(keys, flat, set_mode);
        if (cpp_timing_) { t_launch_ += tick() - tp0; t_calls_++; }
        stat_push_local_ += n - (int64_t)remote.size();
        stat_push_keys_ += n;
      } else {
        // lock-free parallel metadata pass (atomic reads; see meta rules)
        std::vector<int64_t> offs(n);
        if (uniform_len_ >= 0) {
          for (int64_t i = 0; i < n; ++i) {
            TORCH_CHECK((uint64_t)kp[i] < (uint64_t)num_keys_, "key out of range: ", kp[i]);
            offs[i] = i * (int64_t)uniform_len_;
          }
        } else {
          for (int64_t i = 0; i < n; ++i) {
            TORCH_CHECK((uint64_t)kp[i] < (uint64_t)num_keys_, "key out of range: ", kp[i]);
            offs[i] = cum;
            cum += len_of(kp[i]);
          }
        }
        constexpr int64_t G = 2048;
        int64_t nchunks = (n + G - 1) / G;
        struct Part {
          HostBatch merge, assign, merge_spill;
          std::vector<Remote> remote;
          std::vector<Key> spilled;
          int64_t n_repl = 0;
        };
        std::vector<Part> parts(nchunks);
        pass_pool_.run(nchunks, [&](int64_t c0, int64_t c1) {
          for (int64_t c = c0; c < c1; ++c) {
            Part& P = parts[c];
            int64_t e = std::min(n, (c + 1) * G);
            for (int64_t i = c * G; i < e; ++i) {
              Key k = kp[i];
              int32_t l = len_of(k);
              int64_t m = meta_[k].load(std::memory_order_acquire);  // ONE miss: flags+loc
              uint8_t f = mflags(m);
              if ((f & F_PRESENT) && (f & F_OWNER)) {
                int64_t off = mloc(m);
                // spilled merges go through the non-atomic RMW kernel
                // (PCIe atomics are ~144x slower; dups re-routed below)
                // The non-atomic RMW merge relies on STREAM ORDER to
                // serialize concurrent merges to one slot: safe only
                // when every worker launches on the device's default
                // stream (our convention). A worker on a custom stream
                // falls back to the atomic path.
                bool spill_merge = heat_ && (off & SPILL_BIT) && !set_mode && dev_.is_cuda() &&
                                   on_default_stream();
                if (spill_merge)
                  P.merge_spill.add(off, offs[i], l);
                else
                  (set_mode ? P.assign : P.merge).add(off, offs[i], l);
                // version bumps are observable only once a replica was
                // granted (F_HASREP) — unreplicated keys skip the touch
                if (f & F_HASREP) version_[k].fetch_add(1, std::memory_order_relaxed);
                if (heat_) {
                  heat_[k].fetch_add(1, std::memory_order_relaxed);
                  if (off & SPILL_BIT) P.spilled.push_back(k);
                }
              } else if ((f & F_PRESENT) && !set_mode) {
                // replica/stub: merge locally, flush at next sync round
                P.merge.add(mloc(m), offs[i], l);
                meta_[k].fetch_or((int64_t)F_UPDATED);
                P.n_repl++;
              } else {
                P.remote.push_back({k, offs[i], l});
              }
            }
          }
        });
        for (auto& P : parts) {
          std::pair<HostBatch*, HostBatch*> prs[2] = {{&merge, &P.merge}, {&assign, &P.assign}};
          for (auto& pr : prs) {
            pr.first->src.insert(pr.first->src.end(), pr.second->src.begin(), pr.second->src.end());
            pr.first->dst.insert(pr.first->dst.end(), pr.second->dst.begin(), pr.second->dst.end());
            pr.first->len.insert(pr.first->len.end(), pr.second->len.begin(), pr.second->len.end());
          }
          remote.insert(remote.end(), P.remote.begin(), P.remote.end());
          stat_push_replica_ += P.n_repl;
        }
        record_spill_touches(parts);
        stat_push_local_ += n - (int64_t)remote.size();
        stat_push_keys_ += n;
        // spilled merges: unique destinations use the RMW kernel; a slot
        // hit twice in this batch falls back to the atomic path
        HostBatch rmw;
        {
          HostBatch all_spill;
          for (auto& P : parts) {
            all_spill.src.insert(all_spill.src.end(), P.merge_spill.src.begin(), P.merge_spill.src.end());
            all_spill.dst.insert(all_spill.dst.end(), P.merge_spill.dst.begin(), P.merge_spill.dst.end());
            all_spill.len.insert(all_spill.len.end(), P.merge_spill.len.begin(), P.merge_spill.len.end());
          }
          if (all_spill.size()) {
            std::unordered_set<int64_t> seen;
            seen.reserve(all_spill.size() * 2);
            for (size_t i = 0; i < all_spill.size(); ++i) {
              if (seen.insert(all_spill.src[i]).second)
                rmw.add(all_spill.src[i], all_spill.dst[i], all_spill.len[i]);
              else
                merge.add(all_spill.src[i], all_spill.dst[i], all_spill.len[i]);
            }
          }
        }
        if (cpp_timing_) { int64_t t1 = tick(); t_pass_ += t1 - tp0; tp0 = t1; }
        run_scatter(merge, flat, false);
        run_scatter(assign, flat, true);
        run_scatter_rmw(rmw, flat);
        if (cpp_timing_) { t_launch_ += tick() - tp0; t_calls_++; }
      }
    }
    stat_pushes_ += 1;

    if (remote.empty()) return -1;
    int64_t ts;
    {
      std::lock_guard<std::mutex> g(tickets_mu_);
      ts = next_ts_++;
      auto t = std::make_unique<Ticket>();
      t->expected = (int)remote.size();
      tickets_[ts] = std::move(t);
    }
    if (uniform_len_ >= 0) {
      const int32_t l = uniform_len_;
      std::map<std::pair<int, int>, std::pair<std::vector<Key>, std::vector<int64_t>>> groups;
      for (auto& r : remote) {
        auto& g = groups[{channel_of(r.k), directions(r.k)}];
        g.first.push_back(r.k);
        g.second.push_back(r.off / l);  // row index into flat
      }
      auto rows = flat.view({n, (int64_t)l});
      for (auto& [cd, g] : groups) {
        auto idx = torch::from_blob(g.second.data(), {(int64_t)g.second.size()},
                                    torch::TensorOptions().dtype(torch::kInt64))
                       .clone();
        if (dev_.is_cuda()) idx = idx.to(dev_, true);
        OutRec rec{cd.second, M_PUSH_REQ_BULK, 0, rank_, ts,
                   set_mode ? (1LL << 32) : 0, rows.index_select(0, idx).reshape({-1})};
        rec.keys = std::move(g.first);
        enqueue_out(cd.first, std::move(rec));
      }
    } else {
      for (auto& r : remote) {
        torch::Tensor pay = flat.narrow(0, r.off, r.len).clone();
        enqueue_out(channel_of(r.k),
                    OutRec{directions(r.k), set_mode ? M_SET_REQ : M_PUSH_REQ, r.k, rank_, ts, 0, pay});
      }
    }
    return ts;
  }

  // PullIfLocal: all-or-nothing local pull (reference coloc_kv_worker.h)
  bool pull_if_local(torch::Tensor keys, torch::Tensor vals) {
    check_keys(keys);
    check_val_size(keys, vals.numel(), "pull_if_local");
    int64_t n = keys.numel();
    const int64_t* kp = keys.data_ptr<int64_t>();
    torch::Tensor vals_dev =
        vals.device() == dev_ ? vals
                              : torch::empty({vals.numel()},
                                             torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
    HostBatch local;
    int64_t cum = 0;
    {
      InflightGuard g(this);
      for (int64_t i = 0; i < n; ++i) {
        Key k = kp[i];
        int32_t l = len_of(k);
        {
          int64_t m = meta_[k].load(std::memory_order_acquire);
          uint8_t f = mflags(m);
          if (!((f & F_PRESENT) && !(f & F_STUB))) return false;
          local.add(mloc(m), cum, l);
        }
        cum += l;
      }
      run_gather(local, vals_dev);
    }
    if (vals_dev.data_ptr() != vals.data_ptr()) vals.view({-1}).copy_(vals_dev);
    return true;
  }

  bool is_local(Key k) {
    uint8_t f = mflags(meta_[k].load(std::memory_order_acquire));
    return (f & F_PRESENT) && !(f & F_STUB);
  }

  // Intent: announce access to keys in clock window [start, end)
  // (reference coloc_kv_worker.h:368-408). No-op on a single node.
  void intent(int wid, torch::Tensor keys, Clock start, Clock end) {
    check_keys(keys);
    check_key_range(keys);
    if (world_ == 1) return;  // single node: intents are no-ops (reference
                              // coloc_kv_worker.h:728)
    if (end == 0) end = start + 1;
    int64_t n = keys.numel();
    const int64_t* kp = keys.data_ptr<int64_t>();
    std::vector<std::vector<Key>> per_ch(nch_);
    for (int64_t i = 0; i < n; ++i) per_ch[channel_of(kp[i])].push_back(kp[i]);
    for (int c = 0; c < nch_; ++c) {
      if (per_ch[c].empty()) continue;
      std::lock_guard<std::mutex> g(channels_[c].mu);
      channels_[c].intent_queue.push_back(IntentReq{wid, start, end, std::move(per_ch[c])});
    }
  }

  void advance_clock(int wid) { clocks_[wid]++; }
  Clock current_clock(int wid) { return clocks_[wid].load(); }
  std::vector<Clock> worker_clocks() {
    std::vector<Clock> v;
    for (auto& c : clocks_) v.push_back(c.load());
    return v;
  }

  // ------------------------------------------------ tickets / waiting

  bool is_finished(int64_t ts) {
    if (ts < 0) return true;
    std::lock_guard<std::mutex> g(tickets_mu_);
    return tickets_.find(ts) == tickets_.end();
  }

  // ops issued after a transport failure raise instead of creating
  // tickets that can never complete (the sync threads are dead)
  inline void check_not_failed() {
    if (failed_flag_.load(std::memory_order_relaxed)) {
      std::lock_guard<std::mutex> g(tickets_mu_);
      TORCH_CHECK(false, "adapm server failed: ", failed_reason_);
    }
  }

  void wait(int64_t ts) {
    if (ts < 0) return;
    std::unique_lock<std::mutex> g(tickets_mu_);
    tickets_cv_.wait(g, [&] {
      return tickets_.find(ts) == tickets_.end() || failed_flag_.load(std::memory_order_relaxed);
    });
    TORCH_CHECK(!failed_flag_.load(std::memory_order_relaxed),
                "adapm server failed while waiting: ", failed_reason_);
    auto it = failed_tickets_.find(ts);
    if (it != failed_tickets_.end()) {
      std::string why = std::move(it->second);
      failed_tickets_.erase(it);
      TORCH_CHECK(false, "adapm op ", ts, " failed: ", why);
    }
  }

  void wait_all() {
    std::unique_lock<std::mutex> g(tickets_mu_);
    tickets_cv_.wait(g, [&] {
      return tickets_.empty() || failed_flag_.load(std::memory_order_relaxed);
    });
    TORCH_CHECK(!failed_flag_.load(std::memory_order_relaxed),
                "adapm server failed while waiting: ", failed_reason_);
    if (!failed_tickets_.empty()) {
      std::string why = failed_tickets_.begin()->second;
      size_t n = failed_tickets_.size();
      failed_tickets_.clear();
      TORCH_CHECK(false, "adapm: ", n, " async op(s) failed, first: ", why);
    }
  }

  // WaitSync support (reference coloc_kv_worker.h:517-550): callers grab
  // round_counts() then wait_rounds(counts + 2).
  std::vector<int64_t> round_counts() {
    std::vector<int64_t> v;
    for (auto& c : channels_) v.push_back(c.rounds.load());
    return v;
  }
  void wait_rounds(std::vector<int64_t> targets) {
    std::unique_lock<std::mutex> g(rounds_mu_);
    rounds_cv_.wait(g, [&] {
      if (failed_flag_.load(std::memory_order_relaxed)) return true;
      for (int c = 0; c < nch_; ++c)
        if (channels_[c].rounds.load() < targets[c]) return false;
      return true;
    });
    if (failed_flag_.load(std::memory_order_relaxed)) {
      std::lock_guard<std::mutex> tg(tickets_mu_);
      TORCH_CHECK(false, "adapm server failed: ", failed_reason_);
    }
  }

  // Strong WaitSync (globally-idle rounds): wait until each channel has
  // completed a round that was the 2nd consecutive round in which NO rank
  // had outbound traffic — at that point every in-flight delta, forward
  // and refresh has drained, so the caller observes all prior pushes
  // (stronger than the reference's fixed 2-round window, which forwarded
  // deltas can escape under relocation churn).
  std::vector<int64_t> idle_counts() {
    std::vector<int64_t> v;
    for (auto& c : channels_) v.push_back(c.idle2_events.load());
    return v;
  }
  void wait_idle(std::vector<int64_t> targets) {
    std::unique_lock<std::mutex> g(rounds_mu_);
    rounds_cv_.wait(g, [&] {
      if (failed_flag_.load(std::memory_order_relaxed)) return true;
      for (int c = 0; c < nch_; ++c)
        if (channels_[c].idle2_events.load() < targets[c]) return false;
      return true;
    });
    if (failed_flag_.load(std::memory_order_relaxed)) {
      std::lock_guard<std::mutex> tg(tickets_mu_);
      TORCH_CHECK(false, "adapm server failed: ", failed_reason_);
    }
  }

  // ------------------------------------------------ sync round: phase A out

  void enqueue_out(int ch, OutRec r) {
    std::lock_guard<std::mutex> g(channels_[ch].mu);
    channels_[ch].out_queue.push_back(std::move(r));
  }

  void set_intent_ahead(Clock ahead) { intent_ahead_ = ahead; }

  // Phase A collect: drain intents, create replica stubs, expire intents /
  // drop replicas, extract replica deltas, drain pending remote ops; build
  // per-dest messages. Returns [(dest, meta[n,5] int64 cpu, payload f32 dev)].
  // MUST be called by the channel's single sync thread.
  std::vector<std::tuple<int, torch::Tensor, torch::Tensor>> sync_collect(int ch) {
    ChannelState& C = channels_[ch];
    auto tick = [&]() { return cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0; };
    int64_t tc0 = tick();
    std::deque<IntentReq> intents_in;
    std::deque<OutRec> ops_out;
    {
      std::lock_guard<std::mutex> g(C.mu);
      intents_in.swap(C.intent_queue);
      ops_out.swap(C.out_queue);
      for (auto& fi : C.future_intents) intents_in.push_back(std::move(fi));
      C.future_intents.clear();
    }

    // 1. register new intents (reference sync_manager.h registerNewIntents).
    // Stub slots are ZEROED BEFORE the key is published (flags set): a
    // worker push that lands right after publication would otherwise be
    // wiped by a later zero (exact-sum loss, found by the stress hunt).
    // Only this channel's thread creates stubs for its keys, so the
    // absent-check before allocation cannot race another creator.
    HostBatch zero_batch;
    struct NewStub {
      Key k;
      int64_t v_off, s_off;
    };
    std::vector<NewStub> new_stubs;
    std::unordered_set<Key> new_stub_keys;
    std::vector<IntentReq> due;
    for (auto& req : intents_in) {
      Clock now = clocks_[req.wid].load();
      if (req.end <= now) continue;  // already expired
      if (req.start > now + intent_ahead_) {
        std::lock_guard<std::mutex> g(C.mu);
        C.future_intents.push_back(std::move(req));
        continue;
      }
      for (Key k : req.keys) {
        if (!(meta_[k].load(std::memory_order_acquire) & F_PRESENT)) {
          if (!new_stub_keys.insert(k).second) continue;  // dedup within round
          int32_t l = len_of(k);
          layout_identity_.store(false, std::memory_order_release);
          int64_t v_off = slab_.alloc(l);
          int64_t s_off = slab_.alloc(l);
          zero_batch.add(0, v_off, l);  // freelist reuse leaves stale data
          zero_batch.add(0, s_off, l);
          new_stubs.push_back({k, v_off, s_off});
        }
      }
      due.push_back(std::move(req));
    }
    run_zero(zero_batch);  // CPU: inline; GPU: stream-ordered before any
                           // later worker merge kernel on these slots
    for (auto& ns : new_stubs) {
      std::lock_guard<std::mutex> lk(stripe(ns.k));
      sync_loc_[ns.k] = ns.s_off;
      version_[ns.k] = 0;
      meta_[ns.k].store(mpack(ns.v_off, F_PRESENT | F_STUB), std::memory_order_release);
      trace_event(ns.k, "REPLICA_SETUP");
    }
    {
      std::lock_guard<std::mutex> g(C.mu);
      for (auto& req : due) {
        for (Key k : req.keys) intent_cnt_[k].fetch_add(1, std::memory_order_relaxed);
        C.intent_expiry.push(ChannelState::IntentExpiry{
            req.end, req.wid, std::make_shared<std::vector<Key>>(std::move(req.keys))});
      }
      for (auto& ns : new_stubs) C.replicas.insert(ns.k);
    }

    if (cpp_timing_) { int64_t t1 = tick(); t_sy_intents_ += t1 - tc0; tc0 = t1; }
    // 2. expire due intents (heap pop — no full-map sweep) and snapshot
    // the replica set
    std::vector<Key> replica_snapshot;
    {
      std::lock_guard<std::mutex> g(C.mu);
      while (!C.intent_expiry.empty()) {
        const auto& top = C.intent_expiry.top();
        if (top.end > clocks_[top.wid].load()) break;
        auto keys = top.keys;
        C.intent_expiry.pop();
        for (Key k : *keys) intent_cnt_[k].fetch_sub(1, std::memory_order_relaxed);
      }
      replica_snapshot.assign(C.replicas.begin(), C.replicas.end());
    }

    // 2b. sync threshold (reference --sys.sync.threshold,
    // readAndPotentiallyDropReplica sync_manager.h:601-662): updated
    // replicas whose pending-delta L2 norm is below the threshold skip
    // this round — UPDATED stays set and sync_state untouched, so the
    // delta keeps accumulating and ships later (or with the drop, which
    // always carries a payload). The norm is computed on-device; the
    // readback is the round's only sync point and only exists when a
    // threshold is configured.
    std::unordered_set<Key> below_threshold;
    if (sync_threshold_ > 0.0) {
      HostBatch nb;
      std::vector<int64_t> nsync;
      std::vector<Key> cand;
      for (Key k : replica_snapshot) {
        if (intent_cnt_[k].load(std::memory_order_relaxed) == 0)
          continue;  // would drop: always ships
        std::lock_guard<std::mutex> lk(stripe(k));
        int64_t m = meta_[k].load();
        uint8_t f = mflags(m);
        if (!(f & F_PRESENT) || (f & F_OWNER)) continue;
        if (!(f & F_UPDATED) || (f & F_STUB)) continue;
        nb.add(mloc(m), (int64_t)cand.size(), len_of(k));
        nsync.push_back(sync_loc_[k]);
        cand.push_back(k);
      }
      if (!cand.empty()) {
        quiesce();  // norms must follow in-flight push kernels
        auto norms = torch::empty({(int64_t)cand.size()},
                                  torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
        run_delta_sqnorm(nb, nsync, norms);
        auto h = norms.cpu();
        const float* hp = h.data_ptr<float>();
        float thr2 = (float)(sync_threshold_ * sync_threshold_);
        for (size_t i = 0; i < cand.size(); ++i)
          if (hp[i] < thr2) below_threshold.insert(cand[i]);
      }
    }

    // 3. per replica: extract delta / drop (reference readAndPotentiallyDropReplica)
    struct DeltaRec {
      Key k;
      int64_t f0, f1;
      int64_t val_off, sync_off;
      int32_t len;
    };
    std::vector<DeltaRec> deltas;
    std::vector<std::pair<int64_t, int32_t>> frees;
    for (Key k : replica_snapshot) {
      bool has_intent = intent_cnt_[k].load(std::memory_order_relaxed) > 0;
      bool erase_from_replicas = false;
      {
        std::lock_guard<std::mutex> lk(stripe(k));
        int64_t m = meta_[k].load();
        uint8_t f = mflags(m);
        if (!(f & F_PRESENT) || (f & F_OWNER)) {
          erase_from_replicas = true;  // became owner via relocation
        } else {
          bool updated = (f & F_UPDATED) && !below_threshold.count(k);
          bool is_new = f & F_STUB;
          bool drop = !has_intent && !is_new;
          // dropped replicas ALWAYS carry a payload: a concurrent local
          // push between our flags read and the absent-store would
          // otherwise be lost (its kernel lands before our extract by
          // quiesce + stream order, so the extracted delta captures it)
          bool payload = updated || drop;
          int64_t fl = (payload ? D_HAS_PAYLOAD : 0) | (drop ? D_DROPPING : 0) |
                       (has_intent ? D_WANT_REFRESH : 0) | (is_new ? D_NEW : 0);
          deltas.push_back(DeltaRec{k, (int64_t)version_[k].load(), fl,
                                    payload ? mloc(m) : -1, sync_loc_[k], len_of(k)});
          if (updated && !drop) meta_[k].fetch_and(~(int64_t)F_UPDATED);
          if (drop) {
            layout_identity_.store(false, std::memory_order_release);
            frees.push_back({mloc(m), len_of(k)});
            frees.push_back({sync_loc_[k], len_of(k)});
            meta_[k].store(0);
            sync_loc_[k] = -1;
            erase_from_replicas = true;
            stat_drops_ += 1;
            trace_event(k, "REPLICA_DROP");
          }
        }
      }
      if (erase_from_replicas) {
        std::lock_guard<std::mutex> g(C.mu);
        C.replicas.erase(k);
      }
    }
    if (cpp_timing_) { int64_t t1 = tick(); t_sy_replicas_ += t1 - tc0; tc0 = t1;
                       n_sy_deltas_ += (int64_t)deltas.size(); }
    // no worker op may still be between its metadata read and kernel
    // launch: (a) dropped slots must not be reused under it, (b) the
    // delta-extract kernels must be enqueued AFTER any concurrent push
    // kernel whose UPDATED bit we just consumed.
    if (!frees.empty() || !deltas.empty()) quiesce();

    // 4. build per-destination messages
    struct Msg {
      std::vector<int64_t> meta;
      HostBatch gathers;
      std::vector<std::pair<torch::Tensor, int64_t>> copies;
      HostBatch extracts;
      std::vector<int64_t> extract_sync;
      int64_t payload_floats = 0;
    };
    std::unordered_map<int, Msg> msgs;
    auto add_rec = [&](int dest, int64_t code, Key k, int64_t f0, int64_t f1, int64_t f2) -> Msg& {
      Msg& m = msgs[dest];
      m.meta.insert(m.meta.end(), {code, k, f0, f1, f2});
      return m;
    };

    stat_replica_records_ += (int64_t)deltas.size();
    if (uniform_len_ >= 0) {
      // BULK delta emission: one M_DELTA_BULK record per (dest,
      // has-payload) instead of a 5-word record per replica — the
      // per-key record parse + response-object cost dominated round
      // host time at relocation churn.
      struct BD {
        std::vector<Key> keys;
        std::vector<int64_t> info, offs, soffs;
      };
      std::unordered_map<int, std::array<BD, 2>> bulk;
      const int32_t l = uniform_len_;
      for (auto& d : deltas) {
        int dest = directions(d.k);
        if (dest == rank_) continue;  // raced with becoming owner
        int hp = (d.f1 & D_HAS_PAYLOAD) ? 1 : 0;
        BD& bd = bulk[dest][hp];
        bd.keys.push_back(d.k);
        bd.info.push_back((d.f0 << 8) | (d.f1 & 0xff));
        if (hp) {
          stat_replica_payloads_ += 1;
          bd.offs.push_back(d.val_off);
          bd.soffs.push_back(d.sync_off);
        }
      }
      for (auto& [dest, arr] : bulk) {
        Msg& m = msgs[dest];
        for (int hp = 0; hp < 2; ++hp) {
          BD& bd = arr[hp];
          if (bd.keys.empty()) continue;
          int64_t nk = (int64_t)bd.keys.size();
          m.meta.insert(m.meta.end(), {M_DELTA_BULK, nk, rank_, hp, 0});
          m.meta.insert(m.meta.end(), bd.keys.begin(), bd.keys.end());
          m.meta.insert(m.meta.end(), bd.info.begin(), bd.info.end());
          for (int64_t i = 0; i < (int64_t)bd.offs.size(); ++i) {
            m.extracts.add(bd.offs[i], m.payload_floats, l);
            m.extract_sync.push_back(bd.soffs[i]);
            m.payload_floats += l;
          }
        }
      }
    } else {
      for (auto& d : deltas) {
        int dest = directions(d.k);
        if (dest == rank_) continue;  // raced with becoming owner
        Msg& m = add_rec(dest, M_DELTA, d.k, d.f0, d.f1, rank_);
        if (d.f1 & D_HAS_PAYLOAD) {
          stat_replica_payloads_ += 1;
          m.extracts.add(d.val_off, m.payload_floats, d.len);
          m.extract_sync.push_back(d.sync_off);
          m.payload_floats += d.len;
        }
      }
    }
    for (auto& r : ops_out) {
      if (r.code == M_PULL_REQ_BULK || r.code == M_PUSH_REQ_BULK) {
        if (r.dest == rank_) {  // stale direction: handle locally
          if (r.code == M_PULL_REQ_BULK)
            handle_pull_bulk(ch, C, (int)r.f0, r.f1, (int)r.f2, r.keys.data(), r.aux.data(),
                             (int64_t)r.keys.size());
          else
            handle_push_bulk(ch, C, (int)r.f0, r.f1, r.f2, r.keys.data(),
                             (int64_t)r.keys.size(), r.payload);
          continue;
        }
        Msg& m = add_rec(r.dest, r.code, (int64_t)r.keys.size(), r.f0, r.f1, r.f2);
        m.meta.insert(m.meta.end(), r.keys.begin(), r.keys.end());
        if (r.code == M_PULL_REQ_BULK)
          m.meta.insert(m.meta.end(), r.aux.begin(), r.aux.end());
        if (r.payload.defined()) {
          m.copies.push_back({r.payload, m.payload_floats});
          m.payload_floats += r.payload.numel();
        }
        continue;
      }
      int dest = directions(r.key);  // re-resolve at send time
      if (dest == rank_) {
        apply_local_record(ch, C, r);
        continue;
      }
      Msg& m = add_rec(dest, r.code, r.key, r.f0, r.f1, r.f2);
      if (r.payload.defined()) {
        m.copies.push_back({r.payload, m.payload_floats});
        m.payload_floats += r.payload.numel();
      }
    }

    // 5. materialize tensors; extract kernels write into the payloads
    std::vector<std::tuple<int, torch::Tensor, torch::Tensor>> out;
    for (auto& [dest, m] : msgs) {
      auto meta = torch::from_blob(m.meta.data(), {(int64_t)m.meta.size()},
                                   torch::TensorOptions().dtype(torch::kInt64))
                      .clone();
      auto payload = torch::empty({m.payload_floats},
                                  torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      run_extract(m.extracts, m.extract_sync, payload);
      run_gather(m.gathers, payload);
      for (auto& [t, poff] : m.copies) payload.narrow(0, poff, t.numel()).copy_(t, true);
      out.push_back({dest, meta, payload});
      stat_bytes_sent_ += meta.numel() * 8 + m.payload_floats * 4;
    }

    // 6. free dropped slots (slab reuse is stream-ordered; metadata was
    // already cleared under stripe locks + quiesce)
    for (auto& [off, len] : frees) slab_.free_(off, len);

    if (cpp_timing_) t_sy_build_ += tick() - tc0;
    return out;
  }

  // A queued record whose destination resolves to ourselves (the key came
  // home, or we are/became the owner). Applies it as sync_process would.
  void apply_local_record(int ch, ChannelState& C, const OutRec& r) {
    switch (r.code) {
      case M_PUSH_REQ:
      case M_SET_REQ: {
        bool owner = false;
        HostBatch b;
        {
          std::lock_guard<std::mutex> lk(stripe(r.key));
          int64_t m = meta_[r.key].load();
          if (m & F_OWNER) {
            owner = true;
            b.add(mloc(m), 0, len_of(r.key));
            if (m & F_HASREP) version_[r.key]++;
          }
        }
        if (owner) {
          run_scatter(b, r.payload, r.code == M_SET_REQ);
          if ((int)r.f0 == rank_) {
            complete_ticket(r.f1, 1);
          } else {
            std::lock_guard<std::mutex> g(C.mu);
            C.responses.push_back(RespRec{(int)r.f0, M_PUSH_ACK, r.key, r.f1, 1, 0, -1, 0, false, {}});
          }
        } else {
          requeue_bounded(ch, r, /*hops_field=*/2);
        }
        break;
      }
      case M_PULL_REQ: {
        bool owner = false;
        int64_t voff = -1;
        int32_t l = len_of(r.key);
        {
          std::lock_guard<std::mutex> lk(stripe(r.key));
          int64_t m = meta_[r.key].load();
          if (m & F_OWNER) {
            owner = true;
            voff = mloc(m);
          }
        }
        if (owner) {
          if ((int)r.f0 == rank_) {
            torch::Tensor tmp =
                torch::empty({l}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
            HostBatch b;
            b.add(voff, 0, l);
            run_gather(b, tmp);
            deliver_pull(r.f1, r.f2 & 0xffffffff, tmp);
          } else {
            std::lock_guard<std::mutex> g(C.mu);
            C.responses.push_back(
                RespRec{(int)r.f0, M_PULL_RESP, r.key, r.f1, r.f2 & 0xffffffff, 0, voff, l, false, {}});
          }
        } else {
          requeue_bounded(ch, r, /*hops_field=*/2);
        }
        break;
      }
      case M_DELTA: {
        bool owner;
        HostBatch b;
        {
          std::lock_guard<std::mutex> lk(stripe(r.key));
          int64_t m = meta_[r.key].load();
          owner = m & F_OWNER;
          if (owner && r.payload.defined()) {
            b.add(mloc(m), 0, len_of(r.key));
            if (m & F_HASREP) version_[r.key]++;
          }
        }
        if (owner) {
          run_scatter(b, r.payload, false);
          handle_owner_delta(ch, C, r.key, (int)r.f2, r.f0, r.f1 & 0xffff,
                             (int)(r.f1 >> 32));
        } else {
          requeue_bounded(ch, r, /*hops_field=*/1);
        }
        break;
      }
      default:
        break;
    }
  }

  // re-enqueue a record whose believed destination was wrong; bounded hops.
  // At the hop cap (64 — a directory pathology, not normal churn):
  //  - replica DELTAS are never dropped (that would silently lose pushed
  //    updates; reference never drops — addressbook routing converges at
  //    the manager). They keep retrying toward the manager, counted in
  //    stats as overhops.
  //  - ticketed Push/Set/Pull requests fail LOUDLY: the origin's ticket is
  //    marked failed (wait(ts) throws) instead of acking a dropped op.
  void requeue_bounded(int ch, const OutRec& r, int hops_field) {
    OutRec nr = r;
    int64_t* hf = hops_field == 1 ? &nr.f1 : &nr.f2;
    int hops = (int)(*hf >> 32);
    if (hops >= max_hops_) {
      if (r.code == M_DELTA) {
        stat_delta_overhops_ += 1;  // keep chasing; hops stays at the cap
        nr.dest = manager_of(nr.key);
        enqueue_out(ch, std::move(nr));
        return;
      }
      stat_dropped_records_ += 1;
      static const char* why = "remote op dropped at hop limit (directory pathology)";
      if ((int)r.f0 == rank_) {
        fail_ticket(r.f1, 1, why);
      } else {
        std::lock_guard<std::mutex> g(channels_[ch].mu);
        channels_[ch].responses.push_back(
            RespRec{(int)r.f0, M_NACK, r.key, r.f1, 1, 0, -1, 0, false, {}});
      }
      return;
    }
    *hf = (*hf & 0xffffffff) | ((int64_t)(hops + 1) << 32);
    nr.dest = manager_of(nr.key);  // the manager always converges
    enqueue_out(ch, std::move(nr));
  }

  // bulk pull request: serve owned keys from the slab (response gathered
  // at respond time, after quiesce), forward the rest per destination.
  void handle_pull_bulk(int ch, ChannelState& C, int origin, int64_t req_id, int hops,
                        const Key* keys, const int64_t* oidx, int64_t nk) {
    const int32_t l = uniform_len_;
    TORCH_CHECK(l >= 0, "bulk records need a uniform-length store");
    RespRec resp;
    resp.dest = origin;
    resp.code = M_PULL_RESP_BULK;
    resp.f0 = req_id;
    std::map<int, std::pair<std::vector<Key>, std::vector<int64_t>>> fwd;
    for (int64_t i = 0; i < nk; ++i) {
      Key k = keys[i];
      int64_t m = meta_[k].load(std::memory_order_acquire);
      if (m & F_OWNER) {
        resp.keys.push_back(k);
        resp.aux.push_back(oidx[i]);
        resp.slab_offs.push_back(mloc(m));
      } else {
        auto& g = fwd[directions(k)];
        g.first.push_back(k);
        g.second.push_back(oidx[i]);
      }
    }
    if (!resp.keys.empty()) {
      hop_hist_[hops > 8 ? 8 : hops] += (int64_t)resp.keys.size();
      stat_remote_pulls_served_ += (int64_t)resp.keys.size();
      resp.key = (int64_t)resp.keys.size();
      std::lock_guard<std::mutex> g(C.mu);
      C.responses.push_back(std::move(resp));
    }
    for (auto& [d, g] : fwd) {
      if (hops >= max_hops_) {
        stat_dropped_records_ += (int64_t)g.first.size();
        if (origin == rank_) {
          fail_ticket(req_id, (int)g.first.size(), "remote pull dropped at hop limit");
        } else {
          std::lock_guard<std::mutex> gm(C.mu);
          C.responses.push_back(RespRec{origin, M_NACK, 0, req_id,
                                        (int64_t)g.first.size(), 0, -1, 0, false, {}});
        }
        continue;
      }
      stat_forwards_ += (int64_t)g.first.size();
      OutRec rec{d == rank_ ? manager_of(g.first[0]) : d, M_PULL_REQ_BULK, 0, origin, req_id,
                 hops + 1, {}};
      rec.keys = std::move(g.first);
      rec.aux = std::move(g.second);
      enqueue_out(ch, std::move(rec));
    }
  }

  // bulk push request: batched merge into owned rows, ack count, forward
  // the rest (with their payload rows subset).
  void handle_push_bulk(int ch, ChannelState& C, int origin, int64_t req_id, int64_t f2,
                        const Key* keys, int64_t nk, torch::Tensor rows_flat) {
    const int32_t l = uniform_len_;
    TORCH_CHECK(l >= 0, "bulk records need a uniform-length store");
    bool set_mode = (f2 >> 32) & 1;
    int hops = (int)(f2 & 0xffffffff);
    HostBatch apply;
    std::map<int, std::pair<std::vector<Key>, std::vector<int64_t>>> fwd;  // keys, row idx
    int64_t applied = 0;
    for (int64_t i = 0; i < nk; ++i) {
      Key k = keys[i];
      int64_t m = meta_[k].load(std::memory_order_acquire);
      if (m & F_OWNER) {
        apply.add(mloc(m), i * (int64_t)l, l);
        if (m & F_HASREP) version_[k].fetch_add(1, std::memory_order_relaxed);
        applied++;
      } else {
        auto& g = fwd[directions(k)];
        g.first.push_back(k);
        g.second.push_back(i);
      }
    }
    run_scatter(apply, rows_flat, set_mode);
    if (applied > 0) {
      hop_hist_[hops > 8 ? 8 : hops] += applied;
      stat_remote_pushes_served_ += applied;
      if (origin == rank_) {
        complete_ticket(req_id, (int)applied);
      } else {
        std::lock_guard<std::mutex> g(C.mu);
        C.responses.push_back(RespRec{origin, M_PUSH_ACK, 0, req_id, applied, 0, -1, 0, false, {}});
      }
    }
    auto rows = rows_flat.view({nk, (int64_t)l});
    for (auto& [d, g] : fwd) {
      if (hops >= max_hops_) {
        stat_dropped_records_ += (int64_t)g.first.size();
        if (origin == rank_) {
          fail_ticket(req_id, (int)g.first.size(), "remote push dropped at hop limit");
        } else {
          std::lock_guard<std::mutex> gm(C.mu);
          C.responses.push_back(RespRec{origin, M_NACK, 0, req_id,
                                        (int64_t)g.first.size(), 0, -1, 0, false, {}});
        }
        continue;
      }
      stat_forwards_ += (int64_t)g.first.size();
      auto idx = torch::from_blob(g.second.data(), {(int64_t)g.second.size()},
                                  torch::TensorOptions().dtype(torch::kInt64))
                     .clone();
      if (dev_.is_cuda()) idx = idx.to(dev_, true);
      OutRec rec{d == rank_ ? manager_of(g.first[0]) : d, M_PUSH_REQ_BULK, 0, origin, req_id,
                 (set_mode ? (1LL << 32) : 0) | (hops + 1), rows.index_select(0, idx).reshape({-1})};
      rec.keys = std::move(g.first);
      enqueue_out(ch, std::move(rec));
    }
  }

  // ------------------------------------------------ sync round: phase A in

  void sync_process(int ch, int src, torch::Tensor meta, torch::Tensor payload) {
    ChannelState& C = channels_[ch];
    int64_t tp0 = cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0;
    meta = meta.contiguous();
    int64_t n_words = meta.numel();
    const int64_t* mp = meta.data_ptr<int64_t>();
    stat_bytes_recv_ += meta.numel() * 8 + payload.numel() * 4;

    HostBatch merges, assigns;
    int64_t poff = 0;
    int64_t pos = 0;

    while (pos + REC_I64 <= n_words) {
      int64_t code = mp[pos + 0];
      Key k = mp[pos + 1];
      int64_t f0 = mp[pos + 2];
      int64_t f1 = mp[pos + 3];
      int64_t f2 = mp[pos + 4];
      pos += REC_I64;
      if (code == M_PULL_REQ_BULK) {
        int64_t nk = k;
        handle_pull_bulk(ch, C, (int)f0, f1, (int)f2, mp + pos, mp + pos + nk, nk);
        pos += 2 * nk;
        continue;
      }
      if (code == M_PUSH_REQ_BULK) {
        int64_t nk = k;
        int64_t rows_floats = nk * (int64_t)uniform_len_;
        handle_push_bulk(ch, C, (int)f0, f1, f2, mp + pos, nk,
                         payload.narrow(0, poff, rows_floats));
        pos += nk;
        poff += rows_floats;
        continue;
      }
      if (code == M_DELTA_BULK) {
        int64_t nk = k;
        bool hp = f1 != 0;
        handle_delta_bulk(ch, C, (int)f0, hp, mp + pos, mp + pos + nk, nk, payload, poff,
                          merges);
        pos += 2 * nk;
        if (hp) poff += nk * (int64_t)uniform_len_;
        continue;
      }
      int32_t l = len_of(k);

      switch (code) {
        case M_DELTA: {
          int origin = (int)f2;
          bool has_payload = f1 & D_HAS_PAYLOAD;
          bool owner;
          {
            std::lock_guard<std::mutex> lk(stripe(k));
            int64_t m = meta_[k].load();
            owner = m & F_OWNER;
            if (owner && has_payload) {
              merges.add(mloc(m), poff, l);
              if (m & F_HASREP) version_[k]++;
            }
          }
          if (!owner) {
            torch::Tensor pay;
            if (has_payload) pay = payload.narrow(0, poff, l).clone();
            requeue_bounded(ch, OutRec{0, M_DELTA, k, f0, f1, f2, pay}, /*hops_field=*/1);
            stat_forwards_ += 1;
          } else {
            handle_owner_delta(ch, C, k, origin, f0, f1 & 0xffff, (int)(f1 >> 32));
          }
          if (has_payload) poff += l;
          break;
        }
        case M_PUSH_REQ:
        case M_SET_REQ: {
          bool owner;
          {
            std::lock_guard<std::mutex> lk(stripe(k));
            int64_t m = meta_[k].load();
            owner = m & F_OWNER;
            if (owner) {
              (code == M_SET_REQ ? assigns : merges).add(mloc(m), poff, l);
              if (m & F_HASREP) version_[k]++;
            }
          }
          if (owner) {
            int hops = (int)(f2 >> 32);
            hop_hist_[hops > 8 ? 8 : hops] += 1;
            std::lock_guard<std::mutex> g(C.mu);
            C.responses.push_back(RespRec{(int)f0, M_PUSH_ACK, k, f1, 1, 0, -1, 0, false, {}});
            stat_remote_pushes_served_ += 1;
          } else {
            torch::Tensor pay = payload.narrow(0, poff, l).clone();
            requeue_bounded(ch, OutRec{0, (MsgCode)code, k, f0, f1, f2, pay}, /*hops_field=*/2);
            stat_forwards_ += 1;
          }
          poff += l;
          break;
        }
        case M_PULL_REQ: {
          bool owner;
          int64_t voff = -1;
          {
            std::lock_guard<std::mutex> lk(stripe(k));
            int64_t m = meta_[k].load();
            owner = m & F_OWNER;
            if (owner) voff = mloc(m);
          }
          if (owner) {
            int hops = (int)(f2 >> 32);
            hop_hist_[hops > 8 ? 8 : hops] += 1;
            std::lock_guard<std::mutex> g(C.mu);
            C.responses.push_back(
                RespRec{(int)f0, M_PULL_RESP, k, f1, f2 & 0xffffffff, 0, voff, l, false, {}});
            stat_remote_pulls_served_ += 1;
          } else {
            requeue_bounded(ch, OutRec{0, M_PULL_REQ, k, f0, f1, f2, {}}, /*hops_field=*/2);
            stat_forwards_ += 1;
          }
          break;
        }
        case M_RESIDENCE: {
          apply_residence(k, (int)f0, (uint32_t)f1);
          break;
        }
        default:
          TORCH_CHECK(false, "unexpected phase-A record code ", code);
      }
    }
    run_scatter(merges, payload, false);
    run_scatter(assigns, payload, true);
    if (cpp_timing_) {
      t_sy_proc_ += std::chrono::steady_clock::now().time_since_epoch().count() - tp0;
      n_sy_proc_recs_ += n_words / REC_I64;
    }
  }

  // BULK owner-side delta handling (uniform stores): one record covers
  // every replica delta from `origin` for this channel; the decisions
  // match handle_owner_delta key-for-key but run with batched channel-
  // mutex acquisitions and ONE bulk refresh response + per-manager bulk
  // residence records instead of a response object per key.
  void handle_delta_bulk(int ch, ChannelState& C, int origin, bool has_payload,
                         const Key* keys, const int64_t* info, int64_t nk,
                         torch::Tensor payload, int64_t poff, HostBatch& merges) {
    const int32_t l = uniform_len_;
    struct Item {
      Key k;
      int64_t ver;
      uint8_t dflags;
      int cls;  // 0=drop, 1=decide, 2=not-owner (requeue)
      int64_t voff = -1;
      uint64_t other_holders = 0;
      bool relocate = false;
      bool granted = false;  // replica granted: record the holder in pass 3
    };
    std::vector<Item> items;
    items.reserve(nk);
    for (int64_t i = 0; i < nk; ++i) {
      Key k = keys[i];
      Item it{k, info[i] >> 8, (uint8_t)(info[i] & 0xff), 1};
      bool owner;
      {
        std::lock_guard<std::mutex> lk(stripe(k));
        int64_t m = meta_[k].load();
        owner = m & F_OWNER;
        if (owner && has_payload) {
          merges.add(mloc(m), poff + i * (int64_t)l, l);
          if (m & F_HASREP) version_[k]++;
        }
      }
      if (!owner) {
        it.cls = 2;
      } else if (it.dflags & D_DROPPING) {
        it.cls = 0;
      }
      items.push_back(it);
    }
    // requeue non-owned records toward the believed owner (rare)
    for (int64_t i = 0; i < nk; ++i) {
      if (items[i].cls != 2) continue;
      torch::Tensor pay;
      if (has_payload) pay = payload.narrow(0, poff + i * (int64_t)l, l).clone();
      requeue_bounded(ch, OutRec{0, M_DELTA, items[i].k, items[i].ver,
                                 (int64_t)items[i].dflags, origin, pay},
                      /*hops_field=*/1);
      stat_forwards_ += 1;
    }
    // pass 1 (one channel lock): drop-holder removals + holder reads
    std::vector<Key> hasrep_clear;
    {
      std::lock_guard<std::mutex> g(C.mu);
      for (auto& it : items) {
        if (it.cls == 0) {
          auto h = C.holders.find(it.k);
          if (h != C.holders.end()) {
            h->second &= ~(1ULL << origin);
            if (h->second == 0) {
              C.holders.erase(h);
              hasrep_clear.push_back(it.k);
            }
          }
        } else if (it.cls == 1) {
          auto h = C.holders.find(it.k);
          it.other_holders =
              (h == C.holders.end() ? 0 : h->second) & ~(1ULL << origin);
        }
      }
    }
    for (Key k : hasrep_clear) meta_[k].fetch_and(~(int64_t)F_HASREP);

    // pass 2 (no channel lock): decisions + metadata transitions
    RespRec rb;  // bulk refresh to `origin`
    rb.dest = origin;
    rb.code = M_REFRESH_BULK;
    std::map<int, std::pair<std::vector<Key>, std::vector<int64_t>>> resid;  // mgr -> keys, oc
    bool any_decided = false;
    for (auto& it : items) {
      if (it.cls != 1) continue;
      any_decided = true;
      bool local_intent = intent_cnt_[it.k].load(std::memory_order_relaxed) > 0;
      bool relocate =
          techniques_ != TECH_REPLICATION_ONLY && !local_intent && it.other_holders == 0;
      if (relocate) {
        auto rit = C.reloc_round.find(it.k);
        if (rit != C.reloc_round.end() && C.rounds.load() - rit->second < 3) relocate = false;
      }
      if (techniques_ == TECH_RELOCATION_ONLY && !relocate) continue;
      if (relocate) {
        layout_identity_.store(false, std::memory_order_release);
        int64_t new_ver;
        {
          std::lock_guard<std::mutex> lk(stripe(it.k));
          int64_t m = meta_[it.k].load();
          if (!(m & F_OWNER)) continue;  // raced
          it.voff = mloc(m);
          meta_[it.k].store(0);
          new_ver = version_[it.k].fetch_add(1) + 1;
        }
        if (use_loc_cache_) loc_cache_[it.k] = origin;
        uint32_t ctr = ++C.reloc_ctr[it.k];
        C.reloc_ctr.erase(it.k);
        C.reloc_round.erase(it.k);
        it.relocate = true;
        rb.keys.push_back(it.k);
        rb.aux.push_back(new_ver);
        rb.aux2.push_back(((int64_t)ctr << 8) | R_RELOCATE);
        rb.slab_offs.push_back(it.voff);
        int mgr = manager_of(it.k);
        if (mgr == rank_) {
          apply_residence(it.k, origin, ctr);
        } else {
          auto& rv = resid[mgr];
          rv.first.push_back(it.k);
          rv.second.push_back(((int64_t)ctr << 8) | origin);
        }
        stat_relocations_ += 1;
        trace_event(it.k, "RELOC_OUT");
      } else {
        bool is_new = it.dflags & D_NEW;
        it.granted = true;
        layout_identity_.store(false, std::memory_order_release);
        meta_[it.k].fetch_or((int64_t)F_HASREP);
        int64_t cur_ver, voff;
        {
          std::lock_guard<std::mutex> lk(stripe(it.k));
          int64_t m = meta_[it.k].load();
          cur_ver = version_[it.k];
          voff = mloc(m);
        }
        if (is_new || cur_ver != it.ver) {
          rb.keys.push_back(it.k);
          rb.aux.push_back(cur_ver);
          rb.aux2.push_back(0);
          rb.slab_offs.push_back(voff);
          if (is_new) stat_replications_ += 1;
        }
      }
    }
    // pass 3 (one channel lock): holder grants + queue the responses
    if (any_decided || !rb.keys.empty() || !resid.empty()) {
      std::lock_guard<std::mutex> g(C.mu);
      for (auto& it : items)
        if (it.granted) C.holders[it.k] |= 1ULL << origin;
      if (!rb.keys.empty()) {
        rb.key = (int64_t)rb.keys.size();
        C.responses.push_back(std::move(rb));
      }
      for (auto& [mgr, rv] : resid) {
        RespRec rr;
        rr.dest = mgr;
        rr.code = M_RESIDENCE_BULK;
        rr.key = (int64_t)rv.first.size();
        rr.keys = std::move(rv.first);
        rr.aux = std::move(rv.second);
        C.responses.push_back(std::move(rr));
      }
    }
  }

  // owner-side replicate-vs-relocate decision (reference sync_manager.h:612-689)
  void handle_owner_delta(int ch, ChannelState& C, Key k, int origin_rank, int64_t reported_ver,
                          int64_t dflags, int hops = 0) {
    (void)ch;
    if (dflags & D_DROPPING) {
      bool none_left = false;
      {
        std::lock_guard<std::mutex> g(C.mu);
        auto it = C.holders.find(k);
        if (it != C.holders.end()) {
          it->second &= ~(1ULL << origin_rank);
          if (it->second == 0) {
            C.holders.erase(it);
            none_left = true;
          }
        }
      }
      // last replica gone: owner pushes stop paying the version touch
      if (none_left) meta_[k].fetch_and(~(int64_t)F_HASREP);
      return;
    }
    bool local_intent = intent_cnt_[k].load(std::memory_order_relaxed) > 0;
    uint64_t other_holders;
    {
      std::lock_guard<std::mutex> g(C.mu);
      auto it = C.holders.find(k);
      other_holders = (it == C.holders.end() ? 0 : it->second) & ~(1ULL << origin_rank);
    }
    bool relocate = false;
    if (techniques_ != TECH_REPLICATION_ONLY && !local_intent && other_holders == 0) relocate = true;
    // Relocation cooldown: a key that just relocated in may not relocate
    // out for 3 rounds. Forwarded requests chase a moving key at one hop
    // per round, so without this a key ping-ponging every round makes the
    // chase a livelock; with cooldown >= 2 the request always catches up.
    if (relocate) {
      auto it = C.reloc_round.find(k);
      if (it != C.reloc_round.end() && C.rounds.load() - it->second < 3) relocate = false;
    }
    if (techniques_ == TECH_RELOCATION_ONLY && !relocate) {
      return;  // cannot replicate: requester keeps its stub, ops stay remote
    }

    if (relocate) {
      layout_identity_.store(false, std::memory_order_release);
      int64_t voff;
      int32_t l = len_of(k);
      int64_t new_ver;
      {
        std::lock_guard<std::mutex> lk(stripe(k));
        int64_t m = meta_[k].load();
        if (!(m & F_OWNER)) return;  // raced
        voff = mloc(m);
        meta_[k].store(0);  // absent: new local ops route remotely
        new_ver = version_[k].fetch_add(1) + 1;
      }
      // NO per-key quiesce here (it ran once per relocation — hundreds
      // of inflight-drain spins per round at churn): the slot at voff
      // is only gathered AND freed in sync_respond, which quiesces once
      // before its gathers — any worker op whose pass saw OWNER has its
      // kernel enqueued by then, so the relocation payload includes
      // every racing merge and the slot-reuse stays stream-ordered.
      if (use_loc_cache_) loc_cache_[k] = origin_rank;
      uint32_t ctr = ++C.reloc_ctr[k];  // travels with ownership (sync thread only)
      C.reloc_ctr.erase(k);
      C.reloc_round.erase(k);
      int mgr = manager_of(k);
      {
        std::lock_guard<std::mutex> g(C.mu);
        C.holders.erase(k);
        C.responses.push_back(RespRec{origin_rank, M_REFRESH, k, new_ver,
                                      R_RELOCATE | ((int64_t)ctr << 8), 0, voff, l,
                                      /*free_after=*/true, {}});
        if (mgr != rank_)
          C.responses.push_back(
              RespRec{mgr, M_RESIDENCE, k, origin_rank, (int64_t)ctr, 0, -1, 0, false, {}});
      }
      if (mgr == rank_) apply_residence(k, origin_rank, ctr);
      stat_relocations_ += 1;
      trace_event(k, "RELOC_OUT");
    } else {
      bool is_new = dflags & D_NEW;
      // a FORWARDED delta means the origin's believed location of this
      // key is stale: without a correction every per-replica poll
      // re-chases through the manager each round (pure overhead, and a
      // permanent multi-hop journey for its eventual data). Send the
      // origin a residence hint (applied to its location cache).
      if (hops > 0 && origin_rank != rank_) {
        auto it = C.reloc_ctr.find(k);
        int64_t rc = it == C.reloc_ctr.end() ? 0 : (int64_t)it->second;
        std::lock_guard<std::mutex> g(C.mu);
        C.responses.push_back(RespRec{origin_rank, M_RESIDENCE, k, rank_, rc, 0, -1, 0, false, {}});
      }
      // a replica of one of our keys now exists: versions become
      // observable, so the fast paths (identity layout, and the
      // per-key F_HASREP version-bump skip) end. HASREP must be set
      // BEFORE reading cur_ver: any push after this read then bumps,
      // so the replica can never miss a refresh.
      layout_identity_.store(false, std::memory_order_release);
      meta_[k].fetch_or((int64_t)F_HASREP);
      {
        std::lock_guard<std::mutex> g(C.mu);
        C.holders[k] |= 1ULL << origin_rank;
      }
      int64_t cur_ver, voff;
      int32_t l = len_of(k);
      {
        std::lock_guard<std::mutex> lk(stripe(k));
        int64_t m = meta_[k].load();
        cur_ver = version_[k];
        voff = mloc(m);
      }
      if (is_new || cur_ver != reported_ver) {
        std::lock_guard<std::mutex> g(C.mu);
        C.responses.push_back(RespRec{origin_rank, M_REFRESH, k, cur_ver, 0, 0, voff, l, false, {}});
        if (is_new) stat_replications_ += 1;
      }
    }
  }

  void apply_residence(Key k, int new_owner, uint32_t ctr) {
    if (manager_of(k) != rank_) {
      // not the manager: the record is a location-cache hint (owner-
      // side correction of our stale believed location)
      if (use_loc_cache_ && new_owner != rank_) loc_cache_[k] = new_owner;
      return;
    }
    int64_t idx = k / world_;
    std::lock_guard<std::mutex> lk(stripe(k));
    if ((int32_t)(ctr - mgr_reloc_ctr_[idx]) > 0) {  // monotonic, wrap-safe
      mgr_reloc_ctr_[idx] = ctr;
      owner_of_[idx] = new_owner;
    }
  }

  // ------------------------------------------------ sync round: phase B

  std::vector<std::tuple<int, torch::Tensor, torch::Tensor>> sync_respond(int ch) {
    ChannelState& C = channels_[ch];
    quiesce();  // refresh/pull-resp gathers must follow in-flight push kernels
    std::vector<RespRec> resp;
    {
      std::lock_guard<std::mutex> g(C.mu);
      resp.swap(C.responses);
    }
    struct Msg {
      std::vector<int64_t> meta;
      HostBatch gathers;
      std::vector<std::pair<torch::Tensor, int64_t>> copies;
      int64_t payload_floats = 0;
    };
    std::unordered_map<int, Msg> msgs;
    std::vector<std::pair<int64_t, int32_t>> frees;

    for (auto& r : resp) {
      Msg& m = msgs[r.dest];
      m.meta.insert(m.meta.end(), {r.code, r.key, r.f0, r.f1, r.f2});
      if (r.code == M_PULL_RESP_BULK) {
        m.meta.insert(m.meta.end(), r.keys.begin(), r.keys.end());
        m.meta.insert(m.meta.end(), r.aux.begin(), r.aux.end());
        const int32_t l = uniform_len_;
        for (size_t i = 0; i < r.slab_offs.size(); ++i) {
          m.gathers.add(r.slab_offs[i], m.payload_floats, l);
          m.payload_floats += l;
        }
        continue;
      }
      if (r.code == M_REFRESH_BULK) {
        m.meta.insert(m.meta.end(), r.keys.begin(), r.keys.end());
        m.meta.insert(m.meta.end(), r.aux.begin(), r.aux.end());
        m.meta.insert(m.meta.end(), r.aux2.begin(), r.aux2.end());
        const int32_t l = uniform_len_;
        for (size_t i = 0; i < r.slab_offs.size(); ++i) {
          m.gathers.add(r.slab_offs[i], m.payload_floats, l);
          m.payload_floats += l;
          if (r.aux2[i] & R_RELOCATE) frees.push_back({r.slab_offs[i], l});
        }
        continue;
      }
      if (r.code == M_RESIDENCE_BULK) {
        m.meta.insert(m.meta.end(), r.keys.begin(), r.keys.end());
        m.meta.insert(m.meta.end(), r.aux.begin(), r.aux.end());
        continue;
      }
      if (r.slab_off >= 0) {
        m.gathers.add(r.slab_off, m.payload_floats, r.len);
        m.payload_floats += r.len;
        if (r.free_after) frees.push_back({r.slab_off, r.len});
      } else if (r.payload.defined()) {
        m.copies.push_back({r.payload, m.payload_floats});
        m.payload_floats += r.payload.numel();
      }
    }
    std::vector<std::tuple<int, torch::Tensor, torch::Tensor>> out;
    for (auto& [dest, m] : msgs) {
      auto meta = torch::from_blob(m.meta.data(), {(int64_t)m.meta.size()},
                                   torch::TensorOptions().dtype(torch::kInt64))
                      .clone();
      auto payload = torch::empty({m.payload_floats},
                                  torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      run_gather(m.gathers, payload);
      for (auto& [t, poff] : m.copies) payload.narrow(0, poff, t.numel()).copy_(t, true);
      out.push_back({dest, meta, payload});
      stat_bytes_sent_ += meta.numel() * 8 + m.payload_floats * 4;
    }
    for (auto& [off, len] : frees) slab_.free_(off, len);
    return out;
  }

  void sync_apply(int ch, int src, torch::Tensor meta, torch::Tensor payload) {
    ChannelState& C = channels_[ch];
    int64_t tp0 = cpp_timing_ ? std::chrono::steady_clock::now().time_since_epoch().count() : 0;
    meta = meta.contiguous();
    int64_t n_words = meta.numel();
    const int64_t* mp = meta.data_ptr<int64_t>();
    stat_bytes_recv_ += meta.numel() * 8 + payload.numel() * 4;

    HostBatch refreshes, acquires;
    std::vector<int64_t> refresh_sync;
    std::vector<std::pair<int64_t, int32_t>> local_frees;
    // Flag transitions (STUB clear / owner upgrade) are DEFERRED until the
    // value kernels are enqueued: a worker pull between the flag change
    // and the kernel would otherwise serve an unrefreshed (zero) value.
    // On GPU, a gather launched after the deferred flag change is
    // stream-ordered behind the refresh kernel, so it reads fresh data.
    struct Post {
      Key k;
      int64_t new_ver;
      bool relocate;
      int64_t acquire_off;  // >=0: set loc_ to this (stub was gone)
      uint32_t ctr;
    };
    std::vector<Post> posts;
    std::vector<Key> reloc_in;
    int64_t poff = 0;
    int64_t pos = 0;

    while (pos + REC_I64 <= n_words) {
      int64_t code = mp[pos + 0];
      Key k = mp[pos + 1];
      int64_t f0 = mp[pos + 2];
      int64_t f1 = mp[pos + 3];
      pos += REC_I64;
      if (code == M_PULL_RESP_BULK) {
        int64_t nk = k;
        apply_pull_resp_bulk(f0, mp + pos, mp + pos + nk, nk, payload, poff, src);
        pos += 2 * nk;
        poff += nk * (int64_t)uniform_len_;
        continue;
      }
      if (code == M_REFRESH_BULK) {
        int64_t nk = k;
        const Key* ks = mp + pos;
        const int64_t* vers = mp + pos + nk;
        const int64_t* fc = mp + pos + 2 * nk;
        const int32_t l = uniform_len_;
        for (int64_t i = 0; i < nk; ++i) {
          Key kk = ks[i];
          bool relocate = fc[i] & R_RELOCATE;
          uint32_t ctr = (uint32_t)(fc[i] >> 8);
          bool handled = false;
          {
            std::lock_guard<std::mutex> lk(stripe(kk));
            int64_t m = meta_[kk].load();
            uint8_t f = mflags(m);
            if ((f & F_PRESENT) && !(f & F_OWNER)) {
              refreshes.add(mloc(m), poff, l);
              refresh_sync.push_back(sync_loc_[kk]);
              posts.push_back({kk, vers[i], relocate, -1, ctr});
              handled = true;
            }
          }
          if (!handled && relocate) {
            layout_identity_.store(false, std::memory_order_release);
            int64_t voff = slab_.alloc(l);
            acquires.add(voff, poff, l);
            posts.push_back({kk, vers[i], true, voff, ctr});
          }
          poff += l;
        }
        pos += 3 * nk;
        continue;
      }
      if (code == M_RESIDENCE_BULK) {
        int64_t nk = k;
        const Key* ks = mp + pos;
        const int64_t* oc = mp + pos + nk;
        for (int64_t i = 0; i < nk; ++i)
          apply_residence(ks[i], (int)(oc[i] & 0xff), (uint32_t)(oc[i] >> 8));
        pos += 2 * nk;
        continue;
      }
      int32_t l = len_of(k);

      switch (code) {
        case M_REFRESH: {
          bool relocate = f1 & R_RELOCATE;
          uint32_t ctr = (uint32_t)(f1 >> 8);
          bool handled = false;
          {
            std::lock_guard<std::mutex> lk(stripe(k));
            int64_t m = meta_[k].load();
            uint8_t f = mflags(m);
            if ((f & F_PRESENT) && !(f & F_OWNER)) {
              // delta-form apply: val += state - sync; sync = state
              refreshes.add(mloc(m), poff, l);
              refresh_sync.push_back(sync_loc_[k]);
              posts.push_back({k, f0, relocate, -1, ctr});
              handled = true;
            }
          }
          if (!handled && relocate) {
            // we dropped our stub while the relocation was in flight:
            // ownership transfer is unconditional — accept the value.
            layout_identity_.store(false, std::memory_order_release);
            int64_t voff = slab_.alloc(l);
            acquires.add(voff, poff, l);
            posts.push_back({k, f0, true, voff, ctr});
          }
          poff += l;
          break;
        }
        case M_PULL_RESP: {
          torch::Tensor slice = payload.narrow(0, poff, l);
          deliver_pull(f0, f1, slice);
          if (use_loc_cache_) loc_cache_[k] = src;
          poff += l;
          break;
        }
        case M_PUSH_ACK: {
          // NOTE: no location-cache update here — bulk acks aggregate
          // many keys and carry no meaningful key field.
          complete_ticket(f0, (int)f1);
          break;
        }
        case M_NACK: {
          fail_ticket(f0, (int)f1, "remote op dropped at hop limit (directory pathology)");
          break;
        }
        case M_RESIDENCE: {
          apply_residence(k, (int)f0, (uint32_t)f1);
          break;
        }
        default:
          TORCH_CHECK(false, "unexpected phase-B record code ", code);
      }
    }
    run_refresh(refreshes, refresh_sync, payload);
    run_scatter(acquires, payload, /*set=*/true);
    // deferred flag transitions (see comment above)
    for (auto& p : posts) {
      int32_t l = len_of(p.k);
      {
        std::lock_guard<std::mutex> lk(stripe(p.k));
        version_[p.k] = (uint32_t)p.new_ver;
        if (p.acquire_off >= 0) {
          sync_loc_[p.k] = -1;
          meta_[p.k].store(mpack(p.acquire_off, F_PRESENT | F_OWNER));
        } else if (p.relocate) {
          local_frees.push_back({sync_loc_[p.k], l});
          sync_loc_[p.k] = -1;
          // keep the value slot, become owner. A worker-push fetch_or
          // of F_UPDATED racing this store can only lose the UPDATED
          // bit, which is meaningless on an owned key (the merge kernel
          // itself already targeted this same slot).
          meta_[p.k].store(mpack(mloc(meta_[p.k].load()), F_PRESENT | F_OWNER));
        } else {
          meta_[p.k].fetch_and(~(int64_t)F_STUB);  // preserves a concurrent UPDATED
        }
      }
      if (p.relocate) {
        C.reloc_ctr[p.k] = p.ctr;
        C.reloc_round[p.k] = C.rounds.load();
        reloc_in.push_back(p.k);
        if (use_loc_cache_) loc_cache_[p.k] = -1;
        stat_relocated_in_ += 1;
        trace_event(p.k, "RELOC_IN");
      }
    }
    if (!reloc_in.empty()) {  // one lock for all replica-set erases
      std::lock_guard<std::mutex> g(C.mu);
      for (Key k : reloc_in) C.replicas.erase(k);
    }
    for (auto& [off, len] : local_frees) slab_.free_(off, len);
    if (cpp_timing_) {
      t_sy_apply_ += std::chrono::steady_clock::now().time_since_epoch().count() - tp0;
      n_sy_apply_recs_ += n_words / REC_I64;
    }
  }

  // bulk pull response: ONE batched copy kernel from the payload rows into
  // the ticket's output tensor (reusing the gather kernel with the payload
  // as the source arena).
  void apply_pull_resp_bulk(int64_t ts, const Key* keys, const int64_t* oidx, int64_t nk,
                            torch::Tensor payload, int64_t poff, int src) {
    const int32_t l = uniform_len_;
    torch::Tensor out;
    HostBatch hb;
    int delivered = 0;
    {
      std::lock_guard<std::mutex> g(tickets_mu_);
      auto it = tickets_.find(ts);
      if (it == tickets_.end()) return;
      Ticket& t = *it->second;
      out = t.out;
      for (int64_t i = 0; i < nk; ++i) {
        int64_t oi = oidx[i];
        if (t.out_off[oi] < 0) continue;  // duplicate
        hb.add(poff + i * (int64_t)l, t.out_off[oi], l);
        t.out_off[oi] = -1 - t.out_off[oi];
        delivered++;
      }
      // received is NOT bumped here: a multi-channel pull gets one bulk
      // response per channel, applied by different sync threads. If this
      // response's rows counted before its copy ran, a racing response
      // could reach expected, erase the ticket and wake the caller while
      // our rows are still unwritten (observed as zero-pulls under heavy
      // oversubscription).
    }
    if (hb.size()) {
      SlabBases pb{payload.data_ptr<float>(), nullptr};
      if (dev_.is_cuda()) {
        auto d = to_dev(hb);
        ops_gather_gpu(pb, d.b, out.data_ptr<float>(), current_stream(dev_));
      } else {
        auto d = to_dev(hb);
        ops_gather_cpu(pb, d.b, out.data_ptr<float>());
      }
    }
    if (use_loc_cache_) {
      for (int64_t i = 0; i < nk; ++i) loc_cache_[keys[i]] = src;
    }
    if (delivered) {
      std::lock_guard<std::mutex> g(tickets_mu_);
      auto it = tickets_.find(ts);
      if (it == tickets_.end()) return;
      Ticket& t = *it->second;
      t.received += delivered;
      if (t.received >= t.expected) {
        if (t.caller_out.defined()) t.caller_out.view({-1}).copy_(t.out.view({-1}));
        if (!t.fail.empty()) failed_tickets_[ts] = std::move(t.fail);
        tickets_.erase(it);
        tickets_cv_.notify_all();
      }
    }
  }

  void deliver_pull(int64_t ts, int64_t out_index, torch::Tensor data) {
    std::lock_guard<std::mutex> g(tickets_mu_);
    auto it = tickets_.find(ts);
    if (it == tickets_.end()) return;
    Ticket& t = *it->second;
    if (t.out_off[out_index] < 0) return;  // duplicate response: ignore
    t.out.view({-1})
        .narrow(0, t.out_off[out_index], t.out_len[out_index])
        .copy_(data.view({-1}), /*non_blocking=*/true);
    t.out_off[out_index] = -1 - t.out_off[out_index];  // mark answered
    t.received++;
    if (t.received >= t.expected) {
      if (t.caller_out.defined()) t.caller_out.view({-1}).copy_(t.out.view({-1}));
      if (!t.fail.empty()) failed_tickets_[ts] = std::move(t.fail);
      tickets_.erase(it);
      tickets_cv_.notify_all();
    }
  }

  void complete_ticket(int64_t ts, int cnt) {
    std::lock_guard<std::mutex> g(tickets_mu_);
    auto it = tickets_.find(ts);
    if (it == tickets_.end()) return;
    it->second->received += cnt;
    if (it->second->received >= it->second->expected) {
      if (!it->second->fail.empty()) failed_tickets_[ts] = std::move(it->second->fail);
      tickets_.erase(it);
      tickets_cv_.notify_all();
    }
  }

  // hop-limit give-up: count the records toward the ticket (so it
  // completes) but mark it failed — wait(ts) throws instead of
  // pretending the dropped op succeeded (reference never drops:
  // addressbook.h routing always converges at the manager).
  void fail_ticket(int64_t ts, int cnt, const char* why) {
    std::lock_guard<std::mutex> g(tickets_mu_);
    auto it = tickets_.find(ts);
    if (it == tickets_.end()) return;
    Ticket& t = *it->second;
    t.received += cnt;
    if (t.fail.empty()) t.fail = why;
    if (t.received >= t.expected) {
      failed_tickets_[ts] = std::move(t.fail);
      tickets_.erase(it);
      tickets_cv_.notify_all();
    }
  }

  // Failure path (reference heartbeat/dead-node handling is skeletal,
  // van.cc:515-527: detection + surface to the app). When the transport
  // fails (peer died), the Python sync loop calls this: every blocked
  // Wait() returns, and later ops raise instead of hanging.
  void fail(std::string reason) {
    {
      std::lock_guard<std::mutex> g(tickets_mu_);
      failed_reason_ = std::move(reason);
      failed_flag_.store(true, std::memory_order_relaxed);
      tickets_.clear();
    }
    tickets_cv_.notify_all();
    rounds_cv_.notify_all();
  }
  std::string failed_reason() {
    std::lock_guard<std::mutex> g(tickets_mu_);
    return failed_reason_;
  }

  void sync_finish(int ch, bool globally_idle = false) {
    if (globally_idle) {
      // self-handled records can requeue work invisibly to peers; a
      // round with DATA still queued locally is not idle. Payload-less
      // forwarded polls (M_DELTA without D_HAS_PAYLOAD, M_PULL_REQ*)
      // carry no updates and do not block idleness — a stale location
      // cache can keep a poll chasing for a few rounds and strong
      // WaitSync must not hang on that.
      std::lock_guard<std::mutex> g(channels_[ch].mu);
      if (!channels_[ch].responses.empty()) globally_idle = false;
      for (const auto& r : channels_[ch].out_queue) {
        bool data = (r.payload.defined() && r.payload.numel() > 0) ||
                    r.code == M_PUSH_REQ || r.code == M_SET_REQ || r.code == M_PUSH_REQ_BULK;
        if (data) {
          globally_idle = false;
          break;
        }
      }
    }
    {
      std::lock_guard<std::mutex> g(rounds_mu_);
      ChannelState& C = channels_[ch];
      C.rounds++;
      if (globally_idle) {
        if (++C.idle_streak >= 2) C.idle2_events++;
      } else {
        C.idle_streak = 0;
      }
    }
    rounds_cv_.notify_all();
  }

  // watchdog/observability: counts of everything still in flight
  py::dict debug_pending() {
    py::dict d;
    {
      std::lock_guard<std::mutex> g(tickets_mu_);
      d["tickets"] = tickets_.size();
    }
    py::list outq, respq, rounds;
    for (auto& C : channels_) {
      std::lock_guard<std::mutex> g(C.mu);
      outq.append(C.out_queue.size());
      respq.append(C.responses.size());
      rounds.append(C.rounds.load());
    }
    d["out_queues"] = outq;
    d["responses"] = respq;
    d["rounds"] = rounds;
    return d;
  }

  // ------------------------------------------------ fused app operator

  // Fused ComplEx train step: the kernel reads rows straight from the
  // HBM slab and atomically accumulates AdaGrad deltas back — no
  // intermediate pull/push buffers. Requires: single rank (all keys
  // local-owned), uniform lengths, identity layout, GPU store. The
  // classic pull/kernel/push path remains the general case. Counts the
  // same pull+push key-ops in the stats (the store IS read and updated
  // per key).
  torch::Tensor kge_step_fused(torch::Tensor keys_s, torch::Tensor keys_r, torch::Tensor keys_o,
                               torch::Tensor keys_neg, int64_t N, int64_t D, double lr,
                               double eps) {
    TORCH_CHECK(world_ == 1, "kge_step_fused requires a single rank (all keys local)");
    TORCH_CHECK(uniform_len_ == 2 * D, "kge_step_fused: store rows must be [emb|accum] = 2D");
    TORCH_CHECK(layout_identity_.load(), "kge_step_fused requires the identity layout");
    TORCH_CHECK(dev_.is_cuda(), "kge_step_fused is the GPU fast path");
    for (auto* t : {&keys_s, &keys_r, &keys_o, &keys_neg}) check_keys(*t);
    int64_t B = keys_s.numel();
    TORCH_CHECK(keys_r.numel() == B && keys_o.numel() == B && keys_neg.numel() == B * N);
    auto rng_check = [&](const torch::Tensor& t) { check_key_range(t); };
    rng_check(keys_s);
    rng_check(keys_r);
    rng_check(keys_o);
    rng_check(keys_neg);

    auto loss = torch::empty({B}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
    auto ks = keys_s.to(dev_, true);
    auto kr = keys_r.to(dev_, true);
    auto ko = keys_o.to(dev_, true);
    auto kn = keys_neg.to(dev_, true);
    {
      InflightGuard g(this);
      kge_complex_step_fused_gpu(slab_.data, ks.data_ptr<int64_t>(), kr.data_ptr<int64_t>(),
                                 ko.data_ptr<int64_t>(), kn.data_ptr<int64_t>(),
                                 loss.data_ptr<float>(), (int)B, (int)N, (int)D,
                                 Slab::padded(uniform_len_), world_, rank_, (float)lr,
                                 (float)eps, current_stream(dev_));
    }
    int64_t total = 3 * B + B * N;
    stat_pull_keys_ += total;
    stat_pull_local_ += total;
    stat_push_keys_ += total;
    stat_push_local_ += total;
    stat_pulls_ += 1;
    stat_pushes_ += 1;
    return loss;
  }

  // ---- fused slab-direct steps at world>1 / relocated layouts --------
  //
  // The identity-layout fused step above requires world==1. At world>1
  // the multi-GPU hot path would otherwise fall off the fused cliff onto
  // the classic pull/kernel/push path for EVERY sample, even though
  // intent-driven relocation makes >95% of keys local. This general
  // variant runs a host metadata pass (the same per-key cost the classic
  // path pays anyway) that resolves each key's slab offset, compacts the
  // samples whose keys are ALL local into an offsets-mode fused kernel
  // launch (row_at world==0), and returns the indices of the samples
  // with any remote/spilled/stub key — the caller routes those through
  // the classic path. Owned keys get their version bumped and replica
  // keys their UPDATED flag set (same bookkeeping as push()), so the
  // sync protocol ships the fused updates like any other.
  struct FusedResolve {
    std::vector<std::vector<int64_t>> offs;  // per input array, compacted
    std::vector<int64_t> missed;             // sample indices (ascending)
    int64_t b_hit = 0;
  };

  FusedResolve resolve_fused(const std::vector<std::pair<const int64_t*, int>>& arrs,
                             int64_t B) {
    FusedResolve out;
    const int na = (int)arrs.size();
    constexpr int64_t G = 1024;
    int64_t nchunks = (B + G - 1) / G;
    struct Part {
      std::vector<std::vector<int64_t>> offs;
      std::vector<int64_t> missed;
    };
    std::vector<Part> parts(nchunks);
    pass_pool_.run(nchunks, [&](int64_t c0, int64_t c1) {
      for (int64_t c = c0; c < c1; ++c) {
        Part& P = parts[c];
        P.offs.resize(na);
        int64_t e = std::min(B, (c + 1) * G);
        std::vector<int64_t> tmp;
        for (int64_t b = c * G; b < e; ++b) {
          tmp.clear();
          bool ok = true;
          for (int a = 0; a < na && ok; ++a) {
            const int64_t* kp = arrs[a].first;
            int cnt = arrs[a].second;
            for (int i = 0; i < cnt; ++i) {
              Key k = kp[b * cnt + i];
              int64_t m = meta_[k].load(std::memory_order_acquire);
              uint8_t f = mflags(m);
              if (!(f & F_PRESENT) || (f & F_STUB)) { ok = false; break; }
              if (m & MSPILL) { ok = false; break; }  // classic path handles spill
              tmp.push_back(mloc(m));
            }
          }
          if (!ok) {
            P.missed.push_back(b);
            continue;
          }
          // bookkeeping so the sync protocol ships these updates
          size_t t = 0;
          for (int a = 0; a < na; ++a) {
            const int64_t* kp = arrs[a].first;
            int cnt = arrs[a].second;
            for (int i = 0; i < cnt; ++i) {
              Key k = kp[b * cnt + i];
              int64_t m = meta_[k].load(std::memory_order_acquire);
              if (m & F_OWNER) {
                if (m & F_HASREP) version_[k].fetch_add(1, std::memory_order_relaxed);
              } else {
                meta_[k].fetch_or((int64_t)F_UPDATED);
              }
              P.offs[a].push_back(tmp[t++]);
            }
          }
        }
      }
    });
    out.offs.resize(na);
    for (auto& P : parts) {
      for (int a = 0; a < na; ++a)
        out.offs[a].insert(out.offs[a].end(), P.offs[a].begin(), P.offs[a].end());
      out.missed.insert(out.missed.end(), P.missed.begin(), P.missed.end());
    }
    out.b_hit = na ? (int64_t)out.offs[0].size() / arrs[0].second : 0;
    return out;
  }

  static torch::Tensor i64vec_to_tensor(const std::vector<int64_t>& v) {
    auto t = torch::empty({(int64_t)v.size()}, torch::TensorOptions().dtype(torch::kInt64));
    if (!v.empty()) std::memcpy(t.data_ptr<int64_t>(), v.data(), v.size() * sizeof(int64_t));
    return t;
  }

  torch::Tensor offs_to_dev(const std::vector<int64_t>& v) {
    auto t = torch::from_blob((void*)v.data(), {(int64_t)v.size()},
                              torch::TensorOptions().dtype(torch::kInt64))
                 .clone();
    return dev_.is_cuda() ? t.to(dev_, /*non_blocking=*/true) : t;
  }

  std::tuple<torch::Tensor, torch::Tensor> kge_step_fused_general(
      torch::Tensor keys_s, torch::Tensor keys_r, torch::Tensor keys_o, torch::Tensor keys_neg,
      int64_t N, int64_t D, double lr, double eps) {
    check_not_failed();
    TORCH_CHECK(uniform_len_ == 2 * D, "kge_step_fused: store rows must be [emb|accum] = 2D");
    for (auto* t : {&keys_s, &keys_r, &keys_o, &keys_neg}) {
      check_keys(*t);
      check_key_range(*t);
    }
    int64_t B = keys_s.numel();
    TORCH_CHECK(keys_r.numel() == B && keys_o.numel() == B && keys_neg.numel() == B * N);

    torch::Tensor loss, missed_t;
    {
      InflightGuard g(this);
      auto res = resolve_fused({{keys_s.data_ptr<int64_t>(), 1},
                                {keys_r.data_ptr<int64_t>(), 1},
                                {keys_o.data_ptr<int64_t>(), 1},
                                {keys_neg.data_ptr<int64_t>(), (int)N}},
                               B);
      int64_t Bh = res.b_hit;
      loss = torch::empty({Bh}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      if (Bh > 0) {
        if (dev_.is_cuda()) {
          auto os = offs_to_dev(res.offs[0]);
          auto orr = offs_to_dev(res.offs[1]);
          auto oo = offs_to_dev(res.offs[2]);
          auto on = offs_to_dev(res.offs[3]);
          kge_complex_step_fused_gpu(slab_.data, os.data_ptr<int64_t>(), orr.data_ptr<int64_t>(),
                                     oo.data_ptr<int64_t>(), on.data_ptr<int64_t>(),
                                     loss.data_ptr<float>(), (int)Bh, (int)N, (int)D,
                                     Slab::padded(uniform_len_), /*world=offsets-mode*/ 0, rank_,
                                     (float)lr, (float)eps, current_stream(dev_));
        } else {
          std::lock_guard<std::mutex> vg(cpu_val_mu_);
          kge_complex_step_fused_offs_cpu(slab_.data, res.offs[0].data(), res.offs[1].data(),
                                          res.offs[2].data(), res.offs[3].data(),
                                          loss.data_ptr<float>(), (int)Bh, (int)N, (int)D,
                                          (float)lr, (float)eps);
        }
      }
      missed_t = i64vec_to_tensor(res.missed);
      int64_t total = (3 + N) * Bh;
      stat_pull_keys_ += total;
      stat_pull_local_ += total;
      stat_push_keys_ += total;
      stat_push_local_ += total;
      stat_pulls_ += 1;
      stat_pushes_ += 1;
    }
    return {loss, missed_t};
  }

  std::tuple<torch::Tensor, torch::Tensor> w2v_step_fused_general(
      torch::Tensor keys_ctr, torch::Tensor keys_ctx, torch::Tensor keys_neg, int64_t N,
      int64_t D, double lr, double eps) {
    check_not_failed();
    TORCH_CHECK(uniform_len_ == 2 * D, "w2v_step_fused: store rows must be [emb|accum] = 2D");
    for (auto* t : {&keys_ctr, &keys_ctx, &keys_neg}) {
      check_keys(*t);
      check_key_range(*t);
    }
    int64_t B = keys_ctr.numel();
    TORCH_CHECK(keys_ctx.numel() == B && keys_neg.numel() == B * N);
    torch::Tensor loss, missed_t;
    {
      InflightGuard g(this);
      auto res = resolve_fused({{keys_ctr.data_ptr<int64_t>(), 1},
                                {keys_ctx.data_ptr<int64_t>(), 1},
                                {keys_neg.data_ptr<int64_t>(), (int)N}},
                               B);
      int64_t Bh = res.b_hit;
      loss = torch::empty({Bh}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      if (Bh > 0) {
        if (dev_.is_cuda()) {
          auto oc = offs_to_dev(res.offs[0]);
          auto ox = offs_to_dev(res.offs[1]);
          auto on = offs_to_dev(res.offs[2]);
          w2v_sgns_step_fused_gpu(slab_.data, oc.data_ptr<int64_t>(), ox.data_ptr<int64_t>(),
                                  on.data_ptr<int64_t>(), loss.data_ptr<float>(), (int)Bh,
                                  (int)N, (int)D, Slab::padded(uniform_len_), 0, (float)lr,
                                  (float)eps, current_stream(dev_));
        } else {
          std::lock_guard<std::mutex> vg(cpu_val_mu_);
          w2v_sgns_step_fused_offs_cpu(slab_.data, res.offs[0].data(), res.offs[1].data(),
                                       res.offs[2].data(), loss.data_ptr<float>(), (int)Bh,
                                       (int)N, (int)D, (float)lr, (float)eps);
        }
      }
      missed_t = i64vec_to_tensor(res.missed);
      int64_t total = (2 + N) * Bh;
      stat_pull_keys_ += total;
      stat_pull_local_ += total;
      stat_push_keys_ += total;
      stat_push_local_ += total;
      stat_pulls_ += 1;
      stat_pushes_ += 1;
    }
    return {loss, missed_t};
  }

  std::tuple<torch::Tensor, torch::Tensor> mf_step_fused_general(torch::Tensor keys_w,
                                                                 torch::Tensor keys_h,
                                                                 torch::Tensor x, int64_t R,
                                                                 double lr, double lambda,
                                                                 double eps) {
    check_not_failed();
    TORCH_CHECK(uniform_len_ == 2 * R, "mf_step_fused: store rows must be [emb|accum] = 2R");
    TORCH_CHECK(x.scalar_type() == torch::kFloat32, "ratings must be float32");
    for (auto* t : {&keys_w, &keys_h}) {
      check_keys(*t);
      check_key_range(*t);
    }
    int64_t B = keys_w.numel();
    TORCH_CHECK(keys_h.numel() == B && x.numel() == B);
    torch::Tensor loss, missed_t;
    {
      InflightGuard g(this);
      auto res = resolve_fused(
          {{keys_w.data_ptr<int64_t>(), 1}, {keys_h.data_ptr<int64_t>(), 1}}, B);
      int64_t Bh = res.b_hit;
      loss = torch::empty({Bh}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      if (Bh > 0) {
        // compact the ratings to the hit samples
        auto xc = x.contiguous();
        torch::Tensor xh = torch::empty({Bh}, torch::TensorOptions().dtype(torch::kFloat32));
        {
          const float* xp = xc.data_ptr<float>();
          float* xo = xh.data_ptr<float>();
          size_t mi = 0;
          int64_t w = 0;
          for (int64_t b = 0; b < B; ++b) {
            if (mi < res.missed.size() && res.missed[mi] == b) { mi++; continue; }
            xo[w++] = xp[b];
          }
        }
        if (dev_.is_cuda()) {
          auto ow = offs_to_dev(res.offs[0]);
          auto oh = offs_to_dev(res.offs[1]);
          auto xd = xh.to(dev_, true);
          mf_update_step_fused_gpu(slab_.data, ow.data_ptr<int64_t>(), oh.data_ptr<int64_t>(),
                                   xd.data_ptr<float>(), loss.data_ptr<float>(), (int)Bh,
                                   (int)R, Slab::padded(uniform_len_), 0, (float)lr,
                                   (float)lambda, (float)eps, current_stream(dev_));
        } else {
          std::lock_guard<std::mutex> vg(cpu_val_mu_);
          mf_update_step_fused_offs_cpu(slab_.data, res.offs[0].data(), res.offs[1].data(),
                                        xh.data_ptr<float>(), loss.data_ptr<float>(), (int)Bh,
                                        (int)R, (float)lr, (float)lambda, (float)eps);
        }
      }
      missed_t = i64vec_to_tensor(res.missed);
      int64_t total = 2 * Bh;
      stat_pull_keys_ += total;
      stat_pull_local_ += total;
      stat_push_keys_ += total;
      stat_push_local_ += total;
      stat_pulls_ += 1;
      stat_pushes_ += 1;
    }
    return {loss, missed_t};
  }

  torch::Tensor w2v_step_fused(torch::Tensor keys_ctr, torch::Tensor keys_ctx,
                               torch::Tensor keys_neg, int64_t N, int64_t D, double lr,
                               double eps) {
    TORCH_CHECK(world_ == 1, "w2v_step_fused requires a single rank (all keys local)");
    TORCH_CHECK(uniform_len_ == 2 * D, "w2v_step_fused: store rows must be [emb|accum] = 2D");
    TORCH_CHECK(layout_identity_.load(), "w2v_step_fused requires the identity layout");
    TORCH_CHECK(dev_.is_cuda(), "w2v_step_fused is the GPU fast path");
    for (auto* t : {&keys_ctr, &keys_ctx, &keys_neg}) check_keys(*t);
    int64_t B = keys_ctr.numel();
    TORCH_CHECK(keys_ctx.numel() == B && keys_neg.numel() == B * N);
    auto rng_check = [&](const torch::Tensor& t) { check_key_range(t); };
    rng_check(keys_ctr);
    rng_check(keys_ctx);
    rng_check(keys_neg);
    auto loss = torch::empty({B}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
    auto kc = keys_ctr.to(dev_, true);
    auto kx = keys_ctx.to(dev_, true);
    auto kn = keys_neg.to(dev_, true);
    {
      InflightGuard g(this);
      w2v_sgns_step_fused_gpu(slab_.data, kc.data_ptr<int64_t>(), kx.data_ptr<int64_t>(),
                              kn.data_ptr<int64_t>(), loss.data_ptr<float>(), (int)B, (int)N,
                              (int)D, Slab::padded(uniform_len_), world_, (float)lr, (float)eps,
                              current_stream(dev_));
    }
    int64_t total = 2 * B + B * N;
    stat_pull_keys_ += total;
    stat_pull_local_ += total;
    stat_push_keys_ += total;
    stat_push_local_ += total;
    stat_pulls_ += 1;
    stat_pushes_ += 1;
    return loss;
  }

  torch::Tensor mf_step_fused(torch::Tensor keys_w, torch::Tensor keys_h, torch::Tensor x,
                              int64_t R, double lr, double lambda, double eps) {
    TORCH_CHECK(world_ == 1, "mf_step_fused requires a single rank (all keys local)");
    TORCH_CHECK(uniform_len_ == 2 * R, "mf_step_fused: store rows must be [emb|accum] = 2R");
    TORCH_CHECK(layout_identity_.load(), "mf_step_fused requires the identity layout");
    TORCH_CHECK(dev_.is_cuda(), "mf_step_fused is the GPU fast path");
    for (auto* t : {&keys_w, &keys_h}) check_keys(*t);
    int64_t B = keys_w.numel();
    TORCH_CHECK(keys_h.numel() == B && x.numel() == B);
    TORCH_CHECK(x.scalar_type() == torch::kFloat32, "ratings must be float32");
    auto rng_check = [&](const torch::Tensor& t) { check_key_range(t); };
    rng_check(keys_w);
    rng_check(keys_h);
    auto loss = torch::empty({B}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
    auto kw = keys_w.to(dev_, true);
    auto kh = keys_h.to(dev_, true);
    auto xd = x.contiguous().to(dev_, true);
    {
      InflightGuard g(this);
      mf_update_step_fused_gpu(slab_.data, kw.data_ptr<int64_t>(), kh.data_ptr<int64_t>(),
                               xd.data_ptr<float>(), loss.data_ptr<float>(), (int)B, (int)R,
                               Slab::padded(uniform_len_), world_, (float)lr, (float)lambda,
                               (float)eps, current_stream(dev_));
    }
    int64_t total = 2 * B;
    stat_pull_keys_ += total;
    stat_pull_local_ += total;
    stat_push_keys_ += total;
    stat_push_local_ += total;
    stat_pulls_ += 1;
    stat_pushes_ += 1;
    return loss;
  }

  // --------------------------------------------- spill-tier rebalance

  // Which tier a key's value lives in: 0 = device (HBM), 1 = host-spill,
  // -1 = not locally present.
  int key_tier(int64_t k) {
    TORCH_CHECK((uint64_t)k < (uint64_t)num_keys_, "key out of range: ", k);
    int64_t m = meta_[k].load(std::memory_order_acquire);
    uint8_t f = mflags(m);
    if (!(f & F_PRESENT) || (f & F_STUB)) return -1;
    return (m & MSPILL) ? 1 : 0;
  }

  // Promote the hottest host-spilled rows into HBM ("HBM as a cache
  // over pinned host memory" — the tier the all-host reference cannot
  // have). Heat = per-key access count (halved every 8th call → EWMA);
  // candidates come from the touched-list the metadata pass records, so
  // a call is O(touched + moves), NOT O(num_keys). If the device arena
  // has headroom the row simply moves; otherwise it swaps with a cold
  // HBM row found by random sampling (power-of-choices; a 2x-heat
  // hysteresis stops ping-ponging). The move batch is stop-the-world:
  // migrating_ gates InflightGuard entry, quiesce() drains ops that
  // already read metadata, then the loc_ updates and one batched
  // gather+scatter staging pass are enqueued before workers resume —
  // single-stream order makes the move invisible to them. Currently
  // world==1 only (sync threads are not gated; at world==1 they carry
  // no traffic).
  int64_t rebalance_spill(int64_t max_moves = 4096) {
    TORCH_CHECK(world_ == 1, "rebalance_spill currently requires world==1");
    TORCH_CHECK(uniform_len_ >= 0, "rebalance_spill requires a uniform-length store");
    if (!heat_ || slab_.host_capacity == 0 || max_moves <= 0) return 0;
    // one rebalance at a time (concurrent callers would interleave their
    // stop-the-world windows and double-apply swaps)
    std::lock_guard<std::mutex> rg(rebalance_mu_);
    const int32_t l = uniform_len_;
    rebalance_calls_++;

    std::vector<Key> touched;
    {
      std::lock_guard<std::mutex> g(spill_mu_);
      touched.swap(spill_touched_);
    }
    if (touched.empty()) return 0;
    std::sort(touched.begin(), touched.end());
    touched.erase(std::unique(touched.begin(), touched.end()), touched.end());

    struct Cand { Key k; uint32_t h; };
    std::vector<Cand> hot;
    hot.reserve(touched.size());
    for (Key k : touched) hot.push_back({k, heat_[k].load(std::memory_order_relaxed)});
    auto hotter = [](const Cand& a, const Cand& b) { return a.h > b.h; };
    if ((int64_t)hot.size() > max_moves) {
      std::nth_element(hot.begin(), hot.begin() + max_moves - 1, hot.end(), hotter);
      hot.resize(max_moves);
    }
    std::sort(hot.begin(), hot.end(), hotter);

    // eviction candidates: random sample of device-resident keys
    // (collected lazily below, only if promotion needs swaps)
    std::mt19937_64 rng(0x9e3779b97f4a7c15ULL ^ (uint64_t)rebalance_calls_);
    std::vector<Cand> cold;
    // incremental: each call samples one chunk, so sampling cost tracks
    // the number of swaps actually performed, not max_moves
    auto sample_cold = [&]() {
      constexpr int64_t CHUNK = 8192;
      size_t before = cold.size();
      for (int64_t tries = 0; tries < 8 * CHUNK && cold.size() < before + CHUNK; ++tries) {
        Key k = (Key)(rng() % (uint64_t)num_keys_);
        int64_t m = meta_[k].load(std::memory_order_acquire);
        uint8_t f = mflags(m);
        if (!(f & F_PRESENT) || (f & F_STUB)) continue;
        if (m & MSPILL) continue;
        cold.push_back({k, heat_[k].load(std::memory_order_relaxed)});
      }
      std::sort(cold.begin() + before, cold.end(),
                [](const Cand& a, const Cand& b) { return a.h < b.h; });
    };

    // stop the world
    migrating_.store(1, std::memory_order_release);
    quiesce();
    int64_t moves = 0;
    HostBatch from, to;
    std::vector<std::pair<int64_t, int32_t>> frees;  // freed AFTER kernels are enqueued
    int64_t pos = 0;
    size_t ci = 0;
    for (auto& hc : hot) {
      if (moves >= max_moves) break;
      Key ks = hc.k;
      int64_t ms = meta_[ks].load(std::memory_order_acquire);
      uint8_t fs = mflags(ms);
      if (!(fs & F_PRESENT) || (fs & F_STUB)) continue;
      int64_t off_s = mloc(ms);
      if (!(off_s & SPILL_BIT)) continue;
      // free HBM headroom? plain move, no eviction
      int64_t off_new = slab_.try_alloc_device(l);
      if (off_new >= 0) {
        from.add(off_s, pos, l);
        to.add(off_new, pos, l);
        meta_[ks].store(mpack(off_new, fs), std::memory_order_release);
        frees.push_back({off_s, l});
        pos += l;
        moves++;
        continue;
      }
      if (ci >= cold.size()) {
        size_t before = cold.size();
        sample_cold();
        if (cold.size() == before) break;  // nothing evictable found
      }
      // skip cold entries invalidated since sampling
      while (ci < cold.size()) {
        Key kd = cold[ci].k;
        int64_t md = meta_[kd].load(std::memory_order_acquire);
        uint8_t fd = mflags(md);
        if ((fd & F_PRESENT) && !(fd & F_STUB) && !(md & MSPILL)) break;
        ci++;
      }
      if (ci >= cold.size()) continue;
      // hysteresis: a swap must be clearly profitable or keys ping-pong
      if ((int64_t)hc.h <= 2 * (int64_t)cold[ci].h + 1) break;
      Key kd = cold[ci].k;
      int64_t md = meta_[kd].load(std::memory_order_acquire);
      int64_t off_d = mloc(md);
      from.add(off_s, pos, l);
      from.add(off_d, pos + l, l);
      to.add(off_d, pos, l);
      to.add(off_s, pos + l, l);
      meta_[ks].store(mpack(off_d, fs), std::memory_order_release);
      meta_[kd].store(mpack(off_s, mflags(md)), std::memory_order_release);
      pos += 2 * (int64_t)l;
      ci++;
      moves++;
    }
    if (moves > 0) {
      auto tmp = torch::empty({pos}, torch::TensorOptions().dtype(torch::kFloat32).device(dev_));
      run_gather(from, tmp);
      run_scatter(to, tmp, /*set=*/true);
      for (auto& fr : frees) slab_.free_(fr.first, fr.second);
    }
    migrating_.store(0, std::memory_order_release);
    stat_spill_moves_ += moves;

    // EWMA decay: a cheap full halving every 8th call keeps old heat
    // from pinning stale residents (O(num_keys) but branch-free)
    if ((rebalance_calls_ & 7) == 0) {
      at::parallel_for(0, num_keys_, 1 << 20, [&](int64_t b, int64_t e) {
        for (int64_t k = b; k < e; ++k) {
          uint32_t h = heat_[k].load(std::memory_order_relaxed);
          if (h) heat_[k].store(h >> 1, std::memory_order_relaxed);
        }
      });
    }
    return moves;
  }

  // ------------------------------------------------ sampling support

  // "Local" sampling scheme scan: per candidate, scan upward (wrapping in
  // [lo, hi)) until a locally-pullable key is found (reference
  // sampling.h:366-525). Lock-free metadata reads: approximation is fine.
  std::pair<torch::Tensor, int64_t> scan_local(torch::Tensor candidates, Key lo, Key hi) {
    auto c = candidates.contiguous();
    int64_t n = c.numel();
    auto out = torch::empty({n}, torch::TensorOptions().dtype(torch::kInt64));
    const int64_t* cp = c.data_ptr<int64_t>();
    int64_t* op = out.data_ptr<int64_t>();
    std::atomic<int64_t> checks{0};
    constexpr int64_t SG = 4096;
    int64_t snchunks = (n + SG - 1) / SG;
    pass_pool_.run(snchunks, [&](int64_t c0, int64_t c1) {
      int64_t b = c0 * SG, e = std::min(n, c1 * SG);
      int64_t local_checks = 0;
      for (int64_t i = b; i < e; ++i) {
        Key k = cp[i];
        Key start = k;
        while (true) {
          local_checks++;
          uint8_t f = mflags(meta_[k].load(std::memory_order_relaxed));
          if ((f & F_PRESENT) && !(f & F_STUB)) break;
          k++;
          if (k >= hi) k = lo;
          if (k == start) break;  // nothing local in range: keep candidate
        }
        op[i] = k;
      }
      checks += local_checks;
    });
    stat_sampling_checks_ += checks.load();
    return {out, checks.load()};
  }

  // ------------------------------------------------ info / stats

  int64_t get_len(Key k) { return len_of(k); }
  int64_t uniform_len() const { return uniform_len_; }  // -1 if per-key lengths
  bool layout_identity() const { return layout_identity_.load(std::memory_order_acquire); }
  int64_t num_keys() const { return num_keys_; }
  int rank() const { return rank_; }
  int world() const { return world_; }
  int num_channels() const { return nch_; }
  int owner_hint(Key k) { return directions(k); }

  void enable_locality_stats() {
    locality_stats_ = true;
    key_accesses_ = std::vector<std::atomic<uint32_t>>(num_keys_);
    key_local_ = std::vector<std::atomic<uint32_t>>(num_keys_);
    for (int64_t i = 0; i < num_keys_; ++i) {
      key_accesses_[i].store(0);
      key_local_[i].store(0);
    }
  }

  // test hook: poison the location cache so a request routes to a wrong
  // destination and exercises the forward/NACK machinery deterministically
  void debug_set_loc_cache(int64_t k, int r) {
    TORCH_CHECK(use_loc_cache_ && (uint64_t)k < (uint64_t)num_keys_);
    loc_cache_[k] = r;
  }

  // debug/observability: raw metadata snapshot for one key
  // (flags, slab offset, version, believed owner from the directory)
  std::tuple<int, int64_t, int64_t, int> debug_key_state(int64_t k) {
    TORCH_CHECK((uint64_t)k < (uint64_t)num_keys_, "key out of range");
    int owner = -1;
    if ((Key)(k % world_) == (Key)rank_) owner = owner_of_[k / world_];
    int64_t m = meta_[k].load();
    return {(int)mflags(m), (m & F_PRESENT) ? mloc(m) : -1, (int64_t)version_[k].load(), owner};
  }

  void enable_key_trace(torch::Tensor keys) {
    std::lock_guard<std::mutex> g(trace_mu_);
    if (keys.numel() == 1 && keys.data_ptr<int64_t>()[0] == -1) {
      trace_all_ = true;
      return;
    }
    auto kc = keys.contiguous();
    for (int64_t i = 0; i < kc.numel(); ++i) traced_keys_.insert(kc.data_ptr<int64_t>()[i]);
  }

  void trace_event(Key k, const char* ev) {
    if (!trace_all_ && traced_keys_.empty()) return;
    std::lock_guard<std::mutex> g(trace_mu_);
    if (!trace_all_ && !traced_keys_.count(k)) return;
    double t = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0_).count();
    trace_.push_back({t, k, ev});
  }

  // TSV dumps (reference locality_stats.rank.<r>.tsv / traces.<r>.tsv)
  void dump_locality_stats(std::string path) {
    if (!locality_stats_) return;
    FILE* f = fopen(path.c_str(), "w");
    if (!f) throw std::runtime_error("cannot open " + path);
    fprintf(f, "key\taccesses\tlocal\n");
    for (int64_t k = 0; k < num_keys_; ++k) {
      uint32_t a = key_accesses_[k].load();
      if (a) fprintf(f, "%lld\t%u\t%u\n", (long long)k, a, key_local_[k].load());
    }
    fclose(f);
  }

  void dump_traces(std::string path) {
    std::lock_guard<std::mutex> g(trace_mu_);
    FILE* f = fopen(path.c_str(), "w");
    if (!f) throw std::runtime_error("cannot open " + path);
    fprintf(f, "time_s\tkey\tevent\n");
    for (auto& e : trace_) fprintf(f, "%.6f\t%lld\t%s\n", e.t, (long long)e.k, e.ev);
    fclose(f);
  }

  py::dict stats() {
    py::dict d;
    d["pulls"] = stat_pulls_.load();
    d["pushes"] = stat_pushes_.load();
    d["pull_keys"] = stat_pull_keys_.load();
    d["push_keys"] = stat_push_keys_.load();
    d["pull_local"] = stat_pull_local_.load();
    d["push_local"] = stat_push_local_.load();
    d["pull_replica"] = stat_pull_replica_.load();
    d["push_replica"] = stat_push_replica_.load();
    d["remote_pulls_served"] = stat_remote_pulls_served_.load();
    d["remote_pushes_served"] = stat_remote_pushes_served_.load();
    d["relocations_out"] = stat_relocations_.load();
    d["relocations_in"] = stat_relocated_in_.load();
    d["replications"] = stat_replications_.load();
    d["replica_drops"] = stat_drops_.load();
    d["forwards"] = stat_forwards_.load();
    d["dropped_records"] = stat_dropped_records_.load();
    d["delta_overhops"] = stat_delta_overhops_.load();
    {
      py::list hh;
      for (int i = 0; i < 9; ++i) hh.append(hop_hist_[i].load());
      d["hop_hist"] = hh;  // served remote ops by forward-hop count (8 = 8+)
    }
    d["replica_records"] = stat_replica_records_.load();
    d["replica_payloads"] = stat_replica_payloads_.load();
    d["bytes_sent"] = stat_bytes_sent_.load();
    d["bytes_recv"] = stat_bytes_recv_.load();
    d["sampling_checks"] = stat_sampling_checks_.load();
    d["slab_in_use"] = slab_.in_use.load();
    d["slab_capacity"] = slab_.capacity;
    d["host_spill_in_use"] = slab_.host_in_use.load();
    d["spill_rebalance_moves"] = stat_spill_moves_.load();
    d["sync_threshold"] = sync_threshold_;
    d["host_spill_capacity"] = slab_.host_capacity;
    int64_t rounds = 0;
    for (auto& c : channels_) rounds += c.rounds.load();
    d["sync_rounds"] = rounds;
    d["t_pass_ms"] = t_pass_.load() / 1e6;
    d["t_sy_intents_ms"] = t_sy_intents_.load() / 1e6;
    d["t_sy_replicas_ms"] = t_sy_replicas_.load() / 1e6;
    d["t_sy_build_ms"] = t_sy_build_.load() / 1e6;
    d["t_sy_proc_ms"] = t_sy_proc_.load() / 1e6;
    d["t_sy_apply_ms"] = t_sy_apply_.load() / 1e6;
    d["n_sy_deltas"] = n_sy_deltas_.load();
    d["n_sy_proc_recs"] = n_sy_proc_recs_.load();
    d["n_sy_apply_recs"] = n_sy_apply_recs_.load();
    d["t_todev_ms"] = t_todev_.load() / 1e6;
    d["t_launch_ms"] = t_launch_.load() / 1e6;
    d["t_misc_ms"] = t_misc_.load() / 1e6;
    d["t_calls"] = t_calls_.load();
    return d;
  }

  torch::Tensor debug_flags() {
    auto t = torch::empty({(int64_t)num_keys_}, torch::TensorOptions().dtype(torch::kUInt8));
    uint8_t* p = t.data_ptr<uint8_t>();
    for (int64_t i = 0; i < num_keys_; ++i) p[i] = mflags(meta_[i].load());
    return t;
  }

 private:
  void check_keys(const torch::Tensor& keys) {
    TORCH_CHECK(keys.device().is_cpu() && keys.scalar_type() == torch::kInt64 && keys.is_contiguous(),
                "keys must be a contiguous CPU int64 tensor");
  }

  // vectorized key-range validation: branch-free OR-accumulated bound
  // scan (the per-key TORCH_CHECK loop cost ~0.2-0.5 ms per fused step
  // at 155k keys; this autovectorizes)
  void check_key_range(const torch::Tensor& keys) {
    const int64_t* kp = keys.data_ptr<int64_t>();
    int64_t n = keys.numel();
    uint64_t bad = 0;
    const uint64_t lim = (uint64_t)num_keys_;
    for (int64_t i = 0; i < n; ++i) bad |= ((uint64_t)kp[i] >= lim);
    if (bad) {
      for (int64_t i = 0; i < n; ++i)
        TORCH_CHECK((uint64_t)kp[i] < lim, "key out of range: ", kp[i]);
    }
  }

  // always-on value-size validation (reference bindings.cc:174-186
  // validates every call; a mis-sized vals tensor must be a Python
  // error, never kernel UB)
  void check_val_size(const torch::Tensor& keys, int64_t have, const char* op) {
    int64_t need;
    int64_t n = keys.numel();
    if (uniform_len_ >= 0) {
      need = n * (int64_t)uniform_len_;
    } else {
      need = 0;
      const int64_t* kp = keys.data_ptr<int64_t>();
      for (int64_t i = 0; i < n; ++i) {
        TORCH_CHECK((uint64_t)kp[i] < (uint64_t)num_keys_, "key out of range: ", kp[i]);
        need += lens_[kp[i]];
      }
    }
    TORCH_CHECK(have == need, op, ": value tensor has ", have, " floats but ", n,
                " key(s) need ", need);
  }

  int64_t num_keys_;
  int rank_, world_, nch_, log2ch_ = 0, techniques_;
  bool use_loc_cache_;
  torch::Device dev_;
  int32_t uniform_len_ = -1;
  std::vector<int32_t> lens_;

  Slab slab_;
  // Per-key metadata: ONE packed atomic int64 per key (mpack/mflags/
  // mloc) holding flags + spill bit + slab offset, so the lock-free
  // worker metadata pass pays one cache miss per key and every
  // (flags, loc) transition is atomic — no write-ordering rules needed.
  // WRITERS of structural transitions still serialize on the stripe
  // mutexes; racing bit updates use fetch_or/fetch_and (a full store
  // under the stripe lock may only ever clobber a concurrent
  // F_UPDATED fetch_or, and only on transitions where UPDATED becomes
  // meaningless — replica→owner upgrades). version_ stays separate: it
  // is only touched for keys with a granted replica (F_HASREP).
  std::vector<std::atomic<int64_t>> meta_;
  std::vector<int64_t> sync_loc_;
  std::vector<std::atomic<uint32_t>> version_;
  std::vector<int32_t> loc_cache_;
  std::vector<int32_t> owner_of_;
  std::vector<uint32_t> mgr_reloc_ctr_;
  std::unique_ptr<std::mutex[]> locks_;
  std::mutex cpu_val_mu_;

  std::vector<ChannelState> channels_;
  std::unique_ptr<std::atomic<uint16_t>[]> intent_cnt_;  // active-intent count per key
  std::vector<std::atomic<Clock>> clocks_;
  Clock intent_ahead_ = 1LL << 40;  // default: act on intents immediately

  // layout-identity fast path: true until the first structural change
  // (stub / relocation / drop). While true, every key owned here sits at
  // offset (k / world) * padded(uniform_len): the worker metadata pass
  // becomes pure arithmetic (no locks, no random loc_ reads) — the "-1
  // fast path" at full speed. The sync thread clears the flag BEFORE any
  // structural change and then quiesces, so in-flight fast-path ops
  // still see valid offsets (slot reuse is stream-ordered).
  std::atomic<bool> layout_identity_{true};

  std::atomic<int> inflight_{0};
  // forward-hop cap before a request NACKs (test hook: ADAPM_MAX_HOPS)
  int max_hops_ = getenv("ADAPM_MAX_HOPS") ? atoi(getenv("ADAPM_MAX_HOPS")) : 64;
  PassPool pass_pool_;
  std::atomic<int> migrating_{0};                       // spill-rebalance stop-the-world gate
  double sync_threshold_ = 0.0;                         // --sys.sync.threshold equivalent
  std::unique_ptr<std::atomic<uint32_t>[]> heat_;       // per-key access heat (spill stores)
  std::mutex spill_mu_;
  std::mutex rebalance_mu_;
  std::vector<Key> spill_touched_;                      // spilled keys accessed since last rebalance
  int64_t rebalance_calls_ = 0;
  std::atomic<int64_t> stat_spill_moves_{0};
  std::atomic<int64_t> next_ts_{1};
 public:
  // env ADAPM_CPP_TIMING=1: nanosecond accounting of the worker-op host path
  std::atomic<int64_t> t_pass_{0}, t_todev_{0}, t_launch_{0}, t_calls_{0}, t_misc_{0};
  // sync-path phase accounting (ADAPM_CPP_TIMING=1): where round host
  // time goes at high relocation churn
  std::atomic<int64_t> t_sy_intents_{0}, t_sy_replicas_{0}, t_sy_build_{0}, t_sy_proc_{0},
      t_sy_apply_{0}, n_sy_deltas_{0}, n_sy_proc_recs_{0}, n_sy_apply_recs_{0};
  bool cpp_timing_ = getenv("ADAPM_CPP_TIMING") != nullptr;
 private:
  std::mutex tickets_mu_;
  std::condition_variable tickets_cv_;
  std::unordered_map<int64_t, std::unique_ptr<Ticket>> tickets_;
  std::unordered_map<int64_t, std::string> failed_tickets_;  // completed-but-failed: wait() throws
  std::string failed_reason_;
  std::atomic<bool> failed_flag_{false};
  std::mutex rounds_mu_;
  std::condition_variable rounds_cv_;

  // opt-in observability (reference PS_LOCALITY_STATS / PS_TRACE_KEYS,
  // coloc_kv_server_handle.h:86-118, 960-992)
  bool locality_stats_ = false;
  std::vector<std::atomic<uint32_t>> key_accesses_, key_local_;
  std::mutex trace_mu_;
  std::unordered_set<Key> traced_keys_;
  bool trace_all_ = false;
  struct TraceEv { double t; Key k; const char* ev; };
  std::vector<TraceEv> trace_;
  std::chrono::steady_clock::time_point t0_ = std::chrono::steady_clock::now();

  // response-hop histogram (reference sync_manager.h:482-519 hop stats):
  // bucket = min(hops, 8) recorded when a remote request is SERVED
  std::atomic<int64_t> hop_hist_[9] = {};
  std::atomic<int64_t> stat_replica_records_{0}, stat_replica_payloads_{0};
  std::atomic<int64_t> stat_pulls_{0}, stat_pushes_{0}, stat_pull_keys_{0}, stat_push_keys_{0},
      stat_pull_local_{0}, stat_push_local_{0}, stat_pull_replica_{0}, stat_push_replica_{0},
      stat_remote_pulls_served_{0}, stat_remote_pushes_served_{0}, stat_relocations_{0},
      stat_relocated_in_{0}, stat_replications_{0}, stat_drops_{0}, stat_forwards_{0},
      stat_dropped_records_{0}, stat_delta_overhops_{0}, stat_bytes_sent_{0},
      stat_bytes_recv_{0}, stat_sampling_checks_{0};
};

// ------------------------------------------------------- app kernel wrappers

static float* fp(torch::Tensor& t) { return t.data_ptr<float>(); }
static const float* cfp(const torch::Tensor& t) { return t.data_ptr<float>(); }

static void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_contiguous() && t.scalar_type() == torch::kFloat32, name,
              " must be contiguous float32");
}

// ComplEx train step; returns nothing, writes deltas + loss in place.
void kge_complex_step(torch::Tensor s, torch::Tensor r, torch::Tensor o, torch::Tensor neg,
                      torch::Tensor ds, torch::Tensor dr, torch::Tensor do_, torch::Tensor dneg,
                      torch::Tensor loss, int64_t N, int64_t D, double lr, double eps) {
  for (auto* t : {&s, &r, &o, &neg, &ds, &dr, &do_, &dneg, &loss}) check_f32(*t, "kge tensor");
  int B = (int)loss.numel();
  if (s.is_cuda()) {
    TORCH_CHECK(hip_available(), "kge_complex_step: CUDA tensor but no HIP device");
    kge_complex_step_gpu(cfp(s), cfp(r), cfp(o), cfp(neg), fp(ds), fp(dr), fp(do_), fp(dneg),
                         fp(loss), B, (int)N, (int)D, (float)lr, (float)eps,
                         current_stream(s.device()));
  } else {
    kge_complex_step_cpu(cfp(s), cfp(r), cfp(o), cfp(neg), fp(ds), fp(dr), fp(do_), fp(dneg),
                         fp(loss), B, (int)N, (int)D, (float)lr, (float)eps);
  }
}

void rescal_step(torch::Tensor s, torch::Tensor r, torch::Tensor o, torch::Tensor neg,
                 torch::Tensor ds, torch::Tensor drl, torch::Tensor do_, torch::Tensor dneg,
                 torch::Tensor loss, int64_t N, int64_t D, double lr, double eps) {
  for (auto* t : {&s, &r, &o, &neg, &ds, &drl, &do_, &dneg, &loss}) check_f32(*t, "rescal");
  int B = (int)loss.numel();
  TORCH_CHECK(D <= 196, "rescal_step: dim must be <= 196 (R tile staged in LDS)");
  if (s.is_cuda()) {
    TORCH_CHECK(hip_available(), "rescal_step: CUDA tensor but no HIP device");
    rescal_step_gpu(cfp(s), cfp(r), cfp(o), cfp(neg), fp(ds), fp(drl), fp(do_), fp(dneg),
                    fp(loss), B, (int)N, (int)D, (float)lr, (float)eps,
                    current_stream(s.device()));
  } else {
    rescal_step_cpu(cfp(s), cfp(r), cfp(o), cfp(neg), fp(ds), fp(drl), fp(do_), fp(dneg),
                    fp(loss), B, (int)N, (int)D, (float)lr, (float)eps);
  }
}

void kge_complex_score(torch::Tensor s, torch::Tensor r, torch::Tensor cand,
                       torch::Tensor scores, int64_t D) {
  for (auto* t : {&s, &r, &cand, &scores}) check_f32(*t, "kge tensor");
  int B = (int)scores.size(0), E = (int)scores.size(1);
  if (s.is_cuda()) {
    TORCH_CHECK(hip_available(), "kge_complex_score: CUDA tensor but no HIP device");
    if (D % 4 == 0) {
      // GEMM-shaped B x E scoring -> MFMA-tiled kernel (matrix cores;
      // exact f32, bitwise an fmaf chain — reference evaluates the
      // same psi against every entity, knowledge_graph_embeddings.cc:716-774)
      auto qbuf = torch::empty({(int64_t)B * D},
                               torch::TensorOptions().dtype(torch::kFloat32).device(s.device()));
      kge_complex_score_mfma_gpu(cfp(s), cfp(r), cfp(cand), fp(scores),
                                 qbuf.data_ptr<float>(), B, E, (int)D,
                                 current_stream(s.device()));
    } else {
      kge_complex_score_gpu(cfp(s), cfp(r), cfp(cand), fp(scores), B, E, (int)D,
                            current_stream(s.device()));
    }
  } else {
    kge_complex_score_cpu(cfp(s), cfp(r), cfp(cand), fp(scores), B, E, (int)D);
  }
}

// Grouped RESCAL step (GPU): triples pre-sorted by relation; group g =
// triples [starts[g], starts[g+1]) sharing relation matrix rm[g]. The
// three dim^2 products run as MFMA-tiled grouped GEMMs (U = S R,
// dS_raw = W R^T, dR_raw = S^T W — the per-triple scalar loops of the
// reference, knowledge_graph_embeddings.cc:895-922, recast as GEMMs);
// the per-triple score/object-grad/W phase and the AdaGrad epilogues
// are elementwise kernels. Numerics note: dR gets AdaGrad applied to
// the GROUP-SUMMED gradient (minibatch semantics) where the classic
// per-triple path transforms each triple's outer product separately
// (both from the same pulled accumulator snapshot).
torch::Tensor rescal_step_grouped(torch::Tensor s, torch::Tensor rm, torch::Tensor o,
                                  torch::Tensor neg, torch::Tensor ds, torch::Tensor drl,
                                  torch::Tensor do_, torch::Tensor dneg, torch::Tensor starts,
                                  int64_t N, int64_t D, double lr, double eps) {
  for (auto* t : {&s, &rm, &o, &neg, &ds, &drl, &do_, &dneg})
    check_f32(*t, "rescal_grouped tensor");
  TORCH_CHECK(s.is_cuda(), "rescal_step_grouped is the GPU path (CPU uses rescal_step)");
  TORCH_CHECK(D % 4 == 0, "rescal_step_grouped: D must be a multiple of 4");
  TORCH_CHECK(starts.scalar_type() == torch::kInt32 && starts.device().is_cpu() &&
              starts.is_contiguous());
  int G = (int)starts.numel() - 1;
  int B = (int)(s.numel() / (2 * D));
  TORCH_CHECK(rm.numel() == (int64_t)G * 2 * D * D, "rm must be [G][2*D*D]");
  TORCH_CHECK(starts.data_ptr<int32_t>()[G] == B, "starts[-1] must equal B");
  auto dev = s.device();
  auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(dev);
  auto U = torch::empty({(int64_t)B * D}, opts);
  auto W = torch::empty({(int64_t)B * D}, opts);
  auto dRraw = torch::empty({(int64_t)G * D * D}, opts);
  auto loss = torch::empty({B}, opts);
  auto starts_d = starts.to(dev, /*non_blocking=*/true);
  void* st = current_stream(dev);
  const int32_t* sp = starts.data_ptr<int32_t>();
  int ntiles = (int)((D + 63) / 64);
  int tiles_bn = 0, tiles_dd = 0;
  for (int g = 0; g < G; ++g) {
    int Bg = sp[g + 1] - sp[g];
    tiles_bn += ((Bg + 15) / 16) * ntiles;
    tiles_dd += ((int)(D + 15) / 16) * ntiles;
  }
  // U = S @ R  (per group)
  mfma_grouped_gemm_gpu(cfp(s), W.data_ptr<float>(), cfp(rm), U.data_ptr<float>(),
                        starts_d.data_ptr<int32_t>(), G, (int)D, (int)(2 * D), (int)D,
                        2 * D * D, (int)D, 0, 0, tiles_bn, st);
  // scores / object grads / W
  rescal_mid_gpu(U.data_ptr<float>(), cfp(o), cfp(neg), fp(do_), fp(dneg),
                 W.data_ptr<float>(), loss.data_ptr<float>(), B, (int)N, (int)D, (float)lr,
                 (float)eps, st);
  // dS_raw = W @ R^T into ds's first-D columns, then AdaGrad in place
  mfma_grouped_gemm_gpu(cfp(s), W.data_ptr<float>(), cfp(rm), fp(ds),
                        starts_d.data_ptr<int32_t>(), G, (int)D, (int)(2 * D), (int)D,
                        2 * D * D, (int)(2 * D), 0, 1, tiles_bn, st);
  adagrad_rows_gpu(cfp(ds), cfp(s), fp(ds), B, (int)D, (int)(2 * D), (int)(2 * D), (float)lr,
                   (float)eps, st);
  // dR_raw = S^T @ W per group, then AdaGrad into drl
  mfma_grouped_gemm_gpu(cfp(s), W.data_ptr<float>(), cfp(rm), dRraw.data_ptr<float>(),
                        starts_d.data_ptr<int32_t>(), G, (int)D, (int)(2 * D), (int)D,
                        2 * D * D, (int)D, (int64_t)D * D, 2, tiles_dd, st);
  adagrad_rows_gpu(dRraw.data_ptr<float>(), cfp(rm), fp(drl), G, (int)(D * D),
                   (int)(D * D), (int)(2 * D * D), (float)lr, (float)eps, st);
  return loss;
}

void w2v_sgns_step(torch::Tensor ctr, torch::Tensor ctx, torch::Tensor neg, torch::Tensor dctr,
                   torch::Tensor dctx, torch::Tensor dneg, torch::Tensor loss, int64_t N,
                   int64_t D, double lr, double eps) {
  for (auto* t : {&ctr, &ctx, &neg, &dctr, &dctx, &dneg, &loss}) check_f32(*t, "w2v tensor");
  int B = (int)loss.numel();
  if (ctr.is_cuda()) {
    TORCH_CHECK(hip_available(), "w2v_sgns_step: CUDA tensor but no HIP device");
    w2v_sgns_step_gpu(cfp(ctr), cfp(ctx), cfp(neg), fp(dctr), fp(dctx), fp(dneg), fp(loss), B,
                      (int)N, (int)D, (float)lr, (float)eps, current_stream(ctr.device()));
  } else {
    w2v_sgns_step_cpu(cfp(ctr), cfp(ctx), cfp(neg), fp(dctr), fp(dctx), fp(dneg), fp(loss), B,
                      (int)N, (int)D, (float)lr, (float)eps);
  }
}

void mf_update_step(torch::Tensor w, torch::Tensor h, torch::Tensor x, torch::Tensor dw,
                    torch::Tensor dh, torch::Tensor loss, int64_t R, double lr, double lambda,
                    double eps) {
  for (auto* t : {&w, &h, &x, &dw, &dh, &loss}) check_f32(*t, "mf tensor");
  int B = (int)loss.numel();
  if (w.is_cuda()) {
    TORCH_CHECK(hip_available(), "mf_update_step: CUDA tensor but no HIP device");
    mf_update_step_gpu(cfp(w), cfp(h), cfp(x), fp(dw), fp(dh), fp(loss), B, (int)R, (float)lr,
                       (float)lambda, (float)eps, current_stream(w.device()));
  } else {
    mf_update_step_cpu(cfp(w), cfp(h), cfp(x), fp(dw), fp(dh), fp(loss), B, (int)R, (float)lr,
                       (float)lambda, (float)eps);
  }
}

// MF NZSL+L2 loss reduction (reference apps/mf/loss.h); returns (se, reg)
torch::Tensor mf_loss(torch::Tensor w, torch::Tensor h, torch::Tensor x, int64_t R,
                      double lambda) {
  for (auto* t : {&w, &h, &x}) check_f32(*t, "mf loss tensor");
  int B = (int)x.numel();
  auto out = torch::zeros({2}, torch::TensorOptions().dtype(torch::kFloat32).device(w.device()));
  if (w.is_cuda()) {
    mf_loss_gpu(cfp(w), cfp(h), cfp(x), fp(out), B, (int)R, (float)lambda,
                current_stream(w.device()));
  } else {
    mf_loss_cpu(cfp(w), cfp(h), cfp(x), fp(out), B, (int)R, (float)lambda);
  }
  return out;
}

// alias-table draw: prob/alias on the op device; returns int64 keys there
torch::Tensor alias_draw(torch::Tensor prob, torch::Tensor alias, int64_t seed, int64_t N) {
  TORCH_CHECK(prob.is_contiguous() && prob.scalar_type() == torch::kFloat32);
  TORCH_CHECK(alias.is_contiguous() && alias.scalar_type() == torch::kInt32);
  auto out = torch::empty({N}, torch::TensorOptions().dtype(torch::kInt64).device(prob.device()));
  if (prob.is_cuda()) {
    alias_draw_gpu(prob.data_ptr<float>(), alias.data_ptr<int32_t>(), prob.numel(),
                   (uint64_t)seed, N, out.data_ptr<int64_t>(), current_stream(prob.device()));
  } else {
    alias_draw_cpu(prob.data_ptr<float>(), alias.data_ptr<int32_t>(), prob.numel(),
                   (uint64_t)seed, N, out.data_ptr<int64_t>());
  }
  return out;
}

}  // namespace adapm

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using namespace adapm;
  m.def("hip_available", &hip_available);
  m.def("kge_complex_step", &kge_complex_step, py::call_guard<py::gil_scoped_release>());
  m.def("alias_draw", &alias_draw, py::call_guard<py::gil_scoped_release>());
  m.def("rescal_step", &rescal_step, py::call_guard<py::gil_scoped_release>());
  m.def("rescal_step_grouped", &rescal_step_grouped, py::call_guard<py::gil_scoped_release>());
  m.def("kge_complex_score", &kge_complex_score, py::call_guard<py::gil_scoped_release>());
  m.def("w2v_sgns_step", &w2v_sgns_step, py::call_guard<py::gil_scoped_release>());
  m.def("mf_update_step", &mf_update_step, py::call_guard<py::gil_scoped_release>());
  m.def("mf_loss", &mf_loss, py::call_guard<py::gil_scoped_release>());
  py::class_<Server>(m, "Server")
      .def(py::init<int64_t, torch::Tensor, int, int, int, int, std::string, double, int, bool,
                    int64_t, int64_t, double>(),
           py::arg("num_keys"), py::arg("value_lengths"), py::arg("rank"), py::arg("world"),
           py::arg("num_channels"), py::arg("num_workers"), py::arg("device"),
           py::arg("capacity_factor") = 2.0, py::arg("techniques") = 0,
           py::arg("location_caches") = true, py::arg("device_cap_floats") = 0,
           py::arg("host_spill_floats") = 0, py::arg("sync_threshold") = 0.0)
      .def("pull", &Server::pull, py::call_guard<py::gil_scoped_release>())
      .def("push", &Server::push, py::call_guard<py::gil_scoped_release>())
      .def("pull_if_local", &Server::pull_if_local, py::call_guard<py::gil_scoped_release>())
      .def("is_local", &Server::is_local)
      .def("intent", &Server::intent, py::call_guard<py::gil_scoped_release>())
      .def("advance_clock", &Server::advance_clock)
      .def("current_clock", &Server::current_clock)
      .def("worker_clocks", &Server::worker_clocks)
      .def("wait", &Server::wait, py::call_guard<py::gil_scoped_release>())
      .def("wait_all", &Server::wait_all, py::call_guard<py::gil_scoped_release>())
      .def("is_finished", &Server::is_finished)
      .def("round_counts", &Server::round_counts)
      .def("wait_rounds", &Server::wait_rounds, py::call_guard<py::gil_scoped_release>())
      .def("idle_counts", &Server::idle_counts)
      .def("wait_idle", &Server::wait_idle, py::call_guard<py::gil_scoped_release>())
      .def("set_intent_ahead", &Server::set_intent_ahead)
      .def("sync_collect", &Server::sync_collect, py::call_guard<py::gil_scoped_release>())
      .def("sync_process", &Server::sync_process, py::call_guard<py::gil_scoped_release>())
      .def("sync_respond", &Server::sync_respond, py::call_guard<py::gil_scoped_release>())
      .def("sync_apply", &Server::sync_apply, py::call_guard<py::gil_scoped_release>())
      .def("sync_finish", &Server::sync_finish, py::arg("ch"),
           py::arg("globally_idle") = false, py::call_guard<py::gil_scoped_release>())
      .def("fail", &Server::fail, py::call_guard<py::gil_scoped_release>())
      .def("failed_reason", &Server::failed_reason)
      .def("scan_local", &Server::scan_local, py::call_guard<py::gil_scoped_release>())
      .def("kge_step_fused", &Server::kge_step_fused, py::call_guard<py::gil_scoped_release>())
      .def("w2v_step_fused", &Server::w2v_step_fused, py::call_guard<py::gil_scoped_release>())
      .def("mf_step_fused", &Server::mf_step_fused, py::call_guard<py::gil_scoped_release>())
      .def("kge_step_fused_general", &Server::kge_step_fused_general,
           py::call_guard<py::gil_scoped_release>())
      .def("w2v_step_fused_general", &Server::w2v_step_fused_general,
           py::call_guard<py::gil_scoped_release>())
      .def("mf_step_fused_general", &Server::mf_step_fused_general,
           py::call_guard<py::gil_scoped_release>())
      .def("rebalance_spill", &Server::rebalance_spill, py::arg("max_moves") = 4096,
           py::call_guard<py::gil_scoped_release>())
      .def("key_tier", &Server::key_tier)
      .def("get_len", &Server::get_len)
      .def("uniform_len", &Server::uniform_len)
      .def("layout_identity", &Server::layout_identity)
      .def("num_keys", &Server::num_keys)
      .def("rank", &Server::rank)
      .def("world", &Server::world)
      .def("num_channels", &Server::num_channels)
      .def("owner_hint", &Server::owner_hint)
      .def("enable_locality_stats", &Server::enable_locality_stats)
      .def("enable_key_trace", &Server::enable_key_trace)
      .def("debug_key_state", &Server::debug_key_state)
      .def("debug_set_loc_cache", &Server::debug_set_loc_cache)
      .def("debug_pending", &Server::debug_pending)
      .def("dump_locality_stats", &Server::dump_locality_stats)
      .def("dump_traces", &Server::dump_traces)
      .def("stats", &Server::stats)
      .def("debug_flags", &Server::debug_flags);
}
