// curvine_amd native data plane: HBM block arenas, pinned staging
// pipelines, and CDNA4 kernel wrappers (kernels.hip).
//
// This is the MI355X-era equivalent of the reference's native layer
// (/root/reference/crates/core/curvine-sys: sendfile/splice/fadvise are the
// CPU fallback path there; here the hot path is hipMemcpyAsync through a
// pinned ring + device kernels).  Torch-free on purpose: consumers pass raw
// device pointers (e.g. torch tensor.data_ptr()) for GPU-to-GPU reads.
//
// Arena model: one big hipMalloc per (device, arena) — the HBM tier of the
// worker block store (BdevLayout/BdevOffsetAllocator analog,
// /root/reference/crates/adapters/curvine-storage-local/src/layout/
// bdev_layout.rs:30-111); offset allocation lives in Python
// (curvine_amd/worker/arena_alloc.py), this layer moves bytes.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <sys/mman.h>

#include <atomic>
#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include <hip/hip_runtime.h>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess)                                                      \
      throw std::runtime_error(std::string(#expr) + ": " +                     \
                               hipGetErrorString(_e));                         \
  } while (0)

// ---------------------------------------------------------------------------
// CRC32C (Castagnoli): host tables, SW slice-by-8, GF(2) combine
// ---------------------------------------------------------------------------

static uint32_t crc_tab[8][256];
static uint32_t comb_mat[24][32];  // shift by (CRC_SUB << k) bytes
static const uint32_t CRC_POLY = 0x82F63B78u;  // reflected Castagnoli
static const int CRC_SUB = 4096;
static const int WG = 256;

static void gf2_square(uint32_t dst[32], const uint32_t src[32]) {
  for (int i = 0; i < 32; ++i) {
    uint32_t v = src[i], out = 0;
    for (int b = 0; b < 32; ++b)
      if ((v >> b) & 1) out ^= src[b];
    dst[i] = out;
  }
}

static uint32_t gf2_apply(const uint32_t mat[32], uint32_t crc) {
  uint32_t out = 0;
  for (int i = 0; i < 32; ++i)
    if ((crc >> i) & 1) out ^= mat[i];
  return out;
}

static void crc_init_tables() {
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = i;
    for (int k = 0; k < 8; ++k) c = (c >> 1) ^ ((c & 1) ? CRC_POLY : 0);
    crc_tab[0][i] = c;
  }
  for (int t = 1; t < 8; ++t)
    for (uint32_t i = 0; i < 256; ++i)
      crc_tab[t][i] = (crc_tab[t - 1][i] >> 8) ^ crc_tab[0][crc_tab[t - 1][i] & 0xFF];

  // M1 = append-one-zero-bit operator; comb_mat[0] = M1^(8*CRC_SUB)
  uint32_t m[32], tmp[32];
  m[0] = CRC_POLY;
  for (int i = 1; i < 32; ++i) m[i] = 1u << (i - 1);
  // raise to 8*CRC_SUB = 2^15 -> square 15 times
  int shift_bits_log2 = 0;
  {
    uint64_t bits = 8ull * CRC_SUB;  // 32768 = 2^15
    while ((1ull << shift_bits_log2) < bits) shift_bits_log2++;
  }
  for (int s = 0; s < shift_bits_log2; ++s) {
    gf2_square(tmp, m);
    std::memcpy(m, tmp, sizeof(m));
  }
  std::memcpy(comb_mat[0], m, sizeof(m));
  for (int k = 1; k < 24; ++k) gf2_square(comb_mat[k], comb_mat[k - 1]);
}

static uint32_t crc32c_sw(const uint8_t* p, size_t n, uint32_t crc_in) {
  uint32_t crc = crc_in ^ 0xFFFFFFFFu;
  while (n && ((uintptr_t)p & 7)) { crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xFF]; --n; }
#if defined(__SSE4_2__)
  while (n >= 8) {
    crc = (uint32_t)__builtin_ia32_crc32di(crc, *(const uint64_t*)p);
    p += 8; n -= 8;
  }
  while (n) { crc = __builtin_ia32_crc32qi(crc, *p++); --n; }
#else
  while (n >= 8) {
    uint64_t v = *(const uint64_t*)p;
    uint32_t lo = (uint32_t)v ^ crc, hi = (uint32_t)(v >> 32);
    crc = crc_tab[7][lo & 0xFF] ^ crc_tab[6][(lo >> 8) & 0xFF] ^
          crc_tab[5][(lo >> 16) & 0xFF] ^ crc_tab[4][lo >> 24] ^
          crc_tab[3][hi & 0xFF] ^ crc_tab[2][(hi >> 8) & 0xFF] ^
          crc_tab[1][(hi >> 16) & 0xFF] ^ crc_tab[0][hi >> 24];
    p += 8; n -= 8;
  }
  while (n) { crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xFF]; --n; }
#endif
  return crc ^ 0xFFFFFFFFu;
}

// shift a finalized crc over `len` zero bytes (zlib crc32_combine algebra)
static uint32_t crc32c_shift(uint32_t crc, uint64_t len) {
  if (!len) return crc;
  uint32_t m[32], sq[32];
  m[0] = CRC_POLY;
  for (int i = 1; i < 32; ++i) m[i] = 1u << (i - 1);
  // m = M1 (one bit). apply for each set bit of 8*len.
  uint64_t bits = 8ull * len;
  while (bits) {
    if (bits & 1) crc = gf2_apply(m, crc);
    bits >>= 1;
    if (bits) { gf2_square(sq, m); std::memcpy(m, sq, sizeof(m)); }
  }
  return crc;
}

static uint32_t crc32c_combine(uint32_t crc1, uint32_t crc2, uint64_t len2) {
  return crc32c_shift(crc1, len2) ^ crc2;
}

// ---------------------------------------------------------------------------
// Kernels — single translation unit (device symbols + hipMemcpyToSymbol
// need no -fgpu-rdc this way)
// ---------------------------------------------------------------------------
#include "kernels.hip"
#include "lz4.hip"

// ---------------------------------------------------------------------------
// GPU availability
// ---------------------------------------------------------------------------

static int cached_device_count = -2;
static int device_count() {
  if (cached_device_count == -2) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    cached_device_count = n;
  }
  return cached_device_count;
}

// ---------------------------------------------------------------------------
// Arena
// ---------------------------------------------------------------------------

#define SPOOL 16  // parallel copy streams per device arena

struct Arena {
  void* base = nullptr;
  size_t cap = 0;
  int device = -1;       // -1 = host memory (CPU fallback / MEM tier)
  bool pinned = false;   // host arena allocated with hipHostMalloc
  hipStream_t stream = nullptr;       // copy stream
  hipStream_t kstream = nullptr;      // kernel stream
  // stream pool: concurrent readers each grab a slot so large copies
  // don't serialize on one stream/lock
  hipStream_t spool[SPOOL] = {};
  std::mutex spool_mu[SPOOL];
  std::atomic<uint32_t> srr{0};
  // pinned staging ring (device arenas)
  std::vector<void*> pin;
  std::vector<hipEvent_t> ev;
  size_t pin_sz = 0;
  // device scratch for gather/crc results
  void* scratch = nullptr;
  size_t scratch_sz = 0;
  bool crc_tables_uploaded = false;
  // true when base came from hipIpcOpenMemHandle (another process's
  // arena): destroy closes the mapping instead of freeing
  bool ipc_imported = false;
  std::mutex mu;

  bool is_dev() const { return device >= 0; }
};

static std::vector<Arena*> g_arenas;
static std::mutex g_arenas_mu;

static Arena* get_arena(int h) {
  std::lock_guard<std::mutex> g(g_arenas_mu);
  if (h < 0 || h >= (int)g_arenas.size() || !g_arenas[h])
    throw std::runtime_error("bad arena handle");
  return g_arenas[h];
}

static int arena_create(int device, size_t cap, size_t staging_bytes,
                        int staging_count, bool host_pinned) {
  auto* a = new Arena();
  a->device = device;
  a->cap = cap;
  if (device >= 0) {
    if (device >= device_count()) {
      delete a;
      throw std::runtime_error("no such GPU device");
    }
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipMalloc(&a->base, cap));
    HIP_CHECK(hipStreamCreateWithFlags(&a->stream, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&a->kstream, hipStreamNonBlocking));
    for (int i = 0; i < SPOOL; ++i)
      HIP_CHECK(hipStreamCreateWithFlags(&a->spool[i], hipStreamNonBlocking));
    a->pin_sz = staging_bytes;
    a->pin.resize(staging_count);
    a->ev.resize(staging_count);
    for (int i = 0; i < staging_count; ++i) {
      HIP_CHECK(hipHostMalloc(&a->pin[i], staging_bytes, hipHostMallocDefault));
      HIP_CHECK(hipEventCreateWithFlags(&a->ev[i], hipEventDisableTiming));
    }
  } else {
    if (host_pinned && device_count() > 0) {
      HIP_CHECK(hipHostMalloc(&a->base, cap, hipHostMallocDefault));
      a->pinned = true;   // pinning faults every page in
    } else {
      // 2 MiB-aligned + MADV_HUGEPAGE BEFORE the faulting memset: plain
      // malloc starts on 4 KiB pages and copies run TLB-bound at ~2
      // GiB/s until khugepaged collapses them (~7.6 GiB/s after);
      // faulting straight onto huge pages gets the fast path from the
      // first write
      size_t aligned = (cap + ((2u << 20) - 1)) & ~size_t((2u << 20) - 1);
      if (posix_memalign(&a->base, 2u << 20, aligned) != 0) {
        delete a;
        throw std::bad_alloc();
      }
#ifdef MADV_HUGEPAGE
      madvise(a->base, aligned, MADV_HUGEPAGE);
#endif
      std::memset(a->base, 0, cap);
    }
  }
  std::lock_guard<std::mutex> g(g_arenas_mu);
  for (size_t i = 0; i < g_arenas.size(); ++i)
    if (!g_arenas[i]) { g_arenas[i] = a; return (int)i; }
  g_arenas.push_back(a);
  return (int)g_arenas.size() - 1;
}

static void arena_destroy(int h) {
  Arena* a;
  {
    std::lock_guard<std::mutex> g(g_arenas_mu);
    if (h < 0 || h >= (int)g_arenas.size() || !g_arenas[h]) return;
    a = g_arenas[h];
    g_arenas[h] = nullptr;
  }
  if (a->is_dev()) {
    hipSetDevice(a->device);
    hipStreamSynchronize(a->stream);
    for (auto* p : a->pin) hipHostFree(p);
    for (auto e : a->ev) hipEventDestroy(e);
    if (a->scratch) hipFree(a->scratch);
    if (a->ipc_imported) hipIpcCloseMemHandle(a->base);
    else hipFree(a->base);
    hipStreamDestroy(a->stream);
    hipStreamDestroy(a->kstream);
    for (int i = 0; i < SPOOL; ++i)
      if (a->spool[i]) hipStreamDestroy(a->spool[i]);
  } else if (a->pinned) {
    hipHostFree(a->base);
  } else {
    std::free(a->base);
  }
  delete a;
}

// ---- cross-process arena sharing (hipIpc, dmabuf mode) ----
//
// A worker exports its device arena once; a colocated client process
// (FUSE daemon, another engine) opens the handle and reads published
// extents with direct D2H DMA in ITS OWN process — the production
// short-circuit for the one-process-per-GPU deployment shape.

static py::bytes arena_ipc_handle(int h) {
  Arena* a = get_arena(h);
  if (!a->is_dev())
    throw std::runtime_error("ipc export: device arenas only");
  hipIpcMemHandle_t ih;
  HIP_CHECK(hipSetDevice(a->device));
  HIP_CHECK(hipIpcGetMemHandle(&ih, a->base));
  return py::bytes((const char*)&ih, sizeof(ih));
}

static int arena_ipc_open(py::bytes handle, size_t cap, int device) {
  std::string hb = handle;
  if (hb.size() != sizeof(hipIpcMemHandle_t))
    throw std::runtime_error("ipc open: bad handle size");
  if (device < 0 || device >= device_count())
    throw std::runtime_error("ipc open: bad device");
  hipIpcMemHandle_t ih;
  std::memcpy(&ih, hb.data(), sizeof(ih));
  auto* a = new Arena();
  a->device = device;
  a->cap = cap;
  a->ipc_imported = true;
  HIP_CHECK(hipSetDevice(device));
  hipError_t e = hipIpcOpenMemHandle(&a->base, ih,
                                     hipIpcMemLazyEnablePeerAccess);
  if (e != hipSuccess) {
    delete a;
    throw std::runtime_error(std::string("hipIpcOpenMemHandle: ") +
                             hipGetErrorString(e));
  }
  HIP_CHECK(hipStreamCreateWithFlags(&a->stream, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&a->kstream, hipStreamNonBlocking));
  for (int i = 0; i < SPOOL; ++i)
    HIP_CHECK(hipStreamCreateWithFlags(&a->spool[i], hipStreamNonBlocking));
  // no staging ring: reads ride the pooled bounce pairs / pinned dests
  std::lock_guard<std::mutex> g(g_arenas_mu);
  for (size_t i = 0; i < g_arenas.size(); ++i)
    if (!g_arenas[i]) { g_arenas[i] = a; return (int)i; }
  g_arenas.push_back(a);
  return (int)g_arenas.size() - 1;
}

static void ensure_scratch(Arena* a, size_t n) {
  if (a->scratch_sz >= n) return;
  if (a->scratch) HIP_CHECK(hipFree(a->scratch));
  HIP_CHECK(hipMalloc(&a->scratch, n));
  a->scratch_sz = n;
}

static void ensure_crc_tables(Arena* a) {
  if (a->crc_tables_uploaded) return;
  crc_init_tables();
  HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(g_crc_tab), crc_tab, sizeof(crc_tab)));
  HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(g_comb_mat), comb_mat, sizeof(comb_mat)));
  a->crc_tables_uploaded = true;
}

// ---- host<->arena copies with pinned-ring pipelining ----

static void check_range(Arena* a, uint64_t off, uint64_t n) {
  if (off + n > a->cap) throw std::runtime_error("arena range out of bounds");
}

static bool is_pinned_host(const void* p) {
  hipPointerAttribute_t at;
  if (hipPointerGetAttributes(&at, p) != hipSuccess) {
    (void)hipGetLastError();   // clear the error
    return false;
  }
  return at.type == hipMemoryTypeHost;
}

// small or pinned-destination copies skip the ring: one hipMemcpyAsync on
// a pooled stream (parallel across caller threads, no global lock)
static const uint64_t RING_THRESHOLD = 2u << 20;

// ------------------------------------------------------- pinned bounce pool

// hipHostMalloc costs ~1 ms: a per-device free-list of double-buffer
// bounce pairs makes HBM stream opens O(microseconds) after warmup
struct BouncePair {
  void* pin[2] = {nullptr, nullptr};
  hipEvent_t ev[2] = {};
  int device = -1;
};

static constexpr size_t kBounceSz = 8 << 20;

struct BouncePool {
  std::mutex mu;
  std::unordered_map<int, std::vector<BouncePair*>> free_by_dev;
};
static BouncePool g_bounce;

static BouncePair* bounce_acquire(int device) {
  {
    std::lock_guard<std::mutex> g(g_bounce.mu);
    auto& v = g_bounce.free_by_dev[device];
    if (!v.empty()) {
      BouncePair* b = v.back();
      v.pop_back();
      return b;
    }
  }
  auto* b = new BouncePair();
  b->device = device;
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipHostMalloc(&b->pin[0], kBounceSz, hipHostMallocDefault));
  HIP_CHECK(hipHostMalloc(&b->pin[1], kBounceSz, hipHostMallocDefault));
  HIP_CHECK(hipEventCreateWithFlags(&b->ev[0], hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&b->ev[1], hipEventDisableTiming));
  return b;
}

static void bounce_release(BouncePair* b) {
  std::lock_guard<std::mutex> g(g_bounce.mu);
  auto& v = g_bounce.free_by_dev[b->device];
  if (v.size() >= 32) {
    hipEventDestroy(b->ev[0]);
    hipEventDestroy(b->ev[1]);
    hipHostFree(b->pin[0]);
    hipHostFree(b->pin[1]);
    delete b;
    return;
  }
  v.push_back(b);
}

// per-caller-thread stream: zero contention for concurrent small copies
// (the 4K-IOPS path).  hipMemcpy on the null stream and ROCm's internal
// pageable-staging lock both serialize; a thread_local stream does not.
// Streams are destroyed when their thread exits (no leak under thread
// churn) — except during process teardown, when the HIP runtime may
// already be gone.
static std::atomic<bool> g_process_exiting{false};
static struct ExitFlagSetter {
  ~ExitFlagSetter() { g_process_exiting.store(true); }
} g_exit_flag_setter;

struct TlsStreams {
  hipStream_t s[64] = {};
  ~TlsStreams() {
    if (g_process_exiting.load()) return;
    for (auto x : s)
      if (x) (void)hipStreamDestroy(x);
  }
};

static hipStream_t thread_stream(int device) {
  thread_local TlsStreams tls;
  if (device < 0 || device >= 64) throw std::runtime_error("bad device");
  if (!tls.s[device]) {
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreateWithFlags(&tls.s[device], hipStreamNonBlocking));
  }
  return tls.s[device];
}

static void dev_read_direct(Arena* a, uint64_t off, uint8_t* dst, uint64_t n) {
  HIP_CHECK(hipSetDevice(a->device));
  hipStream_t s = thread_stream(a->device);
  HIP_CHECK(hipMemcpyAsync(dst, (const uint8_t*)a->base + off, n,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
}

static void dev_write_direct(Arena* a, uint64_t off, const uint8_t* src,
                             uint64_t n) {
  HIP_CHECK(hipSetDevice(a->device));
  hipStream_t s = thread_stream(a->device);
  HIP_CHECK(hipMemcpyAsync((uint8_t*)a->base + off, src, n,
                           hipMemcpyHostToDevice, s));
  HIP_CHECK(hipStreamSynchronize(s));
}

// device -> host buffer, chunked through the pinned ring: D2H DMA of chunk
// k overlaps the CPU memcpy of chunk k-1 (the staging pipeline of
// BASELINE.json's "pinned hipMemcpyAsync on a side stream").
// device -> unpinned host, chunked through a POOLED pinned double-buffer
// on the caller's stream (D2H DMA of chunk k overlaps the memcpy-out of
// chunk k-1); concurrent readers no longer serialize on one arena ring.
static void dev_read(Arena* a, uint64_t off, uint8_t* dst, uint64_t n) {
  if (n <= RING_THRESHOLD || is_pinned_host(dst)) {
    dev_read_direct(a, off, dst, n);
    return;
  }
  HIP_CHECK(hipSetDevice(a->device));
  BouncePair* bp = bounce_acquire(a->device);
  hipStream_t s = thread_stream(a->device);
  const uint8_t* src = (const uint8_t*)a->base + off;
  uint64_t cs = std::min<uint64_t>(kBounceSz,
                                   std::max<uint64_t>(1u << 20, (n + 1) / 2));
  uint64_t nchunks = (n + cs - 1) / cs;
  try {
    for (uint64_t c = 0; c < nchunks; ++c) {
      uint64_t coff = c * cs;
      uint64_t clen = std::min<uint64_t>(cs, n - coff);
      int slot = (int)(c & 1);
      HIP_CHECK(hipMemcpyAsync(bp->pin[slot], src + coff, clen,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipEventRecord(bp->ev[slot], s));
      if (c > 0) {
        // drain the previous chunk while this one is in flight
        int prev = (int)((c - 1) & 1);
        HIP_CHECK(hipEventSynchronize(bp->ev[prev]));
        uint64_t poff = (c - 1) * cs;
        std::memcpy(dst + poff, bp->pin[prev],
                    std::min<uint64_t>(cs, n - poff));
      }
    }
    int last = (int)((nchunks - 1) & 1);
    HIP_CHECK(hipEventSynchronize(bp->ev[last]));
    uint64_t loff = (nchunks - 1) * cs;
    std::memcpy(dst + loff, bp->pin[last],
                std::min<uint64_t>(cs, n - loff));
  } catch (...) {
    (void)hipStreamSynchronize(s);   // no in-flight DMA may outlive bp
    bounce_release(bp);
    throw;
  }
  bounce_release(bp);
}

// host -> device, chunked through a POOLED pinned double-buffer on the
// caller's stream: concurrent writers (multi-file ingest, replication
// pushes) each pipeline their own H2D DMAs instead of serializing on
// one arena-wide staging ring.
static void dev_write(Arena* a, uint64_t off, const uint8_t* src, uint64_t n) {
  if (n <= RING_THRESHOLD || is_pinned_host(src)) {
    dev_write_direct(a, off, src, n);
    return;
  }
  HIP_CHECK(hipSetDevice(a->device));
  BouncePair* bp = bounce_acquire(a->device);
  hipStream_t s = thread_stream(a->device);
  uint8_t* dst = (uint8_t*)a->base + off;
  // chunk so every call gets >= 2 chunks: the staging memcpy of chunk
  // k overlaps the H2D DMA of chunk k-1 even for a single 4 MiB write
  uint64_t cs = std::min<uint64_t>(kBounceSz,
                                   std::max<uint64_t>(1u << 20, (n + 1) / 2));
  uint64_t nchunks = (n + cs - 1) / cs;
  try {
    for (uint64_t c = 0; c < nchunks; ++c) {
      uint64_t coff = c * cs;
      uint64_t clen = std::min<uint64_t>(cs, n - coff);
      int slot = (int)(c & 1);
      if (c >= 2) HIP_CHECK(hipEventSynchronize(bp->ev[slot]));
      std::memcpy(bp->pin[slot], src + coff, clen);
      HIP_CHECK(hipMemcpyAsync(dst + coff, bp->pin[slot], clen,
                               hipMemcpyHostToDevice, s));
      HIP_CHECK(hipEventRecord(bp->ev[slot], s));
    }
    HIP_CHECK(hipStreamSynchronize(s));
  } catch (...) {
    (void)hipStreamSynchronize(s);   // no in-flight DMA may outlive bp
    bounce_release(bp);
    throw;
  }
  bounce_release(bp);
}

static void arena_read(int h, uint64_t off, py::buffer buf, uint64_t buf_off,
                       uint64_t n) {
  Arena* a = get_arena(h);
  py::buffer_info info = buf.request(true);
  uint64_t cap = (uint64_t)info.size * (uint64_t)info.itemsize;
  if (buf_off + n > cap) throw std::runtime_error("dst buffer too small");
  check_range(a, off, n);
  uint8_t* dst = (uint8_t*)info.ptr + buf_off;
  py::gil_scoped_release rel;
  if (a->is_dev()) dev_read(a, off, dst, n);
  else std::memcpy(dst, (uint8_t*)a->base + off, n);
}

static void arena_write(int h, uint64_t off, py::buffer buf, uint64_t buf_off,
                        uint64_t n) {
  Arena* a = get_arena(h);
  py::buffer_info info = buf.request(false);
  uint64_t cap = (uint64_t)info.size * (uint64_t)info.itemsize;
  if (buf_off + n > cap) throw std::runtime_error("src buffer too small");
  check_range(a, off, n);
  const uint8_t* src = (const uint8_t*)info.ptr + buf_off;
  py::gil_scoped_release rel;
  if (a->is_dev()) dev_write(a, off, src, n);
  else std::memcpy((uint8_t*)a->base + off, src, n);
}

// raw-pointer variants (torch tensors, pinned reply buffers, other arenas'
// memory).  Uses the stream POOL: concurrent FUSE channel threads each get
// their own stream, so large D2H copies overlap instead of serializing.
static void arena_read_ptr(int h, uint64_t off, uintptr_t dst, uint64_t n,
                           bool dst_is_device) {
  Arena* a = get_arena(h);
  check_range(a, off, n);
  py::gil_scoped_release rel;
  if (a->is_dev()) {
    HIP_CHECK(hipSetDevice(a->device));
    hipStream_t s = thread_stream(a->device);
    HIP_CHECK(hipMemcpyAsync((void*)dst, (uint8_t*)a->base + off, n,
                             dst_is_device ? hipMemcpyDeviceToDevice
                                           : hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  } else if (dst_is_device) {
    HIP_CHECK(hipMemcpy((void*)dst, (uint8_t*)a->base + off, n,
                        hipMemcpyHostToDevice));
  } else {
    std::memcpy((void*)dst, (uint8_t*)a->base + off, n);
  }
}

static void arena_write_ptr(int h, uint64_t off, uintptr_t src, uint64_t n,
                            bool src_is_device) {
  Arena* a = get_arena(h);
  check_range(a, off, n);
  py::gil_scoped_release rel;
  if (a->is_dev()) {
    HIP_CHECK(hipSetDevice(a->device));
    hipStream_t s = thread_stream(a->device);
    HIP_CHECK(hipMemcpyAsync((uint8_t*)a->base + off, (const void*)src, n,
                             src_is_device ? hipMemcpyDeviceToDevice
                                           : hipMemcpyHostToDevice, s));
    HIP_CHECK(hipStreamSynchronize(s));
  } else if (src_is_device) {
    HIP_CHECK(hipMemcpy((uint8_t*)a->base + off, (const void*)src, n,
                        hipMemcpyDeviceToHost));
  } else {
    std::memcpy((uint8_t*)a->base + off, (const void*)src, n);
  }
}

// arena -> arena (same or cross device / host tiers)
static void arena_copy(int dst_h, uint64_t dst_off, int src_h, uint64_t src_off,
                       uint64_t n) {
  Arena* d = get_arena(dst_h);
  Arena* s = get_arena(src_h);
  check_range(d, dst_off, n);
  check_range(s, src_off, n);
  py::gil_scoped_release rel;
  uint8_t* dp = (uint8_t*)d->base + dst_off;
  uint8_t* sp = (uint8_t*)s->base + src_off;
  if (!d->is_dev() && !s->is_dev()) {
    std::memcpy(dp, sp, n);
    return;
  }
  Arena* deva = d->is_dev() ? d : s;
  std::lock_guard<std::mutex> g(deva->mu);
  HIP_CHECK(hipSetDevice(deva->device));
  hipMemcpyKind kind =
      d->is_dev() ? (s->is_dev() ? hipMemcpyDeviceToDevice : hipMemcpyHostToDevice)
                  : hipMemcpyDeviceToHost;
  if (d->is_dev() && s->is_dev() && d->device != s->device) {
    HIP_CHECK(hipMemcpyPeerAsync(dp, d->device, sp, s->device, n, deva->stream));
  } else {
    HIP_CHECK(hipMemcpyAsync(dp, sp, n, kind, deva->stream));
  }
  HIP_CHECK(hipStreamSynchronize(deva->stream));
}

// ---- kernels ----

static int grid_for(uint64_t work_items) {
  // >> 256 workgroups to fill 8 XCDs; cap for grid-stride loops
  uint64_t g = (work_items + WG - 1) / WG;
  if (g > 8192) g = 8192;
  if (g < 1) g = 1;
  return (int)g;
}

static void arena_fill(int h, uint64_t off, uint64_t n, int value) {
  Arena* a = get_arena(h);
  check_range(a, off, n);
  py::gil_scoped_release rel;
  if (!a->is_dev()) {
    std::memset((uint8_t*)a->base + off, value, n);
    return;
  }
  std::lock_guard<std::mutex> g(a->mu);
  HIP_CHECK(hipSetDevice(a->device));
  (void)hipGetLastError();   // clear stale per-thread state
  hipLaunchKernelGGL(fill_kernel, dim3(grid_for(n / 16 + 1)), dim3(WG), 0,
                     a->kstream, (uint8_t*)a->base + off, (uint8_t)value, n);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(a->kstream));
}

static uint32_t arena_crc32c(int h, uint64_t off, uint64_t n) {
  Arena* a = get_arena(h);
  check_range(a, off, n);
  if (!a->is_dev()) {
    const uint8_t* p = (const uint8_t*)a->base + off;
    py::gil_scoped_release rel;
    crc_init_tables();
    return crc32c_sw(p, n, 0);
  }
  py::gil_scoped_release rel;
  std::lock_guard<std::mutex> g(a->mu);
  HIP_CHECK(hipSetDevice(a->device));
  ensure_crc_tables(a);
  const uint8_t* p = (const uint8_t*)a->base + off;
  if (((uintptr_t)p & 7) != 0) throw std::runtime_error("crc32c needs 8B-aligned offset");
  uint64_t n_sub = n / CRC_SUB;
  uint64_t tail = n - n_sub * CRC_SUB;
  uint32_t result = 0;
  if (n_sub) {
    uint32_t n_wg = (uint32_t)((n_sub + WG - 1) / WG);
    ensure_scratch(a, n_wg * sizeof(uint32_t) + a->pin_sz);
    uint32_t* d_out = (uint32_t*)a->scratch;
    (void)hipGetLastError();   // clear stale per-thread state
    hipLaunchKernelGGL(crc32c_kernel, dim3(n_wg), dim3(WG), 0, a->kstream, p,
                       n_sub, d_out);
    HIP_CHECK(hipGetLastError());
    std::vector<uint32_t> host_out(n_wg);
    HIP_CHECK(hipMemcpyAsync(host_out.data(), d_out, n_wg * sizeof(uint32_t),
                             hipMemcpyDeviceToHost, a->kstream));
    HIP_CHECK(hipStreamSynchronize(a->kstream));
    // combine per-workgroup crcs (each covers WG*CRC_SUB except the last)
    uint64_t remain = n_sub;
    result = host_out[0];
    uint64_t covered0 = std::min<uint64_t>(WG, remain);
    remain -= covered0;
    for (uint32_t w = 1; w < n_wg; ++w) {
      uint64_t cov = std::min<uint64_t>(WG, remain);
      remain -= cov;
      result = crc32c_combine(result, host_out[w], cov * CRC_SUB);
    }
  }
  if (tail) {
    // read the tail back through a pinned slot and crc on host
    uint8_t* pin0 = (uint8_t*)(a->pin.empty() ? nullptr : a->pin[0]);
    std::vector<uint8_t> tmp;
    uint8_t* dst;
    if (pin0 && tail <= a->pin_sz) dst = pin0;
    else { tmp.resize(tail); dst = tmp.data(); }
    HIP_CHECK(hipMemcpyAsync(dst, p + n_sub * CRC_SUB, tail,
                             hipMemcpyDeviceToHost, a->kstream));
    HIP_CHECK(hipStreamSynchronize(a->kstream));
    uint32_t tail_crc = crc32c_sw(dst, tail, 0);
    result = n_sub ? crc32c_combine(result, tail_crc, tail) : tail_crc;
  }
  return result;
}

static uint32_t crc32c_buf(py::buffer buf, uint32_t init) {
  py::buffer_info info = buf.request(false);
  const uint8_t* p = (const uint8_t*)info.ptr;
  size_t n = (size_t)info.size * info.itemsize;
  py::gil_scoped_release rel;
  crc_init_tables();
  return crc32c_sw(p, n, init);
}

// gather device extents into a host buffer: kernel packs extents into
// contiguous scratch on-device, then one pipelined D2H.
static void arena_gather(int h, std::vector<std::pair<uint64_t, uint64_t>> ext,
                         py::buffer out, uint64_t out_off) {
  Arena* a = get_arena(h);
  py::buffer_info info = out.request(true);
  uint64_t total = 0;
  for (auto& e : ext) { check_range(a, e.first, e.second); total += e.second; }
  if (out_off + total > (uint64_t)info.size * info.itemsize)
    throw std::runtime_error("gather dst too small");
  uint8_t* dst = (uint8_t*)info.ptr + out_off;
  py::gil_scoped_release rel;
  if (!a->is_dev()) {
    uint64_t o = 0;
    for (auto& e : ext) {
      std::memcpy(dst + o, (uint8_t*)a->base + e.first, e.second);
      o += e.second;
    }
    return;
  }
  std::lock_guard<std::mutex> g(a->mu);
  HIP_CHECK(hipSetDevice(a->device));
  // build extent + tile tables
  std::vector<Extent> exts(ext.size());
  std::vector<uint32_t> tile_ext;
  std::vector<uint64_t> tile_off;
  uint64_t o = 0;
  for (size_t i = 0; i < ext.size(); ++i) {
    exts[i] = {ext[i].first, o, ext[i].second};
    for (uint64_t t = 0; t < ext[i].second; t += TILE) {
      tile_ext.push_back((uint32_t)i);
      tile_off.push_back(t);
    }
    o += ext[i].second;
  }
  size_t meta = exts.size() * sizeof(Extent) +
                tile_ext.size() * (sizeof(uint32_t) + sizeof(uint64_t));
  ensure_scratch(a, total + meta + 64);
  uint8_t* d_pack = (uint8_t*)a->scratch;
  Extent* d_ext = (Extent*)(d_pack + ((total + 63) & ~63ull));
  uint32_t* d_te = (uint32_t*)(d_ext + exts.size());
  uint64_t* d_to = (uint64_t*)(((uintptr_t)(d_te + tile_ext.size()) + 7) & ~7ull);
  HIP_CHECK(hipMemcpyAsync(d_ext, exts.data(), exts.size() * sizeof(Extent),
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_te, tile_ext.data(),
                           tile_ext.size() * sizeof(uint32_t),
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_to, tile_off.data(),
                           tile_off.size() * sizeof(uint64_t),
                           hipMemcpyHostToDevice, a->kstream));
  int grid = (int)std::min<size_t>(tile_ext.size(), 8192);
  (void)hipGetLastError();   // clear stale per-thread state
  hipLaunchKernelGGL(copy_extents_kernel, dim3(grid), dim3(WG), 0, a->kstream,
                     (const uint8_t*)a->base, d_pack, d_ext, d_te, d_to,
                     (uint32_t)tile_ext.size());
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(a->kstream));
  // pipelined D2H of the packed region
  size_t nb = a->pin.size();
  uint64_t nchunks = (total + a->pin_sz - 1) / a->pin_sz;
  for (uint64_t c = 0; c < nchunks; ++c) {
    uint64_t coff = c * a->pin_sz;
    uint64_t clen = std::min<uint64_t>(a->pin_sz, total - coff);
    size_t slot = c % nb;
    if (c >= nb) HIP_CHECK(hipEventSynchronize(a->ev[slot]));
    HIP_CHECK(hipMemcpyAsync(a->pin[slot], d_pack + coff, clen,
                             hipMemcpyDeviceToHost, a->stream));
    HIP_CHECK(hipEventRecord(a->ev[slot], a->stream));
    if (c + 1 == nchunks || ((c + 1) % nb) == 0) {
      uint64_t first = (c / nb) * nb;
      for (uint64_t d2 = first; d2 <= c; ++d2) {
        size_t ds = d2 % nb;
        HIP_CHECK(hipEventSynchronize(a->ev[ds]));
        uint64_t doff = d2 * a->pin_sz;
        std::memcpy(dst + doff, a->pin[ds],
                    std::min<uint64_t>(a->pin_sz, total - doff));
      }
    }
  }
}

// gather arena extents into a caller-supplied pointer with per-extent dst
// offsets.  Device arena + device dst = pure on-chip D2D via
// copy_extents_kernel (no host hop — the training-ingest fast path: tar
// sample payloads living in the HBM cache land directly in a torch
// device tensor).  Host arena = plain memcpy scatter into a host dst.
static void arena_gather_ptr(
    int h, std::vector<std::tuple<uint64_t, uint64_t, uint64_t>> ext,
    uintptr_t dst_ptr) {
  Arena* a = get_arena(h);
  for (auto& e : ext) check_range(a, std::get<0>(e), std::get<2>(e));
  uint8_t* dst = (uint8_t*)dst_ptr;
  py::gil_scoped_release rel;
  if (!a->is_dev()) {
    for (auto& e : ext)
      std::memcpy(dst + std::get<1>(e), (uint8_t*)a->base + std::get<0>(e),
                  std::get<2>(e));
    return;
  }
  std::lock_guard<std::mutex> g(a->mu);
  HIP_CHECK(hipSetDevice(a->device));
  std::vector<Extent> exts(ext.size());
  std::vector<uint32_t> tile_ext;
  std::vector<uint64_t> tile_off;
  for (size_t i = 0; i < ext.size(); ++i) {
    exts[i] = {std::get<0>(ext[i]), std::get<1>(ext[i]), std::get<2>(ext[i])};
    for (uint64_t t = 0; t < exts[i].len; t += TILE) {
      tile_ext.push_back((uint32_t)i);
      tile_off.push_back(t);
    }
  }
  size_t meta = exts.size() * sizeof(Extent) +
                tile_ext.size() * (sizeof(uint32_t) + sizeof(uint64_t)) + 64;
  ensure_scratch(a, meta);
  Extent* d_ext = (Extent*)a->scratch;
  uint32_t* d_te = (uint32_t*)(d_ext + exts.size());
  uint64_t* d_to = (uint64_t*)(((uintptr_t)(d_te + tile_ext.size()) + 7) & ~7ull);
  HIP_CHECK(hipMemcpyAsync(d_ext, exts.data(), exts.size() * sizeof(Extent),
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_te, tile_ext.data(),
                           tile_ext.size() * sizeof(uint32_t),
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_to, tile_off.data(),
                           tile_off.size() * sizeof(uint64_t),
                           hipMemcpyHostToDevice, a->kstream));
  int grid = (int)std::min<size_t>(tile_ext.size(), 8192);
  (void)hipGetLastError();   // clear stale per-thread state
  hipLaunchKernelGGL(copy_extents_kernel, dim3(grid), dim3(WG), 0, a->kstream,
                     (const uint8_t*)a->base, dst, d_ext, d_te, d_to,
                     (uint32_t)tile_ext.size());
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(a->kstream));
}

static uintptr_t arena_base_ptr(int h) {
  return (uintptr_t)get_arena(h)->base;
}

// batched small reads (the fio iodepth>1 analog): issue every copy async
// on the caller's thread-local stream, sync ONCE — amortizes launch+sync
// cost over the batch (4K random-read IOPS path)
static void arena_read_batch(int h, const std::vector<uint64_t>& offs,
                             const std::vector<uint64_t>& dsts, uint64_t n) {
  Arena* a = get_arena(h);
  if (offs.size() != dsts.size())
    throw std::runtime_error("arena_read_batch: offs/dsts mismatch");
  for (uint64_t off : offs) check_range(a, off, n);
  py::gil_scoped_release rel;
  if (!a->is_dev()) {
    for (size_t i = 0; i < offs.size(); ++i)
      std::memcpy((uint8_t*)dsts[i], (uint8_t*)a->base + offs[i], n);
    return;
  }
  HIP_CHECK(hipSetDevice(a->device));
  hipStream_t s = thread_stream(a->device);
  for (size_t i = 0; i < offs.size(); ++i)
    HIP_CHECK(hipMemcpyAsync((void*)dsts[i], (uint8_t*)a->base + offs[i], n,
                             hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
}

// ---------------------------------------------------------------------------
// Registered readers: a file's arena extent table pinned in C++ so batched
// random preads resolve + issue + sync entirely GIL-free (the 4 KiB IOPS
// path; the client-side sibling of the native FUSE read registration).
// Safety contract (curvine_amd/client/reader.py): the SyncLocalReader
// holds its store readers open (block refcounts defer delete/demote) for
// the registration's whole lifetime and unregisters before closing them.
// ---------------------------------------------------------------------------

struct RdExt { uint64_t file_off, len; int arena; uint64_t arena_off; };
struct RdReader { uint64_t length; std::vector<RdExt> exts; };
static std::mutex g_rd_mu;
static std::unordered_map<int64_t, RdReader> g_rd;
static int64_t g_rd_next = 1;

static int64_t reader_register(
    std::vector<std::tuple<uint64_t, uint64_t, int, uint64_t>> exts,
    uint64_t length) {
  RdReader r;
  r.length = length;
  r.exts.reserve(exts.size());
  uint64_t prev = 0;
  for (auto& e : exts) {
    if (std::get<0>(e) < prev)
      throw std::runtime_error("reader_register: extents must be sorted");
    prev = std::get<0>(e);
    get_arena(std::get<2>(e));   // validate handle
    r.exts.push_back({std::get<0>(e), std::get<1>(e), std::get<2>(e),
                      std::get<3>(e)});
  }
  std::lock_guard<std::mutex> g(g_rd_mu);
  int64_t rid = g_rd_next++;
  g_rd[rid] = std::move(r);
  return rid;
}

static void reader_unregister(int64_t rid) {
  std::lock_guard<std::mutex> g(g_rd_mu);
  g_rd.erase(rid);
}

// fixed-size batched preads: for each file offset, land n bytes at
// dst_ptr + i*stride.  Returns the indices it could NOT serve (reads
// spanning extent boundaries or past EOF) so the caller falls back for
// exactly those.
static py::list reader_pread_batch(int64_t rid,
                                   const std::vector<uint64_t>& offs,
                                   uint64_t n, uintptr_t dst_ptr,
                                   uint64_t stride) {
  RdReader* r;
  {
    std::lock_guard<std::mutex> g(g_rd_mu);
    auto it = g_rd.find(rid);
    if (it == g_rd.end()) throw std::runtime_error("bad reader id");
    r = &it->second;
  }
  std::vector<uint32_t> skipped;
  {
    py::gil_scoped_release rel;
    int used_dev[8];
    int n_dev = 0;
    const auto& exts = r->exts;
    for (size_t i = 0; i < offs.size(); ++i) {
      uint64_t off = offs[i];
      const RdExt* e = nullptr;
      if (off + n <= r->length && !exts.empty()) {
        // binary search: last extent with file_off <= off
        size_t lo = 0, hi = exts.size();
        while (lo < hi) {
          size_t mid = (lo + hi) / 2;
          if (exts[mid].file_off <= off) lo = mid + 1; else hi = mid;
        }
        if (lo > 0 && off - exts[lo - 1].file_off + n <= exts[lo - 1].len)
          e = &exts[lo - 1];
      }
      if (e == nullptr) {
        skipped.push_back((uint32_t)i);
        continue;
      }
      uint64_t boff = off - e->file_off;
      Arena* a = get_arena(e->arena);
      uint8_t* dst = (uint8_t*)dst_ptr + i * stride;
      const uint8_t* src = (uint8_t*)a->base + e->arena_off + boff;
      if (!a->is_dev()) {
        std::memcpy(dst, src, n);
      } else {
        HIP_CHECK(hipSetDevice(a->device));
        HIP_CHECK(hipMemcpyAsync(dst, src, n, hipMemcpyDeviceToHost,
                                 thread_stream(a->device)));
        bool seen = false;
        for (int d = 0; d < n_dev; d++) seen |= (used_dev[d] == a->device);
        if (!seen && n_dev < 8) used_dev[n_dev++] = a->device;
      }
    }
    for (int d = 0; d < n_dev; d++) {
      HIP_CHECK(hipSetDevice(used_dev[d]));
      HIP_CHECK(hipStreamSynchronize(thread_stream(used_dev[d])));
    }
  }
  py::list out;
  for (uint32_t i : skipped) out.append(i);
  return out;
}

// ---------------------------------------------------------------------------
// LZ4 container: compress/decompress (host) + decompress-into-arena (GPU)
// ---------------------------------------------------------------------------

static py::bytes lz4_compress_py(py::buffer buf) {
  py::buffer_info info = buf.request(false);
  const uint8_t* src = (const uint8_t*)info.ptr;
  size_t n = (size_t)info.size * info.itemsize;
  uint32_t n_chunks = (uint32_t)((n + LZ4_CHUNK - 1) / LZ4_CHUNK);
  std::vector<uint32_t> sizes(n_chunks);
  std::vector<uint8_t> out;
  out.reserve(n / 2 + 1024);
  size_t header = 4 + 8 + 4 + 4 + 4ull * n_chunks;
  out.resize(header);
  {
    py::gil_scoped_release rel;
    std::vector<uint8_t> tmp(LZ4_CHUNK + LZ4_CHUNK / 128 + 64);
    for (uint32_t c = 0; c < n_chunks; ++c) {
      size_t cn = std::min<size_t>(LZ4_CHUNK, n - (size_t)c * LZ4_CHUNK);
      size_t cs = lz4_compress_block(src + (size_t)c * LZ4_CHUNK, cn,
                                     tmp.data());
      sizes[c] = (uint32_t)cs;
      out.insert(out.end(), tmp.data(), tmp.data() + cs);
    }
    uint8_t* h = out.data();
    memcpy(h, &LZ4_MAGIC, 4);
    uint64_t raw = n;
    memcpy(h + 4, &raw, 8);
    uint32_t ck = LZ4_CHUNK;
    memcpy(h + 12, &ck, 4);
    memcpy(h + 16, &n_chunks, 4);
    memcpy(h + 20, sizes.data(), 4ull * n_chunks);
  }
  return py::bytes((const char*)out.data(), out.size());
}

struct Lz4Header {
  uint64_t raw_size;
  uint32_t chunk_size, n_chunks;
  const uint32_t* sizes;
  const uint8_t* payload;
};

static Lz4Header lz4_parse(const uint8_t* p, size_t n) {
  if (n < 20) throw std::runtime_error("lz4: short container");
  uint32_t magic;
  memcpy(&magic, p, 4);
  if (magic != LZ4_MAGIC) throw std::runtime_error("lz4: bad magic");
  Lz4Header h;
  memcpy(&h.raw_size, p + 4, 8);
  memcpy(&h.chunk_size, p + 12, 4);
  memcpy(&h.n_chunks, p + 16, 4);
  if (n < 20 + 4ull * h.n_chunks) throw std::runtime_error("lz4: truncated");
  h.sizes = (const uint32_t*)(p + 20);
  h.payload = p + 20 + 4ull * h.n_chunks;
  return h;
}

static py::bytes lz4_decompress_py(py::buffer buf) {
  py::buffer_info info = buf.request(false);
  const uint8_t* src = (const uint8_t*)info.ptr;
  size_t n = (size_t)info.size * info.itemsize;
  Lz4Header h = lz4_parse(src, n);
  std::string out(h.raw_size, '\0');
  {
    py::gil_scoped_release rel;
    const uint8_t* p = h.payload;
    for (uint32_t c = 0; c < h.n_chunks; ++c) {
      size_t cap = std::min<uint64_t>(h.chunk_size,
                                      h.raw_size - (uint64_t)c * h.chunk_size);
      size_t got = lz4_decompress_block_host(
          p, h.sizes[c], (uint8_t*)out.data() + (uint64_t)c * h.chunk_size,
          cap);
      if (got != cap) throw std::runtime_error("lz4: corrupt chunk");
      p += h.sizes[c];
    }
  }
  return py::bytes(out);
}

// decompress a CVLZ container directly into an arena (GPU kernel on device
// arenas: one workgroup per 64K chunk)
static uint64_t arena_lz4_decompress(int ah, uint64_t dst_off, py::buffer buf) {
  Arena* a = get_arena(ah);
  py::buffer_info info = buf.request(false);
  const uint8_t* src = (const uint8_t*)info.ptr;
  size_t n = (size_t)info.size * info.itemsize;
  Lz4Header h = lz4_parse(src, n);
  check_range(a, dst_off, h.raw_size);
  py::gil_scoped_release rel;
  if (!a->is_dev()) {
    const uint8_t* p = h.payload;
    for (uint32_t c = 0; c < h.n_chunks; ++c) {
      size_t cap = std::min<uint64_t>(h.chunk_size,
                                      h.raw_size - (uint64_t)c * h.chunk_size);
      size_t got = lz4_decompress_block_host(
          p, h.sizes[c],
          (uint8_t*)a->base + dst_off + (uint64_t)c * h.chunk_size, cap);
      if (got != cap) throw std::runtime_error("lz4: corrupt chunk");
      p += h.sizes[c];
    }
    return h.raw_size;
  }
  std::lock_guard<std::mutex> g(a->mu);
  HIP_CHECK(hipSetDevice(a->device));
  std::vector<uint32_t> offs(h.n_chunks);
  uint32_t acc = 0;
  for (uint32_t c = 0; c < h.n_chunks; ++c) {
    offs[c] = acc;
    acc += h.sizes[c];
  }
  size_t payload_n = acc;
  size_t meta = 8ull * h.n_chunks + 64;
  ensure_scratch(a, payload_n + meta + 64);
  uint8_t* d_comp = (uint8_t*)a->scratch;
  uint32_t* d_off = (uint32_t*)(d_comp + ((payload_n + 63) & ~63ull));
  uint32_t* d_len = d_off + h.n_chunks;
  int* d_err = (int*)(d_len + h.n_chunks);
  HIP_CHECK(hipMemcpyAsync(d_comp, h.payload, payload_n,
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_off, offs.data(), 4ull * h.n_chunks,
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemcpyAsync(d_len, h.sizes, 4ull * h.n_chunks,
                           hipMemcpyHostToDevice, a->kstream));
  HIP_CHECK(hipMemsetAsync(d_err, 0, 4, a->kstream));
  int grid = (int)std::min<uint32_t>(h.n_chunks, 16384);
  (void)hipGetLastError();   // clear stale per-thread state
  hipLaunchKernelGGL(lz4_decompress_kernel, dim3(grid), dim3(64), 0,
                     a->kstream, d_comp, d_off, d_len,
                     (uint8_t*)a->base + dst_off, h.chunk_size, h.raw_size,
                     h.n_chunks, d_err);
  HIP_CHECK(hipGetLastError());
  int err_host = 0;
  HIP_CHECK(hipMemcpyAsync(&err_host, d_err, 4, hipMemcpyDeviceToHost,
                           a->kstream));
  HIP_CHECK(hipStreamSynchronize(a->kstream));
  if (err_host) throw std::runtime_error(
      "lz4: corrupt chunk " + std::to_string(err_host - 1) + " (device)");
  return h.raw_size;
}

// ---------------------------------------------------------------------------
// DLPack export: zero-copy torch tensors over arena memory (uint8, 1-D).
// Lets RCCL collectives (torch.distributed "nccl" on ROCm) broadcast
// HBM-resident cache blocks over xGMI without a staging copy, and lets
// DataLoaders consume cached bytes as device tensors.
// ABI structs per dlpack v0.8 (stable).
// ---------------------------------------------------------------------------

extern "C" {
typedef struct { int32_t device_type; int32_t device_id; } DLDevice;
typedef struct { uint8_t code; uint8_t bits; uint16_t lanes; } DLDataType;
typedef struct {
  void* data; DLDevice device; int32_t ndim; DLDataType dtype;
  int64_t* shape; int64_t* strides; uint64_t byte_offset;
} DLTensor;
typedef struct DLManagedTensor {
  DLTensor dl_tensor; void* manager_ctx;
  void (*deleter)(struct DLManagedTensor*);
} DLManagedTensor;
}
static const int kDLCPU = 1, kDLROCM = 10;

struct DLWrap { DLManagedTensor mt; int64_t shape[1]; };

static void dl_deleter(DLManagedTensor* mt) {
  delete reinterpret_cast<DLWrap*>(mt->manager_ctx);
}

static void dl_capsule_destructor(PyObject* cap) {
  // torch renames the capsule to "used_dltensor" after consuming it
  if (PyCapsule_IsValid(cap, "dltensor")) {
    auto* mt = (DLManagedTensor*)PyCapsule_GetPointer(cap, "dltensor");
    if (mt && mt->deleter) mt->deleter(mt);
  }
}

static py::object arena_dlpack(int h, uint64_t off, uint64_t n) {
  Arena* a = get_arena(h);
  check_range(a, off, n);
  auto* w = new DLWrap();
  w->shape[0] = (int64_t)n;
  w->mt.dl_tensor.data = (uint8_t*)a->base + off;
  w->mt.dl_tensor.device = {a->is_dev() ? kDLROCM : kDLCPU,
                            a->is_dev() ? a->device : 0};
  w->mt.dl_tensor.ndim = 1;
  w->mt.dl_tensor.dtype = {0 /*kDLInt? 0=int*/, 8, 1};
  w->mt.dl_tensor.dtype.code = 1;  // kDLUInt
  w->mt.dl_tensor.shape = w->shape;
  w->mt.dl_tensor.strides = nullptr;
  w->mt.dl_tensor.byte_offset = 0;
  w->mt.manager_ctx = w;
  w->mt.deleter = dl_deleter;
  PyObject* cap = PyCapsule_New(&w->mt, "dltensor", dl_capsule_destructor);
  return py::reinterpret_steal<py::object>(cap);
}

static py::memoryview arena_host_view(int h, uint64_t off, uint64_t n) {
  Arena* a = get_arena(h);
  if (a->is_dev()) throw std::runtime_error("arena_host_view: device arena");
  check_range(a, off, n);
  return py::memoryview::from_memory((uint8_t*)a->base + off, n, false);
}

// ---------------------------------------------------------------------------
// Pinned host buffers (FUSE reply/request buffers: DMA lands directly in
// the buffer that is writev'd to /dev/fuse — no staging-ring hop)
// ---------------------------------------------------------------------------

struct PinnedBuf { void* ptr = nullptr; size_t n = 0; bool pinned = false; };
static std::vector<PinnedBuf> g_pins;
static std::mutex g_pins_mu;

static int pinned_alloc(size_t n) {
  PinnedBuf b;
  b.n = n;
  if (device_count() > 0) {
    HIP_CHECK(hipHostMalloc(&b.ptr, n, hipHostMallocDefault));
    b.pinned = true;
  } else {
    b.ptr = std::malloc(n);
    if (!b.ptr) throw std::bad_alloc();
  }
  std::lock_guard<std::mutex> g(g_pins_mu);
  for (size_t i = 0; i < g_pins.size(); ++i)
    if (!g_pins[i].ptr) { g_pins[i] = b; return (int)i; }
  g_pins.push_back(b);
  return (int)g_pins.size() - 1;
}

static void pinned_free(int id) {
  std::lock_guard<std::mutex> g(g_pins_mu);
  if (id < 0 || id >= (int)g_pins.size() || !g_pins[id].ptr) return;
  if (g_pins[id].pinned) hipHostFree(g_pins[id].ptr);
  else std::free(g_pins[id].ptr);
  g_pins[id] = PinnedBuf{};
}

static py::memoryview pinned_view(int id) {
  std::lock_guard<std::mutex> g(g_pins_mu);
  if (id < 0 || id >= (int)g_pins.size() || !g_pins[id].ptr)
    throw std::runtime_error("bad pinned buffer id");
  return py::memoryview::from_memory(g_pins[id].ptr, g_pins[id].n, false);
}

static uintptr_t pinned_ptr(int id) {
  std::lock_guard<std::mutex> g(g_pins_mu);
  if (id < 0 || id >= (int)g_pins.size() || !g_pins[id].ptr)
    throw std::runtime_error("bad pinned buffer id");
  return (uintptr_t)g_pins[id].ptr;
}

static py::dict arena_info(int h) {
  Arena* a = get_arena(h);
  py::dict d;
  d["capacity"] = a->cap;
  d["device"] = a->device;
  d["pinned"] = a->pinned;
  d["staging_bytes"] = a->pin_sz;
  d["staging_count"] = a->pin.size();
  return d;
}

// raw host->device copy for staged remote reads into consumer tensors
static void memcpy_h2d(uintptr_t dst, uintptr_t src, uint64_t n) {
  py::gil_scoped_release rel;
  HIP_CHECK(hipMemcpy((void*)dst, (const void*)src, n, hipMemcpyHostToDevice));
}

static void device_sync(int device) {
  py::gil_scoped_release rel;
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipDeviceSynchronize());
}

// native FUSE data loop (needs Arena/get_arena/thread_stream above)
#include "fuse_loop.hip"

// native metadata RPC frontend (pure epoll/C++, no GPU involvement)
#include "meta_server.cpp"
#include "data_server.cpp"
#include "sdk_abi.cpp"

static py::dict device_mem_info(int device) {
  size_t free_b = 0, total_b = 0;
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  py::dict d;
  d["free"] = free_b;
  d["total"] = total_b;
  return d;
}

PYBIND11_MODULE(_native, m) {
  m.doc() = "curvine_amd native data plane (HIP/CDNA4, gfx950)";
  m.def("device_count", &device_count);
  m.def("device_sync", &device_sync);
  m.def("memcpy_h2d", &memcpy_h2d);
  m.def("device_mem_info", &device_mem_info);
  m.def("arena_ipc_handle", &arena_ipc_handle);
  m.def("arena_ipc_open", &arena_ipc_open);
  m.def("arena_create", &arena_create, py::arg("device"), py::arg("capacity"),
        py::arg("staging_bytes") = 4 << 20, py::arg("staging_count") = 8,
        py::arg("host_pinned") = false);
  m.def("arena_destroy", &arena_destroy);
  m.def("arena_read", &arena_read);
  m.def("arena_write", &arena_write);
  m.def("arena_read_ptr", &arena_read_ptr);
  m.def("arena_write_ptr", &arena_write_ptr);
  m.def("arena_copy", &arena_copy);
  m.def("arena_fill", &arena_fill);
  m.def("arena_crc32c", &arena_crc32c);
  m.def("arena_gather", &arena_gather);
  m.def("arena_gather_ptr", &arena_gather_ptr);
  m.def("arena_base_ptr", &arena_base_ptr);
  m.def("arena_read_batch", &arena_read_batch);
  m.def("reader_register", &reader_register);
  m.def("reader_unregister", &reader_unregister);
  m.def("reader_pread_batch", &reader_pread_batch);
  m.def("arena_info", &arena_info);
  m.def("arena_dlpack", &arena_dlpack);
  m.def("arena_host_view", &arena_host_view);
  m.def("pinned_alloc", &pinned_alloc);
  m.def("pinned_free", &pinned_free);
  m.def("pinned_view", &pinned_view);
  m.def("pinned_ptr", &pinned_ptr);
  m.def("fuse_loop_create", &fuse_loop_create);
  m.def("fuse_loop_add_channel", &fuse_loop_add_channel);
  m.def("fuse_loop_register", &fuse_loop_register);
  m.def("fuse_loop_unregister", &fuse_loop_unregister);
  m.def("fuse_loop_register_write", &fuse_loop_register_write);
  m.def("fuse_loop_unregister_write", &fuse_loop_unregister_write);
  m.def("fuse_loop_write_state", &fuse_loop_write_state);
  m.def("fuse_loop_next_forward", &fuse_loop_next_forward);
  m.def("fuse_loop_stats", &fuse_loop_stats);
  m.def("fuse_loop_stop", &fuse_loop_stop);
  m.def("meta_create", &meta_create);
  m.def("meta_stop", &meta_stop_srv);
  m.def("meta_set_serving", &meta_set_serving);
  m.def("meta_upsert", &meta_upsert);
  m.def("meta_upsert_node", &meta_upsert_node);
  m.def("meta_touch", &meta_touch);
  m.def("meta_upsert_with_children", &meta_upsert_with_children);
  m.def("meta_add_child", &meta_add_child);
  m.def("meta_remove_child", &meta_remove_child);
  m.def("meta_drop", &meta_drop);
  m.def("meta_clear", &meta_clear);
  m.def("meta_forward_pop", &meta_forward_pop);
  m.def("meta_eventfd", &meta_eventfd);
  m.def("meta_send", &meta_send);
  m.def("meta_stats", &meta_stats);
  m.def("meta_worker_upsert", &meta_worker_upsert);
  m.def("meta_block_add_loc", &meta_block_add_loc);
  m.def("meta_block_remove_loc", &meta_block_remove_loc);
  m.def("meta_block_drop", &meta_block_drop);
  m.def("meta_take_access", &meta_take_access);
  m.def("data_create", &data_create);
  m.def("data_stop", &data_stop_srv);
  m.def("data_block_publish", &data_block_publish, py::arg("sid"),
        py::arg("block_id"), py::arg("kind"), py::arg("arena"),
        py::arg("aoff"), py::arg("len"), py::arg("path"),
        py::arg("direct") = false);
  m.def("data_block_drop", &data_block_drop);
  m.def("data_block_refs", &data_block_refs);
  m.def("data_write_register", &data_write_register);
  m.def("data_write_unregister", &data_write_unregister);
  m.def("data_forward_pop", &data_forward_pop);
  m.def("data_eventfd", &data_eventfd);
  m.def("data_send", &data_send);
  m.def("data_stats", &data_stats);
  m.def("data_read_into", &data_read_into);
  m.def("data_write_from", &data_write_from);
  m.def("dw_open", &dw_open);
  m.def("dw_write", &dw_write);
  m.def("dw_drain", &dw_drain);
  m.def("dw_commit", &dw_commit);
  m.def("dw_abort", &dw_abort);
  m.def("lz4_compress", &lz4_compress_py);
  m.def("lz4_decompress", &lz4_decompress_py);
  m.def("arena_lz4_decompress", &arena_lz4_decompress);
  m.def("crc32c", &crc32c_buf, py::arg("buf"), py::arg("init") = 0);
  m.def("crc32c_combine", &crc32c_combine);
  m.attr("CRC_SUB") = CRC_SUB;
  m.attr("__hip_arch__") = "gfx950";
}
