// Native FUSE data loop: GIL-free channel threads serving READ straight
// from cache arenas (HBM via per-thread-stream DMA, host via memcpy) into
// pinned reply buffers — one write(2) per reply.
//
// The MI355X analog of the reference's splice-based receiver/sender hot
// path (curvine-fuse/src/session/channel/fuse_receiver.rs:138-204,
// fuse_sender.rs): metadata ops are forwarded to Python (the control
// plane); registered read handles never touch the interpreter.
//
// Safety contract with Python (curvine_amd/fuse/native_loop.py):
//  * a handle is registered only while its store readers are held open
//    (block deletion/demotion defers on reader refcounts), and
//    unregistered (exclusive lock) before those readers close;
//  * in-flight reads hold the shared lock for the whole serve.

#include <errno.h>
#include <poll.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <mutex>
#include <shared_mutex>
#include <thread>
#include <unordered_map>
#include <vector>

struct NExtent { uint64_t file_off, len; int arena; uint64_t arena_off; };
struct NHandle { uint64_t length; std::vector<NExtent> ext; };

// registered write window: sequential WRITEs append into the current
// block's arena extent GIL-free; block boundaries forward to Python
struct WHandle {
  int arena;
  uint64_t arena_off, capacity, start_off;
  std::atomic<uint64_t> written{0};
};

struct FuseInHeader {
  uint32_t len, opcode;
  uint64_t unique, nodeid;
  uint32_t uid, gid, pid;
  uint16_t total_extlen, pad;
};
struct FuseReadIn {
  uint64_t fh, offset;
  uint32_t size, read_flags;
  uint64_t lock_owner;
  uint32_t flags, pad;
};
struct FuseOutHeader { uint32_t len; int32_t error; uint64_t unique; };
struct FuseWriteIn {
  uint64_t fh, offset;
  uint32_t size, write_flags;
  uint64_t lock_owner;
  uint32_t flags, pad;
};
struct FuseWriteOut { uint32_t size, pad; };
static const uint32_t FUSE_OP_READ = 15;
static const uint32_t FUSE_OP_WRITE = 16;

struct FuseLoop {
  std::vector<int> fds;
  size_t bufsize = (1u << 20) + (64u << 10);
  std::unordered_map<uint64_t, NHandle> handles;
  std::unordered_map<uint64_t, std::unique_ptr<WHandle>> whandles;
  std::shared_mutex hmu;
  std::vector<std::thread> threads;
  std::deque<std::pair<int, std::string>> fwd;   // (origin fd, raw request)
  std::mutex fwd_mu;
  std::condition_variable fwd_cv;
  std::atomic<bool> stopping{false};
  std::atomic<uint64_t> reply_errors{0};
  std::atomic<uint64_t> native_reads{0}, native_writes{0}, forwarded{0};
};

static std::vector<FuseLoop*> g_loops;
static std::mutex g_loops_mu;

static FuseLoop* get_loop(int id) {
  std::lock_guard<std::mutex> g(g_loops_mu);
  if (id < 0 || id >= (int)g_loops.size() || !g_loops[id])
    throw std::runtime_error("bad fuse loop id");
  return g_loops[id];
}

// serve one READ natively; returns false if the fh is not registered

// /dev/fuse replies are all-or-nothing; EINTR retries, ENOENT means the
// kernel aborted the request (not an error for us).  Anything else is
// counted — a silently dropped reply would wedge the application's read.
static bool fuse_reply_write(FuseLoop* L, int fd, const uint8_t* buf,
                             size_t len) {
  for (;;) {
    ssize_t w = write(fd, buf, len);
    if (w >= 0) return true;
    if (errno == EINTR) continue;
    if (errno == ENOENT) return true;   // request aborted by the kernel
    L->reply_errors.fetch_add(1, std::memory_order_relaxed);
    return false;
  }
}

static bool serve_read(FuseLoop* L, int fd, const FuseInHeader* h,
                       const FuseReadIn* r, uint8_t* reply) {
  std::shared_lock<std::shared_mutex> lk(L->hmu);
  auto it = L->handles.find(r->fh);
  if (it == L->handles.end()) return false;
  const NHandle& H = it->second;
  uint64_t off = r->offset;
  uint64_t n = r->size;
  if (off >= H.length) n = 0;
  else if (off + n > H.length) n = H.length - off;
  uint8_t* payload = reply + sizeof(FuseOutHeader);
  uint64_t got = 0;
  // extents are sorted by file_off; binary search the first
  const auto& ext = H.ext;
  size_t lo = 0, hi = ext.size();
  while (lo < hi) {
    size_t mid = (lo + hi) / 2;
    if (ext[mid].file_off + ext[mid].len <= off) lo = mid + 1;
    else hi = mid;
  }
  for (size_t i = lo; i < ext.size() && got < n; ++i) {
    const NExtent& e = ext[i];
    if (e.file_off > off + got) break;   // hole (shouldn't happen)
    uint64_t eoff = off + got - e.file_off;
    uint64_t take = std::min(n - got, e.len - eoff);
    Arena* a = get_arena(e.arena);
    const uint8_t* src = (const uint8_t*)a->base + e.arena_off + eoff;
    if (a->is_dev()) {
      hipSetDevice(a->device);
      hipStream_t s = thread_stream(a->device);
      if (hipMemcpyAsync(payload + got, src, take, hipMemcpyDeviceToHost,
                         s) != hipSuccess ||
          hipStreamSynchronize(s) != hipSuccess)
        return false;   // let Python report the error
    } else {
      std::memcpy(payload + got, src, take);
    }
    got += take;
  }
  if (got < n) return false;   // uncovered range: fall back to Python
  FuseOutHeader* oh = (FuseOutHeader*)reply;
  oh->len = (uint32_t)(sizeof(FuseOutHeader) + got);
  oh->error = 0;
  oh->unique = h->unique;
  if (!fuse_reply_write(L, fd, reply, oh->len)) return true;  // counted
  L->native_reads.fetch_add(1, std::memory_order_relaxed);
  return true;
}

// serve one sequential WRITE natively; false -> forward to Python
static bool serve_write(FuseLoop* L, int fd, const FuseInHeader* h,
                        const FuseWriteIn* w, const uint8_t* payload,
                        size_t payload_avail) {
  if (w->size > payload_avail) return false;
  std::shared_lock<std::shared_mutex> lk(L->hmu);
  auto it = L->whandles.find(w->fh);
  if (it == L->whandles.end()) return false;
  WHandle* W = it->second.get();
  uint64_t cur = W->written.load(std::memory_order_acquire);
  if (w->offset != W->start_off + cur) return false;       // non-sequential
  if (cur + w->size > W->capacity) return false;           // block boundary
  if (!W->written.compare_exchange_strong(cur, cur + w->size))
    return false;   // concurrent writer (dup fd): let Python arbitrate
  Arena* a = get_arena(W->arena);
  uint8_t* dst = (uint8_t*)a->base + W->arena_off + cur;
  bool ok = true;
  if (a->is_dev()) {
    hipSetDevice(a->device);
    hipStream_t s = thread_stream(a->device);
    ok = hipMemcpyAsync(dst, payload, w->size, hipMemcpyHostToDevice, s) ==
             hipSuccess &&
         hipStreamSynchronize(s) == hipSuccess;
  } else {
    std::memcpy(dst, payload, w->size);
  }
  if (!ok) {   // roll the reservation back and let Python handle it
    W->written.fetch_sub(w->size);
    return false;
  }
  struct { FuseOutHeader oh; FuseWriteOut wo; } rep;
  rep.oh.len = sizeof(rep);
  rep.oh.error = 0;
  rep.oh.unique = h->unique;
  rep.wo.size = w->size;
  rep.wo.pad = 0;
  if (!fuse_reply_write(L, fd, (const uint8_t*)&rep, sizeof(rep)))
    return true;   // bytes are in the arena; the reply loss is counted
  L->native_writes.fetch_add(1, std::memory_order_relaxed);
  return true;
}

static void loop_thread(FuseLoop* L, int fd) {
  const bool trace = getenv("CV_FUSE_NATIVE_TRACE") != nullptr;
  std::vector<uint8_t> req(L->bufsize);
  uint8_t* reply = nullptr;
  bool reply_pinned = false;
  if (device_count() > 0 &&
      hipHostMalloc((void**)&reply, L->bufsize, hipHostMallocDefault) ==
          hipSuccess) {
    reply_pinned = true;
  } else {
    reply = (uint8_t*)std::malloc(L->bufsize);
  }
  while (!L->stopping.load(std::memory_order_relaxed)) {
    // poll first: a blocking read would never wake on close(fd) at stop
    struct pollfd pfd{fd, POLLIN, 0};
    int pr = poll(&pfd, 1, 500);
    if (pr < 0 && errno != EINTR) break;
    if (pr <= 0) continue;
    ssize_t n = read(fd, req.data(), req.size());
    if (n < 0) {
      if (errno == EINTR || errno == ENOENT) continue;
      if (!L->stopping.load())
        fprintf(stderr, "[fuse_loop] fd %d read error: %s\n", fd,
                strerror(errno));
      break;   // ENODEV (unmount) / EBADF (stop)
    }
    if (n < (ssize_t)sizeof(FuseInHeader)) continue;
    const FuseInHeader* h = (const FuseInHeader*)req.data();
    if (trace)
      fprintf(stderr, "[nloop fd%d] n=%zd op=%u unique=%llu len=%u\n", fd, n,
              h->opcode, (unsigned long long)h->unique, h->len);
    if (h->opcode == FUSE_OP_READ &&
        n >= (ssize_t)(sizeof(FuseInHeader) + sizeof(FuseReadIn))) {
      const FuseReadIn* r =
          (const FuseReadIn*)(req.data() + sizeof(FuseInHeader));
      if (r->size + sizeof(FuseOutHeader) <= L->bufsize &&
          serve_read(L, fd, h, r, reply))
        continue;
    }
    if (h->opcode == FUSE_OP_WRITE &&
        n >= (ssize_t)(sizeof(FuseInHeader) + sizeof(FuseWriteIn))) {
      const FuseWriteIn* w =
          (const FuseWriteIn*)(req.data() + sizeof(FuseInHeader));
      const uint8_t* payload =
          req.data() + sizeof(FuseInHeader) + sizeof(FuseWriteIn);
      size_t avail = n - sizeof(FuseInHeader) - sizeof(FuseWriteIn);
      struct timespec t0, t1;
      if (trace) clock_gettime(CLOCK_MONOTONIC, &t0);
      bool served = serve_write(L, fd, h, w, payload, avail);
      if (trace) {
        clock_gettime(CLOCK_MONOTONIC, &t1);
        fprintf(stderr, "[nloop fd%d] WRITE off=%llu size=%u served=%d %.2fms\n",
                fd, (unsigned long long)w->offset, w->size, (int)served,
                (t1.tv_sec - t0.tv_sec) * 1e3 + (t1.tv_nsec - t0.tv_nsec) / 1e6);
      }
      if (served) continue;
    }
    // anything else: forward the raw request to the Python control plane.
    // The origin fd travels with it: FUSE replies MUST be written to the
    // same device fd that read the request (each clone has its own
    // processing list; replying elsewhere is silently dropped/ENOENT).
    {
      std::lock_guard<std::mutex> g(L->fwd_mu);
      L->fwd.emplace_back(fd, std::string((const char*)req.data(), (size_t)n));
    }
    L->forwarded.fetch_add(1, std::memory_order_relaxed);
    L->fwd_cv.notify_one();
  }
  if (reply_pinned) hipHostFree(reply);
  else std::free(reply);
}

// ---------------------------------------------------------------------------
// bindings
// ---------------------------------------------------------------------------

static int fuse_loop_create(size_t max_write) {
  auto* L = new FuseLoop();
  L->bufsize = max_write + (64u << 10);
  std::lock_guard<std::mutex> g(g_loops_mu);
  g_loops.push_back(L);
  return (int)g_loops.size() - 1;
}

static void fuse_loop_add_channel(int id, int fd) {
  FuseLoop* L = get_loop(id);
  L->fds.push_back(fd);
  L->threads.emplace_back(loop_thread, L, fd);
}

static void fuse_loop_register(int id, uint64_t fh, uint64_t length,
                               std::vector<std::tuple<uint64_t, uint64_t, int,
                                                      uint64_t>> extents) {
  FuseLoop* L = get_loop(id);
  NHandle H;
  H.length = length;
  for (auto& t : extents)
    H.ext.push_back({std::get<0>(t), std::get<1>(t), std::get<2>(t),
                     std::get<3>(t)});
  std::unique_lock<std::shared_mutex> lk(L->hmu);
  L->handles[fh] = std::move(H);
}

static void fuse_loop_register_write(int id, uint64_t fh, int arena,
                                     uint64_t arena_off, uint64_t capacity,
                                     uint64_t start_off) {
  FuseLoop* L = get_loop(id);
  auto W = std::make_unique<WHandle>();
  W->arena = arena;
  W->arena_off = arena_off;
  W->capacity = capacity;
  W->start_off = start_off;
  std::unique_lock<std::shared_mutex> lk(L->hmu);
  L->whandles[fh] = std::move(W);
}

// remove the write window; returns bytes natively written into it
static uint64_t fuse_loop_unregister_write(int id, uint64_t fh) {
  FuseLoop* L = get_loop(id);
  py::gil_scoped_release rel;
  std::unique_lock<std::shared_mutex> lk(L->hmu);
  auto it = L->whandles.find(fh);
  if (it == L->whandles.end()) return 0;
  uint64_t w = it->second->written.load();
  L->whandles.erase(it);
  return w;
}

static uint64_t fuse_loop_write_state(int id, uint64_t fh) {
  FuseLoop* L = get_loop(id);
  std::shared_lock<std::shared_mutex> lk(L->hmu);
  auto it = L->whandles.find(fh);
  return it == L->whandles.end() ? 0 : it->second->written.load();
}

static void fuse_loop_unregister(int id, uint64_t fh) {
  FuseLoop* L = get_loop(id);
  py::gil_scoped_release rel;   // may wait for in-flight reads
  std::unique_lock<std::shared_mutex> lk(L->hmu);
  L->handles.erase(fh);
}

static py::tuple fuse_loop_next_forward(int id, double timeout_s) {
  FuseLoop* L = get_loop(id);
  std::string msg;
  int origin_fd = -1;
  {
    py::gil_scoped_release rel;
    std::unique_lock<std::mutex> lk(L->fwd_mu);
    L->fwd_cv.wait_for(lk, std::chrono::duration<double>(timeout_s),
                       [&] { return !L->fwd.empty() || L->stopping.load(); });
    if (!L->fwd.empty()) {
      origin_fd = L->fwd.front().first;
      msg = std::move(L->fwd.front().second);
      L->fwd.pop_front();
    }
  }   // GIL re-acquired here
  return py::make_tuple(origin_fd, py::bytes(msg));
}

static py::dict fuse_loop_stats(int id) {
  FuseLoop* L = get_loop(id);
  py::dict d;
  d["native_reads"] = L->native_reads.load();
  d["native_writes"] = L->native_writes.load();
  d["forwarded"] = L->forwarded.load();
  d["reply_errors"] = L->reply_errors.load();
  {
    std::shared_lock<std::shared_mutex> lk(L->hmu);
    d["registered_handles"] = L->handles.size();
  }
  return d;
}

static void fuse_loop_stop(int id) {
  FuseLoop* L = get_loop(id);
  L->stopping.store(true);
  py::gil_scoped_release rel;
  for (int fd : L->fds) close(fd);
  L->fwd_cv.notify_all();
  for (auto& t : L->threads)
    if (t.joinable()) t.join();
  L->threads.clear();
}
