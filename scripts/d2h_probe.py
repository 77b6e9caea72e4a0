import sys, time, os, threading
sys.path.insert(0, "/root/repo")
from curvine_amd.native import Arena, PinnedBuffer, load
lib = load()
a = Arena(0, 20 << 30, 8 << 20, 8)

def bw(label, nthreads, chunk, total_per_thread):
    bufs = [PinnedBuffer(chunk) for _ in range(nthreads)]
    errs = []
    def w(t):
        try:
            off = t * total_per_thread
            pos = 0
            while pos < total_per_thread:
                a.read_to_ptr(off + pos, bufs[t].ptr, chunk, False)
                pos += chunk
        except Exception as e:
            errs.append(e)
    ts = [threading.Thread(target=w, args=(t,)) for t in range(nthreads)]
    t0 = time.perf_counter()
    [t.start() for t in ts]; [t.join() for t in ts]
    dt = time.perf_counter() - t0
    if errs: print(label, "ERR", errs[0]); return
    print(f"{label}: {nthreads*total_per_thread/dt/2**30:.1f} GiB/s")
    for b in bufs: b.close()

bw("1 thread x 256MiB chunks", 1, 256 << 20, 1 << 30)
bw("1 thread x 4MiB chunks", 1, 4 << 20, 1 << 30)
bw("4 threads x 4MiB", 4, 4 << 20, 1 << 30)
bw("16 threads x 4MiB", 16, 4 << 20, 1 << 30)
bw("16 threads x 16MiB", 16, 16 << 20, 1 << 30)
bw("32 threads x 4MiB", 32, 4 << 20, 512 << 20)
