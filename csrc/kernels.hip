// CDNA4 (gfx950) data-pipeline kernels for curvine_amd.
//
// These re-implement, GPU-side, the byte-crunching sites of the reference
// cache engine: CRC32C block checksums
// (curvine-tests/src/curvine_bench.rs:37-40 verification analog),
// scatter/gather chunk coalesce (DataSlice split/merge,
// fs_reader_parallel.rs:94-126 analog), and zero-fill for sparse hole
// reads (block_reader_hole.rs analog).
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * memory-bound streaming kernels: 16 B/lane vectorized accesses,
//    grid-stride loops, grids sized >> 256 workgroups to fill 8 XCDs.
//  * wavefront = 64; block dims are multiples of 64.
//  * CRC32C: per-thread slice-by-8 over an LDS-resident table, partial
//    CRCs combined with GF(2) 32x32 matrices (one matrix per tree level,
//    128 B each) — the standard crc-combine trick, evaluated in-kernel.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WG 256            // threads per workgroup
#define CRC_SUB 4096      // bytes per thread for crc kernel

// ---------------------------------------------------------------------------
// CRC32C (Castagnoli 0x1EDC6F41, reflected 0x82F63B78)
// ---------------------------------------------------------------------------

// slice-by-8 tables, filled by host via hipMemcpyToSymbol
__device__ uint32_t g_crc_tab[8][256];
// combine matrices: level k combines two equal sub-crcs of length CRC_SUB<<k
// m[k][i] = column i of the GF2 matrix for "shift crc by (CRC_SUB<<k) zero bytes"
__device__ uint32_t g_comb_mat[24][32];

__device__ __forceinline__ uint32_t gf2_apply(const uint32_t* mat, uint32_t crc) {
  uint32_t out = 0;
#pragma unroll
  for (int i = 0; i < 32; ++i) {
    // bit i of crc selects column i
    out ^= (crc >> i & 1u) ? mat[i] : 0u;
  }
  return out;
}

// Each thread computes the CRC of its CRC_SUB-byte sub-chunk, the workgroup
// tree-combines to one CRC per WG*CRC_SUB "super-chunk", written to out[].
// Host combines the per-workgroup results (equal-length, one matrix apply
// each) and the byte tail. `n_sub` = number of full sub-chunks.
extern "C" __global__ __launch_bounds__(WG) void crc32c_kernel(
    const uint8_t* __restrict__ data, uint64_t n_sub, uint32_t* __restrict__ out) {
  __shared__ uint32_t tab[8][256];
  __shared__ uint32_t partial[WG];
  for (int i = threadIdx.x; i < 8 * 256; i += WG)
    (&tab[0][0])[i] = (&g_crc_tab[0][0])[i];
  __syncthreads();

  uint64_t sub = (uint64_t)blockIdx.x * WG + threadIdx.x;
  uint32_t crc = 0xFFFFFFFFu;
  if (sub < n_sub) {
    const uint8_t* p = data + sub * CRC_SUB;
    // 8 bytes per step, slice-by-8; data is 8B-aligned by construction
    const uint64_t* q = reinterpret_cast<const uint64_t*>(p);
#pragma unroll 4
    for (int i = 0; i < CRC_SUB / 8; ++i) {
      uint64_t v = q[i];
      uint32_t lo = (uint32_t)v ^ crc;
      uint32_t hi = (uint32_t)(v >> 32);
      crc = tab[7][lo & 0xFF] ^ tab[6][(lo >> 8) & 0xFF] ^
            tab[5][(lo >> 16) & 0xFF] ^ tab[4][lo >> 24] ^
            tab[3][hi & 0xFF] ^ tab[2][(hi >> 8) & 0xFF] ^
            tab[1][(hi >> 16) & 0xFF] ^ tab[0][hi >> 24];
    }
  }
  crc ^= 0xFFFFFFFFu;
  partial[threadIdx.x] = crc;
  __syncthreads();

  // tree combine: crc(A||B) = gf2_apply(M_lenB, crc(A)) ^ crc(B).
  // Lanes beyond n_sub hold crc of empty string (0), and M*0^0=... careful:
  // combining with an EMPTY right side must be identity, so we track how
  // many real sub-chunks each accumulated node covers via count arithmetic.
  uint64_t base = (uint64_t)blockIdx.x * WG;
  for (int k = 0, stride = 1; stride < WG; ++k, stride <<= 1) {
    int t = threadIdx.x;
    uint32_t merged = 0; bool act = false;
    if ((t & (2 * stride - 1)) == 0) {
      uint64_t left_cnt = base + t + stride;     // first sub index of right node
      if (left_cnt < n_sub) {
        // right node non-empty: its covered length is min(stride, n_sub-left)
        uint64_t right_n = n_sub - left_cnt;
        if (right_n >= (uint64_t)stride) {
          merged = gf2_apply(g_comb_mat[k], partial[t]) ^ partial[t + stride];
        } else {
          // ragged tail: shift left crc by right_n sub-chunks (apply level
          // matrices per set bit of right_n)
          uint32_t c = partial[t];
          uint64_t r = right_n;
          for (int b = 0; b < 24 && (r >> b); ++b)
            if ((r >> b) & 1) c = gf2_apply(g_comb_mat[b], c);
          merged = c ^ partial[t + stride];
        }
        act = true;
      }
    }
    __syncthreads();
    if (act) partial[threadIdx.x] = merged;
    __syncthreads();
  }
  if (threadIdx.x == 0) out[blockIdx.x] = partial[0];
}

// ---------------------------------------------------------------------------
// Extent copy (gather/scatter coalesce) and fill
// ---------------------------------------------------------------------------

struct Extent { uint64_t src_off, dst_off, len; };

// One workgroup tile = 64 KiB of one extent. Host builds a flat tile table
// (extent index per tile) so the grid is perfectly balanced across CUs.
#define TILE (64 * 1024)

extern "C" __global__ __launch_bounds__(WG) void copy_extents_kernel(
    const uint8_t* __restrict__ src, uint8_t* __restrict__ dst,
    const Extent* __restrict__ extents,
    const uint32_t* __restrict__ tile_ext,   // tile -> extent index
    const uint64_t* __restrict__ tile_off,   // tile -> offset within extent
    uint32_t n_tiles) {
  for (uint32_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    Extent e = extents[tile_ext[tile]];
    uint64_t toff = tile_off[tile];
    uint64_t len = min((uint64_t)TILE, e.len - toff);
    const uint8_t* s = src + e.src_off + toff;
    uint8_t* d = dst + e.dst_off + toff;
    uint64_t mis = ((uintptr_t)s | (uintptr_t)d) & 15;
    if (mis == 0 && (len & 15) == 0) {
      const uint4* s4 = reinterpret_cast<const uint4*>(s);
      uint4* d4 = reinterpret_cast<uint4*>(d);
      for (uint64_t i = threadIdx.x; i < len / 16; i += WG) d4[i] = s4[i];
    } else if ((((uintptr_t)s | (uintptr_t)d | len) & 3) == 0) {
      const uint32_t* s1 = reinterpret_cast<const uint32_t*>(s);
      uint32_t* d1 = reinterpret_cast<uint32_t*>(d);
      for (uint64_t i = threadIdx.x; i < len / 4; i += WG) d1[i] = s1[i];
    } else {
      for (uint64_t i = threadIdx.x; i < len; i += WG) d[i] = s[i];
    }
  }
}

extern "C" __global__ __launch_bounds__(WG) void fill_kernel(
    uint8_t* __restrict__ dst, uint8_t value, uint64_t n) {
  uint64_t n16 = n / 16;
  uint4 v4;
  uint32_t vv = value * 0x01010101u;
  v4.x = v4.y = v4.z = v4.w = vv;
  uint4* d4 = reinterpret_cast<uint4*>(dst);
  uint64_t stride = (uint64_t)gridDim.x * WG;
  for (uint64_t i = (uint64_t)blockIdx.x * WG + threadIdx.x; i < n16; i += stride)
    d4[i] = v4;
  // tail
  for (uint64_t i = n16 * 16 + (uint64_t)blockIdx.x * WG + threadIdx.x; i < n;
       i += stride)
    dst[i] = value;
}

// byte-compare two device ranges -> mismatch count (numerics tests)
extern "C" __global__ __launch_bounds__(WG) void compare_kernel(
    const uint8_t* __restrict__ a, const uint8_t* __restrict__ b, uint64_t n,
    unsigned long long* mismatches) {
  uint64_t stride = (uint64_t)gridDim.x * WG;
  unsigned long long local = 0;
  uint64_t n16 = n / 16;
  const uint4* a4 = reinterpret_cast<const uint4*>(a);
  const uint4* b4 = reinterpret_cast<const uint4*>(b);
  for (uint64_t i = (uint64_t)blockIdx.x * WG + threadIdx.x; i < n16; i += stride) {
    uint4 x = a4[i], y = b4[i];
    if (x.x != y.x || x.y != y.y || x.z != y.z || x.w != y.w) local++;
  }
  for (uint64_t i = n16 * 16 + (uint64_t)blockIdx.x * WG + threadIdx.x; i < n;
       i += stride)
    if (a[i] != b[i]) local++;
  if (local) atomicAdd(mismatches, local);
}
