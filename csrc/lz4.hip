// LZ4 block codec: host compress/decompress + gfx950 device decompress.
//
// Covers the reference's lz4 sites (raft snapshot streams,
// RocksDB block compression; new-engine kernel "lz4_decompress_stream").
// Container "CVLZ" = raw_size u64 | chunk_size u32 | n_chunks u32 |
// comp_size[n] u32 | chunk payloads.  Chunks compress independently so the
// GPU decompresses them in parallel (one workgroup per chunk; the
// sequential nature of LZ4 keeps per-chunk work serial — parallelism comes
// from chunk count, which saturates the chip for multi-MB streams).
//
// Block format per LZ4 spec: token(4b lit|4b match), ext lens (255...),
// literals, 2B little-endian offset, match >= 4.

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <string.h>
#include <vector>

static const uint32_t LZ4_MAGIC = 0x435A4C34u;  // "4LZC"
static const int LZ4_CHUNK = 64 * 1024;

// ---------------------------------------------------------------------------
// host compress (greedy hash table, LZ4-block compatible)
// ---------------------------------------------------------------------------

static inline uint32_t lz4_hash(uint32_t v) {
  return (v * 2654435761u) >> 19;   // 13-bit table
}

static size_t lz4_compress_block(const uint8_t* src, size_t n, uint8_t* dst) {
  uint32_t table[1 << 13];
  memset(table, 0xFF, sizeof(table));
  size_t si = 0, di = 0, anchor = 0;
  if (n >= 13) {
    while (si + 12 < n) {
      uint32_t seq;
      memcpy(&seq, src + si, 4);
      uint32_t h = lz4_hash(seq);
      uint32_t cand = table[h];
      table[h] = (uint32_t)si;
      uint32_t ref_seq = 0;
      if (cand != 0xFFFFFFFFu && si - cand <= 65535) {
        memcpy(&ref_seq, src + cand, 4);
      }
      if (cand == 0xFFFFFFFFu || si - cand > 65535 || ref_seq != seq) {
        si++;
        continue;
      }
      // extend match (leave 5 bytes for the closing literals-only sequence)
      size_t mlen = 4;
      while (si + mlen < n - 5 && src[si + mlen] == src[cand + mlen]) mlen++;
      size_t lit = si - anchor;
      // token
      uint8_t* tok = dst + di++;
      *tok = 0;
      if (lit >= 15) {
        *tok |= 0xF0;
        size_t rest = lit - 15;
        while (rest >= 255) { dst[di++] = 255; rest -= 255; }
        dst[di++] = (uint8_t)rest;
      } else {
        *tok |= (uint8_t)(lit << 4);
      }
      memcpy(dst + di, src + anchor, lit);
      di += lit;
      uint16_t off = (uint16_t)(si - cand);
      dst[di++] = off & 0xFF;
      dst[di++] = off >> 8;
      size_t mrest = mlen - 4;
      if (mrest >= 15) {
        *tok |= 0x0F;
        mrest -= 15;
        while (mrest >= 255) { dst[di++] = 255; mrest -= 255; }
        dst[di++] = (uint8_t)mrest;
      } else {
        *tok |= (uint8_t)mrest;
      }
      si += mlen;
      anchor = si;
    }
  }
  // final literals
  size_t lit = n - anchor;
  uint8_t* tok = dst + di++;
  *tok = 0;
  if (lit >= 15) {
    *tok |= 0xF0;
    size_t rest = lit - 15;
    while (rest >= 255) { dst[di++] = 255; rest -= 255; }
    dst[di++] = (uint8_t)rest;
  } else {
    *tok |= (uint8_t)(lit << 4);
  }
  memcpy(dst + di, src + anchor, lit);
  di += lit;
  return di;
}

static size_t lz4_decompress_block_host(const uint8_t* src, size_t comp_n,
                                        uint8_t* dst, size_t cap) {
  size_t si = 0, di = 0;
  while (si < comp_n) {
    uint8_t tok = src[si++];
    size_t lit = tok >> 4;
    if (lit == 15) {
      uint8_t b;
      do { b = src[si++]; lit += b; } while (b == 255);
    }
    if (di + lit > cap || si + lit > comp_n) return (size_t)-1;
    memcpy(dst + di, src + si, lit);
    di += lit; si += lit;
    if (si >= comp_n) break;   // last sequence has no match
    uint16_t off = src[si] | (src[si + 1] << 8);
    si += 2;
    size_t mlen = (tok & 0xF);
    if (mlen == 15) {
      uint8_t b;
      do { b = src[si++]; mlen += b; } while (b == 255);
    }
    mlen += 4;
    if (off == 0 || off > di || di + mlen > cap) return (size_t)-1;
    // overlapping copy byte-by-byte
    for (size_t i = 0; i < mlen; ++i) { dst[di] = dst[di - off]; di++; }
  }
  return di;
}

// ---------------------------------------------------------------------------
// device decompress: one workgroup per chunk; lane 0 walks the sequences,
// the full wave does the literal copies (match copies are wave-wide when
// offset allows, serial otherwise).
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(64) void lz4_decompress_kernel(
    const uint8_t* __restrict__ comp, const uint32_t* __restrict__ chunk_off,
    const uint32_t* __restrict__ chunk_len, uint8_t* __restrict__ out,
    uint32_t chunk_size, uint64_t raw_size, uint32_t n_chunks,
    int* __restrict__ error_flag) {
  for (uint32_t c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const uint8_t* src = comp + chunk_off[c];
    uint32_t comp_n = chunk_len[c];
    uint8_t* dst = out + (uint64_t)c * chunk_size;
    uint64_t cap = min((uint64_t)chunk_size,
                       raw_size - (uint64_t)c * chunk_size);
    uint32_t si = 0, di = 0;
    int lane = threadIdx.x;
    while (si < comp_n) {
      // every lane parses the same header (uniform addresses broadcast)
      uint8_t tok = src[si++];
      uint32_t lit = tok >> 4;
      if (lit == 15) {
        uint8_t b;
        do { b = src[si++]; lit += b; } while (b == 255);
      }
      if (di + lit > cap || si + lit > comp_n) {
        if (lane == 0) atomicExch(error_flag, 1 + (int)c);
        return;
      }
      for (uint32_t i = lane; i < lit; i += 64) dst[di + i] = src[si + i];
      __threadfence_block();   // cross-lane visibility of the writes
      di += lit; si += lit;
      if (si >= comp_n) break;
      uint32_t off = src[si] | (src[si + 1] << 8);
      si += 2;
      uint32_t mlen = (tok & 0xF);
      if (mlen == 15) {
        uint8_t b;
        do { b = src[si++]; mlen += b; } while (b == 255);
      }
      mlen += 4;
      if (off == 0 || off > di || di + mlen > cap) {
        if (lane == 0) atomicExch(error_flag, 1 + (int)c);
        return;
      }
      // match bytes are periodic with period `off`: every lane reads only
      // PRE-MATCH data (dst[di-off .. di)), so there is no intra-match
      // hazard for any offset
      for (uint32_t i = lane; i < mlen; i += 64)
        dst[di + i] = dst[di - off + (i % off)];
      __threadfence_block();
      di += mlen;
    }
  }
}
