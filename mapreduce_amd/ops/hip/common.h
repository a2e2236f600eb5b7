// Common device helpers for the MapReduce CDNA4 kernels (gfx950 only).
// Wavefront = 64 lanes; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef uint64_t u64;
typedef uint32_t u32;
typedef uint8_t u8;
typedef int64_t i64;

// Sentinel for empty hash-table slots.  A real FNV-1a hash could in
// principle equal this; inserts remap it (see hash_table.hip) so the table
// stays correct for every input.
#define HT_EMPTY 0xFFFFFFFFFFFFFFFFull

DEV u64 mulhi_u64(u64 a, u64 b) { return __umul64hi(a, b); }

// partition id of a 64-bit hash for P partitions: floor(h * P / 2^64).
// Monotonic in h, so a hash-sorted array is partition-contiguous — the
// property that lets the RCCL all-to-all send buffers be plain slices.
DEV u32 partition_of(u64 h, u32 nparts) { return (u32)mulhi_u64(h, (u64)nparts); }

// FNV-1a 64 (must match mapreduce_amd.utils.tuple.fnv1a64)
#define FNV64_OFFSET 0xCBF29CE484222325ull
#define FNV64_PRIME 0x100000001B3ull

// Chunked word hash (must match utils.tuple.wordhash64): one xor-multiply
// per 8 zero-padded little-endian bytes, then a length fold.  6x shorter
// dependent chain than byte-serial FNV-1a; bijective (hence collision-free)
// over single-chunk words.
DEV u64 whash_chunk(u64 h, u64 chunk) { return (h ^ chunk) * FNV64_PRIME; }
DEV u64 whash_fin(u64 h, u64 len) { return (h ^ len) * FNV64_PRIME; }

// SplitMix64 finalizer (must match utils.tuple.splitmix64)
DEV u64 splitmix64_dev(u64 x) {
  u64 z = x + 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

// index of last element <= v in a sorted i64 array (upper_bound - 1)
DEV int ub_minus1(const i64* a, int n, i64 v) {
  int lo = 0, hi = n;  // first index with a[i] > v
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (a[mid] <= v) lo = mid + 1;
    else hi = mid;
  }
  return lo - 1;
}

DEV bool is_ws(u8 c) {
  // Python str.split() whitespace set: \t \n \v \f \r ' '
  return c == ' ' || (c >= 9 && c <= 13);
}

// SWAR is_ws over 8 bytes at once: returns an 8-bit mask, bit i set iff
// byte i of x is whitespace (must match is_ws for all 256 byte values —
// tests tokenize random binary against the Python split oracle).
// ~17 u64 ops replace 8 per-byte classifications (~32 ops).
DEV u32 ws_mask8(u64 x) {
  const u64 L = 0x0101010101010101ULL;
  const u64 H = 0x8080808080808080ULL;
  u64 v = x ^ 0x2020202020202020ULL;          // 0x00 where byte == ' '
  // exact zero-byte detector: the (v - L) & ~v & H shortcut false-fires
  // when the next-lower byte borrows (e.g. "0x21 0x20"); this form has
  // no cross-byte carries (0x7F + 0x7F < 0x100)
  u64 m20 = ~(((v & ~H) + ~H) | v | ~H);      // msb set where byte == ' '
  u64 hi = x & H;                             // bytes >= 128: never ws
  u64 low = x & ~H;
  u64 ge9 = (low + (0x7F - 8) * L) & H;       // low7 >= 9
  u64 ge14 = (low + (0x7F - 13) * L) & H;     // low7 >= 14
  u64 ws = m20 | (ge9 & ~ge14 & ~hi);         // msb-per-byte ws flags
  // pack the 8 msb flags into bits 0..7: bits sit at 8i+7; after >>7
  // they sit at 8i, and OR-folding by {7,14,28} supplies every needed
  // shift 7i (subset sums) while no cross-byte bit can alias into the
  // low byte (8i - 7k lands in [0,7] only for k == i)
  u64 m = ws >> 7;
  m |= m >> 7;
  m |= m >> 14;
  m |= m >> 28;
  return (u32)(m & 0xFF);
}

DEV u64 lane_id() { return __lane_id(); }

// inclusive wave sum over 64 lanes
DEV i64 wave_sum_i64(i64 v) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)
