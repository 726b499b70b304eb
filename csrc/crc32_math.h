// GF(2) linear algebra for parallel CRC32 (zlib-compatible, poly 0xEDB88320).
//
// The engine CRCs cross-silo tensor payloads on the GPU: each device thread
// computes the *linear part* L(m_t) of its slice (init 0, no final xor),
// applies the shift matrix for its suffix length, and the results xor-reduce
// to the linear part of the whole message.  Host-side we prepare:
//   - 8 slice-by-8 byte tables (kernel inner loop),
//   - binary-power shift matrices M_sub^(2^j) (per-thread suffix apply),
//   - the tail shift matrix and the init/final constant
//     crc(M) = L(M) ^ shift_{len}(0xFFFFFFFF) ^ 0xFFFFFFFF.
//
// This header is pure host C++ (also unit-tested against zlib via Python).
#pragma once

#include <cstddef>
#include <cstdint>
#include <cstring>
#include <vector>

namespace rayfed_crc {

constexpr uint32_t kPoly = 0xEDB88320u;  // reflected CRC-32 (IEEE / zlib)

// ---- GF(2) 32x32 matrices: mat[i] = column for input bit i -----------------
inline uint32_t gf2_times(const uint32_t* mat, uint32_t vec) {
  uint32_t sum = 0;
  int i = 0;
  while (vec) {
    if (vec & 1u) sum ^= mat[i];
    vec >>= 1;
    ++i;
  }
  return sum;
}

inline void gf2_square(uint32_t* dst, const uint32_t* mat) {
  for (int i = 0; i < 32; ++i) dst[i] = gf2_times(mat, mat[i]);
}

// Matrix that advances the CRC register past one zero BYTE.
inline void byte_shift_matrix(uint32_t* m8) {
  uint32_t odd[32], even[32];
  // one zero BIT:
  odd[0] = kPoly;
  for (int i = 1; i < 32; ++i) odd[i] = 1u << (i - 1);
  gf2_square(even, odd);   // 2 bits
  gf2_square(odd, even);   // 4 bits
  gf2_square(m8, odd);     // 8 bits = 1 byte
}

// Matrix that advances past `len` zero bytes (len >= 0; len==0 -> identity).
inline void shift_matrix(uint32_t* out, uint64_t len) {
  uint32_t acc[32];
  for (int i = 0; i < 32; ++i) acc[i] = 1u << i;  // identity
  uint32_t p[32];
  byte_shift_matrix(p);
  while (len) {
    if (len & 1u) {
      uint32_t tmp[32];
      for (int i = 0; i < 32; ++i) tmp[i] = gf2_times(p, acc[i]);
      std::memcpy(acc, tmp, sizeof(acc));
    }
    len >>= 1;
    if (!len) break;
    uint32_t sq[32];
    gf2_square(sq, p);
    std::memcpy(p, sq, sizeof(p));
  }
  std::memcpy(out, acc, 32 * sizeof(uint32_t));
}

inline uint32_t shift_apply(uint64_t len, uint32_t v) {
  uint32_t m[32];
  shift_matrix(m, len);
  return gf2_times(m, v);
}

// zlib-style combine of two finalized CRCs: crc(A||B) given crcA, crcB, |B|.
inline uint32_t crc32_combine(uint32_t crc1, uint32_t crc2, uint64_t len2) {
  // Finalized crc = raw ^ 0xFFFFFFFF with raw starting from 0xFFFFFFFF.
  // crc(A||B) = shift_{len2}(crcA) ^ crcB  — the init/final terms cancel
  // exactly as in zlib's crc32_combine.
  return shift_apply(len2, crc1) ^ crc2;
}

// ---- slice-by-8 tables (init-0 linear CRC) ---------------------------------
// tables[k][b] = linear CRC of byte b followed by k zero bytes.
inline void make_slice8_tables(uint32_t* tables /* [8*256] */) {
  for (int b = 0; b < 256; ++b) {
    uint32_t c = static_cast<uint32_t>(b);
    for (int k = 0; k < 8; ++k) c = (c >> 1) ^ ((c & 1u) ? kPoly : 0u);
    tables[b] = c;
  }
  for (int k = 1; k < 8; ++k) {
    for (int b = 0; b < 256; ++b) {
      uint32_t c = tables[(k - 1) * 256 + b];
      tables[k * 256 + b] = (c >> 8) ^ tables[c & 0xFFu];
    }
  }
}

// ---- host reference (for tests / tiny buffers) -----------------------------
inline uint32_t crc32_host(const uint8_t* data, size_t n, uint32_t crc = 0) {
  static uint32_t table[256];
  static bool init = false;
  if (!init) {
    for (int b = 0; b < 256; ++b) {
      uint32_t c = static_cast<uint32_t>(b);
      for (int k = 0; k < 8; ++k) c = (c >> 1) ^ ((c & 1u) ? kPoly : 0u);
      table[b] = c;
    }
    init = true;
  }
  crc ^= 0xFFFFFFFFu;
  for (size_t i = 0; i < n; ++i) crc = (crc >> 8) ^ table[(crc ^ data[i]) & 0xFFu];
  return crc ^ 0xFFFFFFFFu;
}

}  // namespace rayfed_crc
