/* MI355X-native device hashing primitives (PRODUCT code).
 *
 * SHA-256 (FIPS 180-4) and keyed BLAKE2b-256 (RFC 7693) for the sighash and
 * BIP-340 challenge paths. One message per lane, state in VGPRs, fixed-layout
 * specializations for the hot message shapes. Independent implementation from
 * oracle/ok_hash.c (register-resident states, compile-time specialization).
 */
#ifndef KV_HASH_DEVICE_H
#define KV_HASH_DEVICE_H

#ifndef KV_HOST_TEST
#include <hip/hip_runtime.h>
#endif
#include <stdint.h>

namespace kv {

/* ---------------- SHA-256 ---------------- */

__device__ __constant__ static const uint32_t SHA_K[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

__device__ __forceinline__ uint32_t rotr32(uint32_t x, int n) {
  return (x >> n) | (x << (32 - n));
}

__device__ inline void sha256_compress(uint32_t h[8], const uint32_t w_in[16]) {
  uint32_t w[64];
#pragma unroll
  for (int i = 0; i < 16; i++) w[i] = w_in[i];
#pragma unroll
  for (int i = 16; i < 64; i++) {
    uint32_t s0 = rotr32(w[i - 15], 7) ^ rotr32(w[i - 15], 18) ^ (w[i - 15] >> 3);
    uint32_t s1 = rotr32(w[i - 2], 17) ^ rotr32(w[i - 2], 19) ^ (w[i - 2] >> 10);
    w[i] = w[i - 16] + s0 + w[i - 7] + s1;
  }
  uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5], g = h[6],
           hh = h[7];
#pragma unroll
  for (int i = 0; i < 64; i++) {
    uint32_t S1 = rotr32(e, 6) ^ rotr32(e, 11) ^ rotr32(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = hh + S1 + ch + SHA_K[i] + w[i];
    uint32_t S0 = rotr32(a, 2) ^ rotr32(a, 13) ^ rotr32(a, 22);
    uint32_t mj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + mj;
    hh = g; g = f; f = e; e = d + t1; d = c; c = b; b = a; a = t1 + t2;
  }
  h[0] += a; h[1] += b; h[2] += c; h[3] += d;
  h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

__device__ __forceinline__ uint32_t be32(const uint8_t *p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) |
         (uint32_t)p[3];
}

/* SHA256(tag_hash ‖ tag_hash ‖ a32 ‖ b32 ‖ c32) — the BIP-340 tagged hash of
 * 96 bytes of data. tag midstate folded at compile time by passing the
 * premixed state. */
__device__ inline void sha256_tagged96(const uint32_t mid[8], const uint8_t a[32],
                                       const uint8_t b[32], const uint8_t c[32],
                                       uint8_t out[32]) {
  uint32_t h[8];
#pragma unroll
  for (int i = 0; i < 8; i++) h[i] = mid[i];
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 8; i++) w[i] = be32(a + 4 * i);
#pragma unroll
  for (int i = 0; i < 8; i++) w[8 + i] = be32(b + 4 * i);
  sha256_compress(h, w);
#pragma unroll
  for (int i = 0; i < 8; i++) w[i] = be32(c + 4 * i);
  w[8] = 0x80000000u;
#pragma unroll
  for (int i = 9; i < 15; i++) w[i] = 0;
  w[15] = (64 + 96) * 8; /* total length bits */
  sha256_compress(h, w);
#pragma unroll
  for (int i = 0; i < 8; i++) {
    out[4 * i] = (uint8_t)(h[i] >> 24);
    out[4 * i + 1] = (uint8_t)(h[i] >> 16);
    out[4 * i + 2] = (uint8_t)(h[i] >> 8);
    out[4 * i + 3] = (uint8_t)h[i];
  }
}

/* ---------------- keyed BLAKE2b-256 (RFC 7693) ---------------- */

__device__ __constant__ static const uint64_t B2B_IV[8] = {
    0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL, 0x3c6ef372fe94f82bULL,
    0xa54ff53a5f1d36f1ULL, 0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
    0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};

__device__ __constant__ static const uint8_t B2B_SIGMA[12][16] = {
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3},
    {11, 8, 12, 0, 5, 2, 15, 13, 10, 14, 3, 6, 7, 1, 9, 4},
    {7, 9, 3, 1, 13, 12, 11, 14, 2, 6, 5, 10, 4, 0, 15, 8},
    {9, 0, 5, 7, 2, 4, 10, 15, 14, 1, 11, 12, 6, 8, 3, 13},
    {2, 12, 6, 10, 0, 11, 8, 3, 4, 13, 7, 5, 15, 14, 1, 9},
    {12, 5, 1, 15, 14, 13, 4, 10, 0, 7, 6, 3, 9, 2, 8, 11},
    {13, 11, 7, 14, 12, 1, 3, 9, 5, 0, 15, 4, 8, 6, 2, 10},
    {6, 15, 14, 9, 11, 3, 0, 8, 12, 2, 13, 7, 1, 4, 10, 5},
    {10, 2, 8, 4, 7, 6, 1, 5, 15, 11, 9, 14, 3, 12, 13, 0},
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3}};

__device__ __forceinline__ uint64_t rotr64d(uint64_t x, int n) {
  return (x >> n) | (x << (64 - n));
}

struct b2b_state {
  uint64_t h[8];
  uint64_t t;       /* bytes compressed so far (≤ 2^64) */
  uint8_t buf[128]; /* pending block */
  uint32_t buflen;
};

__device__ inline void b2b_compress(b2b_state &S, const uint8_t block[128], int last) {
  uint64_t v[16], m[16];
#pragma unroll
  for (int i = 0; i < 16; i++) {
    uint64_t w = 0;
#pragma unroll
    for (int j = 0; j < 8; j++) w |= (uint64_t)block[i * 8 + j] << (8 * j);
    m[i] = w;
  }
#pragma unroll
  for (int i = 0; i < 8; i++) v[i] = S.h[i];
#pragma unroll
  for (int i = 0; i < 8; i++) v[i + 8] = B2B_IV[i];
  v[12] ^= S.t;
  if (last) v[14] = ~v[14];
#define KV_B2B_G(a, b, c, d, x, y)                                             \
  v[a] = v[a] + v[b] + (x);                                                    \
  v[d] = rotr64d(v[d] ^ v[a], 32);                                             \
  v[c] = v[c] + v[d];                                                          \
  v[b] = rotr64d(v[b] ^ v[c], 24);                                             \
  v[a] = v[a] + v[b] + (y);                                                    \
  v[d] = rotr64d(v[d] ^ v[a], 16);                                             \
  v[c] = v[c] + v[d];                                                          \
  v[b] = rotr64d(v[b] ^ v[c], 63);
#pragma unroll
  for (int r = 0; r < 12; r++) {
    const uint8_t *s = B2B_SIGMA[r];
    KV_B2B_G(0, 4, 8, 12, m[s[0]], m[s[1]]);
    KV_B2B_G(1, 5, 9, 13, m[s[2]], m[s[3]]);
    KV_B2B_G(2, 6, 10, 14, m[s[4]], m[s[5]]);
    KV_B2B_G(3, 7, 11, 15, m[s[6]], m[s[7]]);
    KV_B2B_G(0, 5, 10, 15, m[s[8]], m[s[9]]);
    KV_B2B_G(1, 6, 11, 12, m[s[10]], m[s[11]]);
    KV_B2B_G(2, 7, 8, 13, m[s[12]], m[s[13]]);
    KV_B2B_G(3, 4, 9, 14, m[s[14]], m[s[15]]);
  }
#undef KV_B2B_G
#pragma unroll
  for (int i = 0; i < 8; i++) S.h[i] ^= v[i] ^ v[i + 8];
}

/* init with ≤64-byte key (domain separator), 32-byte output */
__device__ inline void b2b_init_keyed(b2b_state &S, const uint8_t *key, uint32_t keylen) {
#pragma unroll
  for (int i = 0; i < 8; i++) S.h[i] = B2B_IV[i];
  S.h[0] ^= 32ULL | ((uint64_t)keylen << 8) | (1ULL << 16) | (1ULL << 24);
  S.t = 0;
  S.buflen = 0;
  if (keylen) {
#pragma unroll
    for (int i = 0; i < 128; i++) S.buf[i] = 0;
    for (uint32_t i = 0; i < keylen; i++) S.buf[i] = key[i];
    S.buflen = 128;
  }
}

__device__ inline void b2b_update(b2b_state &S, const uint8_t *data, uint32_t len) {
  while (len > 0) {
    if (S.buflen == 128) {
      S.t += 128;
      b2b_compress(S, S.buf, 0);
      S.buflen = 0;
    }
    uint32_t take = 128 - S.buflen;
    if (take > len) take = len;
    /* copy 8B at a time (unaligned-capable memcpy lowering) — the byte loop
     * made the subhash kernel walk the blob one byte-load per byte */
    uint32_t i = 0;
    for (; i + 8 <= take; i += 8) {
      uint64_t w;
      __builtin_memcpy(&w, data + i, 8);
      __builtin_memcpy(S.buf + S.buflen + i, &w, 8);
    }
    for (; i < take; i++) S.buf[S.buflen + i] = data[i];
    S.buflen += take;
    data += take;
    len -= take;
  }
}

__device__ __forceinline__ void b2b_update_u16(b2b_state &S, uint16_t v) {
  uint8_t b[2] = {(uint8_t)v, (uint8_t)(v >> 8)};
  b2b_update(S, b, 2);
}
__device__ __forceinline__ void b2b_update_u32(b2b_state &S, uint32_t v) {
  uint8_t b[4];
#pragma unroll
  for (int i = 0; i < 4; i++) b[i] = (uint8_t)(v >> (8 * i));
  b2b_update(S, b, 4);
}
__device__ __forceinline__ void b2b_update_u64(b2b_state &S, uint64_t v) {
  uint8_t b[8];
#pragma unroll
  for (int i = 0; i < 8; i++) b[i] = (uint8_t)(v >> (8 * i));
  b2b_update(S, b, 8);
}

__device__ inline void b2b_final(b2b_state &S, uint8_t out[32]) {
  S.t += S.buflen;
  for (uint32_t i = S.buflen; i < 128; i++) S.buf[i] = 0;
  b2b_compress(S, S.buf, 1);
#pragma unroll
  for (int i = 0; i < 32; i++) out[i] = (uint8_t)(S.h[i / 8] >> (8 * (i % 8)));
}

/* ---------------- ChaCha20 384-byte expansion (rand_chacha 0.3.1) ---------- */

__device__ __forceinline__ uint32_t rotl32d(uint32_t x, int n) {
  return (x << n) | (x >> (32 - n));
}

__device__ inline void chacha20_block(const uint32_t key[8], uint32_t counter,
                                      uint8_t out[64]) {
  uint32_t s[16];
  s[0] = 0x61707865; s[1] = 0x3320646e; s[2] = 0x79622d32; s[3] = 0x6b206574;
#pragma unroll
  for (int i = 0; i < 8; i++) s[4 + i] = key[i];
  s[12] = counter; s[13] = 0; s[14] = 0; s[15] = 0;
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 16; i++) w[i] = s[i];
#define KV_QR(a, b, c, d)                                                      \
  w[a] += w[b]; w[d] ^= w[a]; w[d] = rotl32d(w[d], 16);                        \
  w[c] += w[d]; w[b] ^= w[c]; w[b] = rotl32d(w[b], 12);                        \
  w[a] += w[b]; w[d] ^= w[a]; w[d] = rotl32d(w[d], 8);                         \
  w[c] += w[d]; w[b] ^= w[c]; w[b] = rotl32d(w[b], 7);
#pragma unroll
  for (int i = 0; i < 10; i++) {
    KV_QR(0, 4, 8, 12); KV_QR(1, 5, 9, 13); KV_QR(2, 6, 10, 14); KV_QR(3, 7, 11, 15);
    KV_QR(0, 5, 10, 15); KV_QR(1, 6, 11, 12); KV_QR(2, 7, 8, 13); KV_QR(3, 4, 9, 14);
  }
#undef KV_QR
#pragma unroll
  for (int i = 0; i < 16; i++) {
    uint32_t v = w[i] + s[i];
    out[4 * i] = (uint8_t)v;
    out[4 * i + 1] = (uint8_t)(v >> 8);
    out[4 * i + 2] = (uint8_t)(v >> 16);
    out[4 * i + 3] = (uint8_t)(v >> 24);
  }
}

} // namespace kv

#endif /* KV_HASH_DEVICE_H */
