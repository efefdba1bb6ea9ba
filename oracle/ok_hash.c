/* ORACLE (test infrastructure only — see oracle/README.md).
 *
 * CPU restatement of the reference's hash constructions:
 *   - keyed BLAKE2b-256  (reference: crypto/hashes/src/hashers.rs:88-117, blake2b_simd
 *     Params::new().hash_length(32).key(domain) — RFC 7693 keyed mode)
 *   - SHA-256 and the domain-prefixed SHA-256 hasher
 *     (reference: crypto/hashes/src/hashers.rs:56-85)
 *   - keyed BLAKE3 (reference: crypto/hashes/src/hashers.rs:119-154, blake3::Hasher::new_keyed
 *     with the domain string zero-padded to 32 bytes)
 *
 * Only tests/, bench.py's cpu_baseline leg and __graft_entry__.smoke() may link this.
 */
#include "oracle.h"
#include <string.h>

/* ---------------- BLAKE2b (RFC 7693) ---------------- */

static const uint64_t B2B_IV[8] = {
    0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL, 0x3c6ef372fe94f82bULL,
    0xa54ff53a5f1d36f1ULL, 0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
    0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};

static const uint8_t B2B_SIGMA[12][16] = {
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

static inline uint64_t rotr64(uint64_t x, int n) { return (x >> n) | (x << (64 - n)); }

static void b2b_compress(ok_blake2b_state *S, const uint8_t block[128], int last) {
  uint64_t v[16], m[16];
  for (int i = 0; i < 16; i++) {
    m[i] = 0;
    for (int j = 0; j < 8; j++) m[i] |= (uint64_t)block[i * 8 + j] << (8 * j);
  }
  for (int i = 0; i < 8; i++) v[i] = S->h[i];
  for (int i = 0; i < 8; i++) v[i + 8] = B2B_IV[i];
  v[12] ^= S->t[0];
  v[13] ^= S->t[1];
  if (last) v[14] = ~v[14];
#define G(a, b, c, d, x, y)                                                    \
  do {                                                                         \
    v[a] = v[a] + v[b] + (x);                                                  \
    v[d] = rotr64(v[d] ^ v[a], 32);                                            \
    v[c] = v[c] + v[d];                                                        \
    v[b] = rotr64(v[b] ^ v[c], 24);                                            \
    v[a] = v[a] + v[b] + (y);                                                  \
    v[d] = rotr64(v[d] ^ v[a], 16);                                            \
    v[c] = v[c] + v[d];                                                        \
    v[b] = rotr64(v[b] ^ v[c], 63);                                            \
  } while (0)
  for (int r = 0; r < 12; r++) {
    const uint8_t *s = B2B_SIGMA[r];
    G(0, 4, 8, 12, m[s[0]], m[s[1]]);
    G(1, 5, 9, 13, m[s[2]], m[s[3]]);
    G(2, 6, 10, 14, m[s[4]], m[s[5]]);
    G(3, 7, 11, 15, m[s[6]], m[s[7]]);
    G(0, 5, 10, 15, m[s[8]], m[s[9]]);
    G(1, 6, 11, 12, m[s[10]], m[s[11]]);
    G(2, 7, 8, 13, m[s[12]], m[s[13]]);
    G(3, 4, 9, 14, m[s[14]], m[s[15]]);
  }
#undef G
  for (int i = 0; i < 8; i++) S->h[i] ^= v[i] ^ v[i + 8];
}

void ok_blake2b_init(ok_blake2b_state *S, const uint8_t *key, size_t keylen,
                     size_t outlen) {
  memset(S, 0, sizeof(*S));
  S->outlen = outlen;
  for (int i = 0; i < 8; i++) S->h[i] = B2B_IV[i];
  /* parameter block word 0: digest_length | key_length<<8 | fanout<<16 | depth<<24 */
  S->h[0] ^= (uint64_t)outlen | ((uint64_t)keylen << 8) | (1ULL << 16) | (1ULL << 24);
  if (keylen > 0) {
    uint8_t block[128] = {0};
    memcpy(block, key, keylen);
    ok_blake2b_update(S, block, 128);
  }
}

void ok_blake2b_update(ok_blake2b_state *S, const uint8_t *data, size_t len) {
  while (len > 0) {
    if (S->buflen == 128) {
      S->t[0] += 128;
      if (S->t[0] < 128) S->t[1]++;
      b2b_compress(S, S->buf, 0);
      S->buflen = 0;
    }
    size_t take = 128 - S->buflen;
    if (take > len) take = len;
    memcpy(S->buf + S->buflen, data, take);
    S->buflen += take;
    data += take;
    len -= take;
  }
}

void ok_blake2b_final(ok_blake2b_state *S, uint8_t *out) {
  S->t[0] += S->buflen;
  if (S->t[0] < S->buflen) S->t[1]++;
  memset(S->buf + S->buflen, 0, 128 - S->buflen);
  b2b_compress(S, S->buf, 1);
  for (size_t i = 0; i < S->outlen; i++) out[i] = (uint8_t)(S->h[i / 8] >> (8 * (i % 8)));
}

void ok_blake2b_keyed(const uint8_t *key, size_t keylen, const uint8_t *data,
                      size_t len, uint8_t out32[32]) {
  ok_blake2b_state S;
  ok_blake2b_init(&S, key, keylen, 32);
  ok_blake2b_update(&S, data, len);
  ok_blake2b_final(&S, out32);
}

/* ---------------- SHA-256 (FIPS 180-4) ---------------- */

static const uint32_t SHA256_K[64] = {
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

static inline uint32_t rotr32(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

static void sha256_compress(uint32_t h[8], const uint8_t block[64]) {
  uint32_t w[64];
  for (int i = 0; i < 16; i++)
    w[i] = ((uint32_t)block[4 * i] << 24) | ((uint32_t)block[4 * i + 1] << 16) |
           ((uint32_t)block[4 * i + 2] << 8) | block[4 * i + 3];
  for (int i = 16; i < 64; i++) {
    uint32_t s0 = rotr32(w[i - 15], 7) ^ rotr32(w[i - 15], 18) ^ (w[i - 15] >> 3);
    uint32_t s1 = rotr32(w[i - 2], 17) ^ rotr32(w[i - 2], 19) ^ (w[i - 2] >> 10);
    w[i] = w[i - 16] + s0 + w[i - 7] + s1;
  }
  uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5], g = h[6], hh = h[7];
  for (int i = 0; i < 64; i++) {
    uint32_t S1 = rotr32(e, 6) ^ rotr32(e, 11) ^ rotr32(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = hh + S1 + ch + SHA256_K[i] + w[i];
    uint32_t S0 = rotr32(a, 2) ^ rotr32(a, 13) ^ rotr32(a, 22);
    uint32_t mj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + mj;
    hh = g; g = f; f = e; e = d + t1; d = c; c = b; b = a; a = t1 + t2;
  }
  h[0] += a; h[1] += b; h[2] += c; h[3] += d; h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

void ok_sha256_init(ok_sha256_state *S) {
  static const uint32_t iv[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                                 0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};
  memcpy(S->h, iv, sizeof(iv));
  S->nbytes = 0;
  S->buflen = 0;
}

void ok_sha256_update(ok_sha256_state *S, const uint8_t *data, size_t len) {
  S->nbytes += len;
  while (len > 0) {
    size_t take = 64 - S->buflen;
    if (take > len) take = len;
    memcpy(S->buf + S->buflen, data, take);
    S->buflen += take;
    data += take;
    len -= take;
    if (S->buflen == 64) {
      sha256_compress(S->h, S->buf);
      S->buflen = 0;
    }
  }
}

void ok_sha256_final(ok_sha256_state *S, uint8_t out32[32]) {
  uint64_t bits = S->nbytes * 8;
  uint8_t pad = 0x80;
  ok_sha256_update(S, &pad, 1);
  S->nbytes -= 1;
  uint8_t zero = 0;
  while (S->buflen != 56) {
    ok_sha256_update(S, &zero, 1);
    S->nbytes -= 1;
  }
  uint8_t lenb[8];
  for (int i = 0; i < 8; i++) lenb[i] = (uint8_t)(bits >> (56 - 8 * i));
  ok_sha256_update(S, lenb, 8);
  for (int i = 0; i < 8; i++) {
    out32[4 * i] = (uint8_t)(S->h[i] >> 24);
    out32[4 * i + 1] = (uint8_t)(S->h[i] >> 16);
    out32[4 * i + 2] = (uint8_t)(S->h[i] >> 8);
    out32[4 * i + 3] = (uint8_t)S->h[i];
  }
}

void ok_sha256(const uint8_t *data, size_t len, uint8_t out32[32]) {
  ok_sha256_state S;
  ok_sha256_init(&S);
  ok_sha256_update(&S, data, len);
  ok_sha256_final(&S, out32);
}

/* Domain-prefixed SHA-256: SHA256(SHA256(domain) || data)
 * (reference: crypto/hashes/src/hashers.rs:56-85 — new_with_prefix(DOMAIN_HASH)) */
void ok_sha256_domain(const uint8_t *domain, size_t domain_len, const uint8_t *data,
                      size_t len, uint8_t out32[32]) {
  uint8_t prefix[32];
  ok_sha256(domain, domain_len, prefix);
  ok_sha256_state S;
  ok_sha256_init(&S);
  ok_sha256_update(&S, prefix, 32);
  ok_sha256_update(&S, data, len);
  ok_sha256_final(&S, out32);
}

/* ---------------- BLAKE3 (keyed + regular) ---------------- */

static const uint32_t B3_IV[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                                  0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};
enum {
  B3_CHUNK_START = 1,
  B3_CHUNK_END = 2,
  B3_PARENT = 4,
  B3_ROOT = 8,
  B3_KEYED_HASH = 16,
};

static void b3_g(uint32_t *s, int a, int b, int c, int d, uint32_t mx, uint32_t my) {
  s[a] = s[a] + s[b] + mx;
  s[d] = rotr32(s[d] ^ s[a], 16);
  s[c] = s[c] + s[d];
  s[b] = rotr32(s[b] ^ s[c], 12);
  s[a] = s[a] + s[b] + my;
  s[d] = rotr32(s[d] ^ s[a], 8);
  s[c] = s[c] + s[d];
  s[b] = rotr32(s[b] ^ s[c], 7);
}

static void b3_compress(const uint32_t cv[8], const uint32_t block[16], uint64_t counter,
                        uint32_t block_len, uint32_t flags, uint32_t out[16]) {
  static const uint8_t P[7][16] = {
      {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
      {2, 6, 3, 10, 7, 0, 4, 13, 1, 11, 12, 5, 9, 14, 15, 8},
      {3, 4, 10, 12, 13, 2, 7, 14, 6, 5, 9, 0, 11, 15, 8, 1},
      {10, 7, 12, 9, 14, 3, 13, 15, 4, 0, 11, 2, 5, 8, 1, 6},
      {12, 13, 9, 11, 15, 10, 14, 8, 7, 2, 5, 3, 0, 1, 6, 4},
      {9, 14, 11, 5, 8, 12, 15, 1, 13, 3, 0, 10, 2, 6, 4, 7},
      {11, 15, 5, 0, 1, 9, 8, 6, 14, 10, 2, 12, 3, 4, 7, 13}};
  uint32_t s[16];
  for (int i = 0; i < 8; i++) s[i] = cv[i];
  for (int i = 0; i < 4; i++) s[8 + i] = B3_IV[i];
  s[12] = (uint32_t)counter;
  s[13] = (uint32_t)(counter >> 32);
  s[14] = block_len;
  s[15] = flags;
  for (int r = 0; r < 7; r++) {
    const uint8_t *p = P[r];
    b3_g(s, 0, 4, 8, 12, block[p[0]], block[p[1]]);
    b3_g(s, 1, 5, 9, 13, block[p[2]], block[p[3]]);
    b3_g(s, 2, 6, 10, 14, block[p[4]], block[p[5]]);
    b3_g(s, 3, 7, 11, 15, block[p[6]], block[p[7]]);
    b3_g(s, 0, 5, 10, 15, block[p[8]], block[p[9]]);
    b3_g(s, 1, 6, 11, 12, block[p[10]], block[p[11]]);
    b3_g(s, 2, 7, 8, 13, block[p[12]], block[p[13]]);
    b3_g(s, 3, 4, 9, 14, block[p[14]], block[p[15]]);
  }
  for (int i = 0; i < 8; i++) {
    out[i] = s[i] ^ s[i + 8];
    out[i + 8] = s[i + 8] ^ cv[i];
  }
}

static void b3_words_from_le(const uint8_t *b, size_t len, uint32_t w[16]) {
  uint8_t tmp[64] = {0};
  memcpy(tmp, b, len);
  for (int i = 0; i < 16; i++)
    w[i] = (uint32_t)tmp[4 * i] | ((uint32_t)tmp[4 * i + 1] << 8) |
           ((uint32_t)tmp[4 * i + 2] << 16) | ((uint32_t)tmp[4 * i + 3] << 24);
}

/* chunk hash: returns output chaining value (8 words); if root_out != NULL and this is
 * the root, writes 32-byte root output instead. Simple recursive tree over full input
 * (oracle-only; not performance relevant). */
static void b3_chunk_cv(const uint32_t key[8], const uint8_t *chunk, size_t len,
                        uint64_t chunk_counter, uint32_t base_flags, int is_root,
                        uint32_t cv_out[8], uint8_t *root_out) {
  uint32_t cv[8];
  memcpy(cv, key, 32);
  size_t nblocks = (len + 63) / 64;
  if (nblocks == 0) nblocks = 1;
  for (size_t i = 0; i < nblocks; i++) {
    size_t blen = (i == nblocks - 1) ? len - 64 * i : 64;
    uint32_t block[16];
    b3_words_from_le(chunk + 64 * i, blen, block);
    uint32_t flags = base_flags;
    if (i == 0) flags |= B3_CHUNK_START;
    if (i == nblocks - 1) {
      flags |= B3_CHUNK_END;
      if (is_root) flags |= B3_ROOT;
    }
    uint32_t out[16];
    b3_compress(cv, block, chunk_counter, (uint32_t)blen, flags, out);
    if (i == nblocks - 1 && is_root && root_out) {
      for (int j = 0; j < 8; j++) {
        root_out[4 * j] = (uint8_t)out[j];
        root_out[4 * j + 1] = (uint8_t)(out[j] >> 8);
        root_out[4 * j + 2] = (uint8_t)(out[j] >> 16);
        root_out[4 * j + 3] = (uint8_t)(out[j] >> 24);
      }
      return;
    }
    memcpy(cv, out, 32);
  }
  memcpy(cv_out, cv, 32);
}

/* recursive subtree over `nchunks` chunks starting at chunk_counter; produces cv */
static void b3_subtree(const uint32_t key[8], const uint8_t *data, size_t len,
                       uint64_t chunk_counter, uint32_t base_flags, int is_root,
                       uint32_t cv_out[8], uint8_t *root_out) {
  if (len <= 1024) {
    b3_chunk_cv(key, data, len, chunk_counter, base_flags, is_root, cv_out, root_out);
    return;
  }
  /* left subtree gets the largest power-of-two chunks strictly less than total */
  size_t chunks = (len + 1023) / 1024;
  size_t left_chunks = 1;
  while (left_chunks * 2 < chunks) left_chunks *= 2;
  size_t left_len = left_chunks * 1024;
  uint32_t lcv[8], rcv[8];
  b3_subtree(key, data, left_len, chunk_counter, base_flags, 0, lcv, NULL);
  b3_subtree(key, data + left_len, len - left_len, chunk_counter + left_chunks,
             base_flags, 0, rcv, NULL);
  uint32_t block[16];
  for (int i = 0; i < 8; i++) { block[i] = lcv[i]; block[i + 8] = rcv[i]; }
  uint32_t flags = base_flags | B3_PARENT;
  if (is_root) flags |= B3_ROOT;
  uint32_t out[16];
  b3_compress(key, block, 0, 64, flags, out);
  if (is_root && root_out) {
    for (int j = 0; j < 8; j++) {
      root_out[4 * j] = (uint8_t)out[j];
      root_out[4 * j + 1] = (uint8_t)(out[j] >> 8);
      root_out[4 * j + 2] = (uint8_t)(out[j] >> 16);
      root_out[4 * j + 3] = (uint8_t)(out[j] >> 24);
    }
    return;
  }
  memcpy(cv_out, out, 32);
}

void ok_blake3(const uint8_t *data, size_t len, uint8_t out32[32]) {
  uint32_t cv[8];
  b3_subtree(B3_IV, data, len, 0, 0, 1, cv, out32);
}

void ok_blake3_keyed(const uint8_t key[32], const uint8_t *data, size_t len,
                     uint8_t out32[32]) {
  uint32_t kw[8];
  for (int i = 0; i < 8; i++)
    kw[i] = (uint32_t)key[4 * i] | ((uint32_t)key[4 * i + 1] << 8) |
            ((uint32_t)key[4 * i + 2] << 16) | ((uint32_t)key[4 * i + 3] << 24);
  uint32_t cv[8];
  b3_subtree(kw, data, len, 0, B3_KEYED_HASH, 1, cv, out32);
}

/* ---------------- ChaCha20 keystream (rand_chacha 0.3.1 semantics) ----------------
 * MuHash element expansion (reference: crypto/muhash/src/lib.rs:153-169):
 * ChaCha20Rng::from_seed(hash32) then fill_bytes(&mut [0u8; 384]).
 * rand_chacha: key = seed, 64-bit stream id = 0 (nonce), 32-bit block counter
 * starting at 0, output = state words serialized little-endian, blocks sequential. */

static inline uint32_t rotl32(uint32_t x, int n) { return (x << n) | (x >> (32 - n)); }

static void chacha20_block(const uint32_t key[8], uint64_t nonce, uint32_t counter,
                           uint8_t out[64]) {
  uint32_t s[16];
  s[0] = 0x61707865; s[1] = 0x3320646e; s[2] = 0x79622d32; s[3] = 0x6b206574;
  for (int i = 0; i < 8; i++) s[4 + i] = key[i];
  /* rand_chacha uses a 64-bit counter in words 12..13 and 64-bit nonce in 14..15.
   * (rand_chacha's c2-chacha backend: [constant, key, counter64, nonce64]) */
  s[12] = counter;
  s[13] = 0;
  s[14] = (uint32_t)nonce;
  s[15] = (uint32_t)(nonce >> 32);
  uint32_t w[16];
  memcpy(w, s, sizeof(s));
#define QR(a, b, c, d)                                                         \
  do {                                                                         \
    w[a] += w[b]; w[d] ^= w[a]; w[d] = rotl32(w[d], 16);                       \
    w[c] += w[d]; w[b] ^= w[c]; w[b] = rotl32(w[b], 12);                       \
    w[a] += w[b]; w[d] ^= w[a]; w[d] = rotl32(w[d], 8);                        \
    w[c] += w[d]; w[b] ^= w[c]; w[b] = rotl32(w[b], 7);                        \
  } while (0)
  for (int i = 0; i < 10; i++) {
    QR(0, 4, 8, 12); QR(1, 5, 9, 13); QR(2, 6, 10, 14); QR(3, 7, 11, 15);
    QR(0, 5, 10, 15); QR(1, 6, 11, 12); QR(2, 7, 8, 13); QR(3, 4, 9, 14);
  }
#undef QR
  for (int i = 0; i < 16; i++) {
    uint32_t v = w[i] + s[i];
    out[4 * i] = (uint8_t)v;
    out[4 * i + 1] = (uint8_t)(v >> 8);
    out[4 * i + 2] = (uint8_t)(v >> 16);
    out[4 * i + 3] = (uint8_t)(v >> 24);
  }
}

void ok_chacha20_block384(const uint8_t seed[32], uint8_t out[384]) {
  uint32_t key[8];
  for (int i = 0; i < 8; i++)
    key[i] = (uint32_t)seed[4 * i] | ((uint32_t)seed[4 * i + 1] << 8) |
             ((uint32_t)seed[4 * i + 2] << 16) | ((uint32_t)seed[4 * i + 3] << 24);
  for (int b = 0; b < 6; b++) chacha20_block(key, 0, (uint32_t)b, out + 64 * b);
}
