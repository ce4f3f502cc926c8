/* blake2b.c — BLAKE2b (RFC 7693) with keyed/personalized init, plus BLAKE2s.
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note). Needed for the halo2
 * Blake2bWrite Fiat–Shamir transcript (reference proof.rs:32,52) and the
 * Blake2s resource-logic commitment (resource_logic_commitment.rs:17-25).
 * Pinned by RFC 7693 known-answer tests in tests/test_oracle.py.
 */
#include <stdint.h>
#include <string.h>

typedef struct {
    uint64_t h[8];
    uint64_t t[2];
    uint8_t buf[128];
    size_t buflen;
    size_t outlen;
} blake2b_state;

static const uint64_t b2b_iv[8] = {
    0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL, 0x3c6ef372fe94f82bULL,
    0xa54ff53a5f1d36f1ULL, 0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
    0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};

static const uint8_t b2b_sigma[12][16] = {
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

static void b2b_compress(blake2b_state* S, const uint8_t block[128], int last) {
    uint64_t v[16], m[16];
    for (int i = 0; i < 16; i++) memcpy(&m[i], block + 8 * i, 8);
    for (int i = 0; i < 8; i++) v[i] = S->h[i];
    for (int i = 0; i < 8; i++) v[i + 8] = b2b_iv[i];
    v[12] ^= S->t[0];
    v[13] ^= S->t[1];
    if (last) v[14] = ~v[14];
#define G(a, b, c, d, x, y)                     \
    do {                                        \
        v[a] = v[a] + v[b] + x;                 \
        v[d] = rotr64(v[d] ^ v[a], 32);         \
        v[c] = v[c] + v[d];                     \
        v[b] = rotr64(v[b] ^ v[c], 24);         \
        v[a] = v[a] + v[b] + y;                 \
        v[d] = rotr64(v[d] ^ v[a], 16);         \
        v[c] = v[c] + v[d];                     \
        v[b] = rotr64(v[b] ^ v[c], 63);         \
    } while (0)
    for (int r = 0; r < 12; r++) {
        const uint8_t* s = b2b_sigma[r];
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

/* personal: 16 bytes or NULL */
void orc_blake2b_init(blake2b_state* S, size_t outlen, const uint8_t* personal) {
    memset(S, 0, sizeof(*S));
    S->outlen = outlen;
    for (int i = 0; i < 8; i++) S->h[i] = b2b_iv[i];
    /* parameter block: digest_length | key_length<<8 | fanout<<16 | depth<<24 */
    S->h[0] ^= (uint64_t)outlen | (1ULL << 16) | (1ULL << 24);
    if (personal) {
        uint64_t p0, p1;
        memcpy(&p0, personal, 8);
        memcpy(&p1, personal + 8, 8);
        S->h[6] ^= p0;
        S->h[7] ^= p1;
    }
}

void orc_blake2b_update(blake2b_state* S, const uint8_t* in, size_t inlen) {
    while (inlen > 0) {
        if (S->buflen == 128) {
            S->t[0] += 128;
            if (S->t[0] < 128) S->t[1]++;
            b2b_compress(S, S->buf, 0);
            S->buflen = 0;
        }
        size_t take = 128 - S->buflen;
        if (take > inlen) take = inlen;
        memcpy(S->buf + S->buflen, in, take);
        S->buflen += take;
        in += take;
        inlen -= take;
    }
}

void orc_blake2b_final(blake2b_state* S, uint8_t* out) {
    S->t[0] += S->buflen;
    if (S->t[0] < S->buflen) S->t[1]++;
    memset(S->buf + S->buflen, 0, 128 - S->buflen);
    b2b_compress(S, S->buf, 1);
    for (size_t i = 0; i < S->outlen; i++) out[i] = (uint8_t)(S->h[i / 8] >> (8 * (i % 8)));
}

/* one-shot helper for ctypes tests */
void orc_blake2b(const uint8_t* in, long inlen, const uint8_t* personal16_or_null,
                 long outlen, uint8_t* out) {
    blake2b_state S;
    orc_blake2b_init(&S, (size_t)outlen, personal16_or_null);
    orc_blake2b_update(&S, in, (size_t)inlen);
    orc_blake2b_final(&S, out);
}

/* ---------------- BLAKE2s (for resource_logic_commitment; 8-byte personal) */

typedef struct {
    uint32_t h[8];
    uint32_t t[2];
    uint8_t buf[64];
    size_t buflen;
    size_t outlen;
} blake2s_state;

static const uint32_t b2s_iv[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                                   0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};

static inline uint32_t rotr32(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

static void b2s_compress(blake2s_state* S, const uint8_t block[64], int last) {
    uint32_t v[16], m[16];
    for (int i = 0; i < 16; i++) memcpy(&m[i], block + 4 * i, 4);
    for (int i = 0; i < 8; i++) v[i] = S->h[i];
    for (int i = 0; i < 8; i++) v[i + 8] = b2s_iv[i];
    v[12] ^= S->t[0];
    v[13] ^= S->t[1];
    if (last) v[14] = ~v[14];
#define G(a, b, c, d, x, y)                     \
    do {                                        \
        v[a] = v[a] + v[b] + x;                 \
        v[d] = rotr32(v[d] ^ v[a], 16);         \
        v[c] = v[c] + v[d];                     \
        v[b] = rotr32(v[b] ^ v[c], 12);         \
        v[a] = v[a] + v[b] + y;                 \
        v[d] = rotr32(v[d] ^ v[a], 8);          \
        v[c] = v[c] + v[d];                     \
        v[b] = rotr32(v[b] ^ v[c], 7);          \
    } while (0)
    for (int r = 0; r < 10; r++) {
        const uint8_t* s = b2b_sigma[r]; /* same sigma table, first 10 rows */
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

void orc_blake2s_init(blake2s_state* S, size_t outlen, const uint8_t* personal8) {
    memset(S, 0, sizeof(*S));
    S->outlen = outlen;
    for (int i = 0; i < 8; i++) S->h[i] = b2s_iv[i];
    S->h[0] ^= (uint32_t)outlen | (1u << 16) | (1u << 24);
    if (personal8) {
        uint32_t p0, p1;
        memcpy(&p0, personal8, 4);
        memcpy(&p1, personal8 + 4, 4);
        S->h[6] ^= p0;
        S->h[7] ^= p1;
    }
}

void orc_blake2s_update(blake2s_state* S, const uint8_t* in, size_t inlen) {
    while (inlen > 0) {
        if (S->buflen == 64) {
            S->t[0] += 64;
            if (S->t[0] < 64) S->t[1]++;
            b2s_compress(S, S->buf, 0);
            S->buflen = 0;
        }
        size_t take = 64 - S->buflen;
        if (take > inlen) take = inlen;
        memcpy(S->buf + S->buflen, in, take);
        S->buflen += take;
        in += take;
        inlen -= take;
    }
}

void orc_blake2s_final(blake2s_state* S, uint8_t* out) {
    S->t[0] += (uint32_t)S->buflen;
    if (S->t[0] < S->buflen) S->t[1]++;
    memset(S->buf + S->buflen, 0, 64 - S->buflen);
    b2s_compress(S, S->buf, 1);
    for (size_t i = 0; i < S->outlen; i++) out[i] = (uint8_t)(S->h[i / 4] >> (8 * (i % 4)));
}

void orc_blake2s(const uint8_t* in, long inlen, const uint8_t* personal8_or_null,
                 long outlen, uint8_t* out) {
    blake2s_state S;
    orc_blake2s_init(&S, (size_t)outlen, personal8_or_null);
    orc_blake2s_update(&S, in, (size_t)inlen);
    orc_blake2s_final(&S, out);
}
