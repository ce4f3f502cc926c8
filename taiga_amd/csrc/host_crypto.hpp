// host_crypto.hpp — product-side host primitives: BLAKE2b (RFC 7693),
// ChaCha20 DRBG, and the halo2 Blake2bWrite transcript restatement.
// PRODUCT CODE (independent of oracle/; byte-behaviour pinned against the
// oracle by the prover parity tests).
//
// Conventions (same as the oracle):
//   transcript: Blake2b-512 personal "Halo2-Transcript";
//     common_point [0x01]‖x‖y, common_scalar [0x02]‖s, squeeze [0x00]+clone
//   DRBG: ChaCha20 (RFC 8439), key=seed, zero nonce, counter from 0;
//     field draw = 64 bytes as LE 512-bit, wide-reduced
#pragma once
#include "pasta_device.hpp"

#include <cstring>
#include <string>
#include <vector>

namespace taiga {

// ---------------- BLAKE2b ----------------
struct Blake2b {
  u64 h[8];
  u64 t[2];
  uint8_t buf[128];
  size_t buflen = 0;
  size_t outlen;

  static constexpr u64 IV[8] = {0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL,
                                0x3c6ef372fe94f82bULL, 0xa54ff53a5f1d36f1ULL,
                                0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
                                0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};

  explicit Blake2b(size_t outlen_ = 64, const uint8_t* personal16 = nullptr)
      : outlen(outlen_) {
    t[0] = t[1] = 0;
    for (int i = 0; i < 8; i++) h[i] = IV[i];
    h[0] ^= (u64)outlen | (1ULL << 16) | (1ULL << 24);
    if (personal16) {
      u64 p0, p1;
      memcpy(&p0, personal16, 8);
      memcpy(&p1, personal16 + 8, 8);
      h[6] ^= p0;
      h[7] ^= p1;
    }
  }

  static inline u64 rotr(u64 x, int n) { return (x >> n) | (x << (64 - n)); }

  void compress(const uint8_t* block, bool last) {
    static const uint8_t SIG[12][16] = {
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
    u64 v[16], m[16];
    for (int i = 0; i < 16; i++) memcpy(&m[i], block + 8 * i, 8);
    for (int i = 0; i < 8; i++) v[i] = h[i];
    for (int i = 0; i < 8; i++) v[i + 8] = IV[i];
    v[12] ^= t[0];
    v[13] ^= t[1];
    if (last) v[14] = ~v[14];
    auto G = [&](int a, int b, int c, int dd, u64 x, u64 y) {
      v[a] = v[a] + v[b] + x;
      v[dd] = rotr(v[dd] ^ v[a], 32);
      v[c] = v[c] + v[dd];
      v[b] = rotr(v[b] ^ v[c], 24);
      v[a] = v[a] + v[b] + y;
      v[dd] = rotr(v[dd] ^ v[a], 16);
      v[c] = v[c] + v[dd];
      v[b] = rotr(v[b] ^ v[c], 63);
    };
    for (int r = 0; r < 12; r++) {
      const uint8_t* s = SIG[r];
      G(0, 4, 8, 12, m[s[0]], m[s[1]]);
      G(1, 5, 9, 13, m[s[2]], m[s[3]]);
      G(2, 6, 10, 14, m[s[4]], m[s[5]]);
      G(3, 7, 11, 15, m[s[6]], m[s[7]]);
      G(0, 5, 10, 15, m[s[8]], m[s[9]]);
      G(1, 6, 11, 12, m[s[10]], m[s[11]]);
      G(2, 7, 8, 13, m[s[12]], m[s[13]]);
      G(3, 4, 9, 14, m[s[14]], m[s[15]]);
    }
    for (int i = 0; i < 8; i++) h[i] ^= v[i] ^ v[i + 8];
  }

  void update(const uint8_t* in, size_t n) {
    while (n) {
      if (buflen == 128) {
        t[0] += 128;
        if (t[0] < 128) t[1]++;
        compress(buf, false);
        buflen = 0;
      }
      size_t take = 128 - buflen;
      if (take > n) take = n;
      memcpy(buf + buflen, in, take);
      buflen += take;
      in += take;
      n -= take;
    }
  }

  void final(uint8_t* out) {
    t[0] += buflen;
    if (t[0] < buflen) t[1]++;
    memset(buf + buflen, 0, 128 - buflen);
    compress(buf, true);
    for (size_t i = 0; i < outlen; i++) out[i] = (uint8_t)(h[i / 8] >> (8 * (i % 8)));
  }
};

// ---------------- ChaCha20 DRBG ----------------
struct Drbg {
  uint32_t key[8];
  uint32_t counter = 0;
  uint8_t buf[64];
  int pos = 64;

  explicit Drbg(const uint8_t seed[32]) {
    for (int i = 0; i < 8; i++)
      key[i] = (uint32_t)seed[4 * i] | ((uint32_t)seed[4 * i + 1] << 8) |
               ((uint32_t)seed[4 * i + 2] << 16) | ((uint32_t)seed[4 * i + 3] << 24);
  }

  static inline uint32_t rotl(uint32_t x, int n) { return (x << n) | (x >> (32 - n)); }

  static void block(const uint32_t key[8], uint32_t counter, uint8_t out[64]) {
    uint32_t st[16] = {0x61707865, 0x3320646e, 0x79622d32, 0x6b206574,
                       key[0], key[1], key[2], key[3], key[4], key[5], key[6], key[7],
                       counter, 0, 0, 0};
    uint32_t x[16];
    memcpy(x, st, sizeof(x));
    auto QR = [&](int a, int b, int c, int d) {
      x[a] += x[b]; x[d] = rotl(x[d] ^ x[a], 16);
      x[c] += x[d]; x[b] = rotl(x[b] ^ x[c], 12);
      x[a] += x[b]; x[d] = rotl(x[d] ^ x[a], 8);
      x[c] += x[d]; x[b] = rotl(x[b] ^ x[c], 7);
    };
    for (int i = 0; i < 10; i++) {
      QR(0, 4, 8, 12); QR(1, 5, 9, 13); QR(2, 6, 10, 14); QR(3, 7, 11, 15);
      QR(0, 5, 10, 15); QR(1, 6, 11, 12); QR(2, 7, 8, 13); QR(3, 4, 9, 14);
    }
    for (int i = 0; i < 16; i++) {
      uint32_t v = x[i] + st[i];
      out[4 * i] = (uint8_t)v;
      out[4 * i + 1] = (uint8_t)(v >> 8);
      out[4 * i + 2] = (uint8_t)(v >> 16);
      out[4 * i + 3] = (uint8_t)(v >> 24);
    }
  }

  void bytes(uint8_t* out, size_t n) {
    while (n) {
      if (pos == 64) {
        block(key, counter++, buf);
        pos = 0;
      }
      size_t take = 64 - (size_t)pos;
      if (take > n) take = n;
      memcpy(out, buf + pos, take);
      pos += (int)take;
      out += take;
      n -= take;
    }
  }

  // next field element (Mont) via 512-bit wide reduction
  template <class C>
  Fd<C> field() {
    uint8_t b[64];
    bytes(b, 64);
    Fd<C> lo, hi, r2;
    memcpy(lo.l, b, 32);
    memcpy(hi.l, b + 32, 32);
#pragma unroll
    for (int i = 0; i < 4; i++) r2.l[i] = C::R2[i];
    Fd<C> lom = fd_mul(lo, r2);
    Fd<C> him = fd_mul(fd_mul(hi, r2), r2);
    return fd_add(lom, him);
  }
};

// one counter-addressed 64-byte cell (witness generation)
template <class C>
inline Fd<C> drbg_cell_field(const uint8_t seed[32], u64 cell) {
  Drbg d(seed);
  d.counter = (uint32_t)cell;
  d.pos = 64;
  uint8_t b[64];
  d.bytes(b, 64);
  Fd<C> lo, hi, r2;
  memcpy(lo.l, b, 32);
  memcpy(hi.l, b + 32, 32);
#pragma unroll
  for (int i = 0; i < 4; i++) r2.l[i] = C::R2[i];
  return fd_add(fd_mul(lo, r2), fd_mul(fd_mul(hi, r2), r2));
}

// wide-reduce 64 bytes (challenge derivation)
template <class C>
inline Fd<C> from_uniform_512(const uint8_t b[64]) {
  Fd<C> lo, hi, r2;
  memcpy(lo.l, b, 32);
  memcpy(hi.l, b + 32, 32);
#pragma unroll
  for (int i = 0; i < 4; i++) r2.l[i] = C::R2[i];
  return fd_add(fd_mul(lo, r2), fd_mul(fd_mul(hi, r2), r2));
}

// bulk parallel field draws: each draw consumes exactly one 64-byte block,
// so with rng.pos == 64 the stream is counter-addressable (identical bytes
// to sequential field() calls)
inline void drbg_fields_par(Drbg& rng, Fd<FpCfg>* out, long n) {
  uint32_t base = rng.counter;
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long i = 0; i < n; i++) {
    uint8_t b[64];
    Drbg::block(rng.key, base + (uint32_t)i, b);
    out[i] = from_uniform_512<FpCfg>(b);
  }
  rng.counter = base + (uint32_t)n;
}

// ---------------- transcript (Blake2bWrite/Read over vesta::Affine) ------
struct Transcript {
  Blake2b st{64, (const uint8_t*)"Halo2-Transcript"};
  std::vector<uint8_t> proof;  // write mode
  const uint8_t* rbuf = nullptr;
  size_t rpos = 0, rlen = 0;
  bool reading = false;

  void init_read(const uint8_t* p, size_t len) {
    rbuf = p;
    rlen = len;
    rpos = 0;
    reading = true;
  }

  void common_point(const VestaAff& p) {
    uint8_t pre = 1, xb[32], yb[32];
    st.update(&pre, 1);
    Fq x = fd_from_mont(p.x), y = fd_from_mont(p.y);
    memcpy(xb, x.l, 32);
    memcpy(yb, y.l, 32);
    st.update(xb, 32);
    st.update(yb, 32);
  }

  void common_scalar(const Fp& s) {
    uint8_t pre = 2, sb[32];
    st.update(&pre, 1);
    Fp v = fd_from_mont(s);
    memcpy(sb, v.l, 32);
    st.update(sb, 32);
  }

  // 32-byte compressed form (x repr + y-odd bit 255)
  static void compress(uint8_t out[32], const VestaAff& p) {
    Fq x = fd_from_mont(p.x), y = fd_from_mont(p.y);
    memcpy(out, x.l, 32);
    out[31] |= (uint8_t)((y.l[0] & 1) << 7);
  }

  int write_point(const VestaAff& p) {
    if (aff_is_identity(p)) return -1;
    common_point(p);
    uint8_t cb[32];
    compress(cb, p);
    proof.insert(proof.end(), cb, cb + 32);
    return 0;
  }

  void write_scalar(const Fp& s) {
    common_scalar(s);
    uint8_t sb[32];
    Fp v = fd_from_mont(s);
    memcpy(sb, v.l, 32);
    proof.insert(proof.end(), sb, sb + 32);
  }

  Fp squeeze() {
    uint8_t pre = 0;
    st.update(&pre, 1);
    Blake2b clone = st;
    uint8_t dig[64];
    clone.final(dig);
    return from_uniform_512<FpCfg>(dig);
  }

  // ---- read mode (verifier) ----
  static bool fq_canonical(const Fq& v) {
    for (int limb = 3; limb >= 0; limb--) {
      if (v.l[limb] > FqCfg::MOD[limb]) return false;
      if (v.l[limb] < FqCfg::MOD[limb]) return true;
    }
    return false;
  }

  // 32-byte compressed -> Mont affine (host Tonelli-Shanks); rejects
  // identity and invalid encodings (transcript points are never identity)
  static bool decompress(VestaAff& out, const uint8_t in[32]) {
    u64 l[4];
    memcpy(l, in, 32);
    unsigned sign = (unsigned)(l[3] >> 63);
    l[3] &= 0x7FFFFFFFFFFFFFFFULL;
    if ((l[0] | l[1] | l[2] | l[3]) == 0) return false;
    Fq x{{l[0], l[1], l[2], l[3]}};
    if (!fq_canonical(x)) return false;
    Fq xm = fd_to_mont(x);
    Fq five{{5, 0, 0, 0}};
    Fq rhs = fd_add(fd_mul(fd_sqr(xm), xm), fd_to_mont(five));
    Fq y;
    if (!fd_sqrt(y, rhs)) return false;
    if (fd_is_odd_std(y) != (bool)sign) y = fd_neg(y);
    out.x = xm;
    out.y = y;
    return true;
  }

  bool read_point(VestaAff& p) {
    if (rpos + 32 > rlen) return false;
    if (!decompress(p, rbuf + rpos)) return false;
    rpos += 32;
    common_point(p);
    return true;
  }

  bool read_scalar(Fp& s) {
    if (rpos + 32 > rlen) return false;
    Fp v;
    memcpy(v.l, rbuf + rpos, 32);
    for (int limb = 3;; limb--) {
      if (v.l[limb] > FpCfg::MOD[limb]) return false;
      if (v.l[limb] < FpCfg::MOD[limb]) break;
      if (limb == 0) return false;
    }
    rpos += 32;
    s = fd_to_mont(v);
    common_scalar(s);
    return true;
  }
};

}  // namespace taiga
