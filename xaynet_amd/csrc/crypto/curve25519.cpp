#include "curve25519.h"

#include "../bigint.h"
#include "../common.h"
#include "sha2.h"

namespace xaynet::crypto {

using u128 = unsigned __int128;

// ===================================================================== fe25519
// Field element mod p = 2^255 - 19, radix 2^51, 5 limbs.

struct fe {
    uint64_t v[5];
};

static const uint64_t MASK51 = 0x7ffffffffffffULL;

static fe fe_zero() { return fe{{0, 0, 0, 0, 0}}; }
static fe fe_one() { return fe{{1, 0, 0, 0, 0}}; }

static void fe_carry(fe& a);

// add/sub outputs are always carry-normalized (limbs < 2^52), so every fe in
// flight is a valid fe_sub/fe_mul input. This is CPU-side protocol crypto, not
// a hot path; uniform invariants beat the few saved cycles.
static fe fe_add(const fe& a, const fe& b) {
    fe r;
    for (int i = 0; i < 5; ++i) r.v[i] = a.v[i] + b.v[i];
    fe_carry(r);
    return r;
}

// a - b + 2*p, then carry
static fe fe_sub(const fe& a, const fe& b) {
    fe r;
    r.v[0] = a.v[0] + 0xfffffffffffdaULL - b.v[0];
    r.v[1] = a.v[1] + 0xffffffffffffeULL - b.v[1];
    r.v[2] = a.v[2] + 0xffffffffffffeULL - b.v[2];
    r.v[3] = a.v[3] + 0xffffffffffffeULL - b.v[3];
    r.v[4] = a.v[4] + 0xffffffffffffeULL - b.v[4];
    fe_carry(r);
    return r;
}

static fe fe_mul(const fe& f, const fe& g) {
    u128 r0 = (u128)f.v[0] * g.v[0] + (u128)(19 * f.v[1]) * g.v[4] + (u128)(19 * f.v[2]) * g.v[3] +
              (u128)(19 * f.v[3]) * g.v[2] + (u128)(19 * f.v[4]) * g.v[1];
    u128 r1 = (u128)f.v[0] * g.v[1] + (u128)f.v[1] * g.v[0] + (u128)(19 * f.v[2]) * g.v[4] +
              (u128)(19 * f.v[3]) * g.v[3] + (u128)(19 * f.v[4]) * g.v[2];
    u128 r2 = (u128)f.v[0] * g.v[2] + (u128)f.v[1] * g.v[1] + (u128)f.v[2] * g.v[0] +
              (u128)(19 * f.v[3]) * g.v[4] + (u128)(19 * f.v[4]) * g.v[3];
    u128 r3 = (u128)f.v[0] * g.v[3] + (u128)f.v[1] * g.v[2] + (u128)f.v[2] * g.v[1] +
              (u128)f.v[3] * g.v[0] + (u128)(19 * f.v[4]) * g.v[4];
    u128 r4 = (u128)f.v[0] * g.v[4] + (u128)f.v[1] * g.v[3] + (u128)f.v[2] * g.v[2] +
              (u128)f.v[3] * g.v[1] + (u128)f.v[4] * g.v[0];

    fe out;
    uint64_t c;
    c = uint64_t(r0 >> 51); out.v[0] = uint64_t(r0) & MASK51; r1 += c;
    c = uint64_t(r1 >> 51); out.v[1] = uint64_t(r1) & MASK51; r2 += c;
    c = uint64_t(r2 >> 51); out.v[2] = uint64_t(r2) & MASK51; r3 += c;
    c = uint64_t(r3 >> 51); out.v[3] = uint64_t(r3) & MASK51; r4 += c;
    c = uint64_t(r4 >> 51); out.v[4] = uint64_t(r4) & MASK51;
    out.v[0] += c * 19;
    c = out.v[0] >> 51; out.v[0] &= MASK51;
    out.v[1] += c;
    return out;
}

static fe fe_sq(const fe& a) { return fe_mul(a, a); }

static fe fe_mul_small(const fe& a, uint64_t s) {
    u128 r0 = (u128)a.v[0] * s, r1 = (u128)a.v[1] * s, r2 = (u128)a.v[2] * s,
         r3 = (u128)a.v[3] * s, r4 = (u128)a.v[4] * s;
    fe out;
    uint64_t c;
    c = uint64_t(r0 >> 51); out.v[0] = uint64_t(r0) & MASK51; r1 += c;
    c = uint64_t(r1 >> 51); out.v[1] = uint64_t(r1) & MASK51; r2 += c;
    c = uint64_t(r2 >> 51); out.v[2] = uint64_t(r2) & MASK51; r3 += c;
    c = uint64_t(r3 >> 51); out.v[3] = uint64_t(r3) & MASK51; r4 += c;
    c = uint64_t(r4 >> 51); out.v[4] = uint64_t(r4) & MASK51;
    out.v[0] += c * 19;
    return out;
}

static void fe_carry(fe& a) {
    uint64_t c;
    for (int pass = 0; pass < 2; ++pass) {
        for (int i = 0; i < 4; ++i) {
            c = a.v[i] >> 51;
            a.v[i] &= MASK51;
            a.v[i + 1] += c;
        }
        c = a.v[4] >> 51;
        a.v[4] &= MASK51;
        a.v[0] += 19 * c;
    }
}

// canonical reduction + serialize
static void fe_tobytes(uint8_t out[32], const fe& in) {
    fe a = in;
    fe_carry(a);
    // reduce mod p fully: add 19, see if it overflows 2^255
    uint64_t c = (a.v[0] + 19) >> 51;
    c = (a.v[1] + c) >> 51;
    c = (a.v[2] + c) >> 51;
    c = (a.v[3] + c) >> 51;
    c = (a.v[4] + c) >> 51;  // 1 iff a >= p
    a.v[0] += 19 * c;
    for (int i = 0; i < 4; ++i) {
        a.v[i + 1] += a.v[i] >> 51;
        a.v[i] &= MASK51;
    }
    a.v[4] &= MASK51;

    uint64_t w0 = a.v[0] | (a.v[1] << 51);
    uint64_t w1 = (a.v[1] >> 13) | (a.v[2] << 38);
    uint64_t w2 = (a.v[2] >> 26) | (a.v[3] << 25);
    uint64_t w3 = (a.v[3] >> 39) | (a.v[4] << 12);
    store64_le(out + 0, w0);
    store64_le(out + 8, w1);
    store64_le(out + 16, w2);
    store64_le(out + 24, w3);
}

static fe fe_frombytes(const uint8_t in[32]) {
    uint64_t w0 = load64_le(in + 0), w1 = load64_le(in + 8), w2 = load64_le(in + 16),
             w3 = load64_le(in + 24);
    fe r;
    r.v[0] = w0 & MASK51;
    r.v[1] = ((w0 >> 51) | (w1 << 13)) & MASK51;
    r.v[2] = ((w1 >> 38) | (w2 << 26)) & MASK51;
    r.v[3] = ((w2 >> 25) | (w3 << 39)) & MASK51;
    r.v[4] = (w3 >> 12) & MASK51;  // top bit dropped
    return r;
}

static fe fe_pow(const fe& a, const uint8_t* exp_le, size_t nbits) {
    fe result = fe_one();
    fe base = a;
    for (size_t i = 0; i < nbits; ++i) {
        if ((exp_le[i / 8] >> (i % 8)) & 1) result = fe_mul(result, base);
        base = fe_sq(base);
    }
    return result;
}

// a^(p-2) = inverse
static fe fe_invert(const fe& a) {
    // p - 2 = 2^255 - 21
    uint8_t e[32];
    for (int i = 0; i < 32; ++i) e[i] = 0xff;
    e[0] = 0xeb;
    e[31] = 0x7f;
    return fe_pow(a, e, 255);
}

// a^((p-3)/8) used for square roots; (p-3)/8 = (2^255-22)/8 = 2^252 - 3
static fe fe_pow252m3(const fe& a) {
    uint8_t e[32];
    for (int i = 0; i < 32; ++i) e[i] = 0xff;
    e[0] = 0xfd;
    e[31] = 0x0f;
    return fe_pow(a, e, 253);
}

static bool fe_isnegative(const fe& a) {
    uint8_t b[32];
    fe_tobytes(b, a);
    return b[0] & 1;
}

static bool fe_iszero(const fe& a) {
    uint8_t b[32];
    fe_tobytes(b, a);
    for (int i = 0; i < 32; ++i)
        if (b[i]) return false;
    return true;
}

static bool fe_eq(const fe& a, const fe& b) {
    uint8_t x[32], y[32];
    fe_tobytes(x, a);
    fe_tobytes(y, b);
    return std::memcmp(x, y, 32) == 0;
}

static fe fe_neg(const fe& a) { return fe_sub(fe_zero(), a); }

static void fe_cswap(fe& a, fe& b, uint64_t swap) {
    uint64_t mask = 0 - swap;
    for (int i = 0; i < 5; ++i) {
        uint64_t t = mask & (a.v[i] ^ b.v[i]);
        a.v[i] ^= t;
        b.v[i] ^= t;
    }
}

// ===================================================================== X25519

static void clamp(uint8_t k[32]) {
    k[0] &= 248;
    k[31] &= 127;
    k[31] |= 64;
}

void x25519(uint8_t out[32], const uint8_t scalar[32], const uint8_t point[32]) {
    uint8_t k[32];
    std::memcpy(k, scalar, 32);
    clamp(k);

    fe x1 = fe_frombytes(point);
    fe x2 = fe_one(), z2 = fe_zero();
    fe x3 = x1, z3 = fe_one();
    uint64_t swap = 0;

    for (int t = 254; t >= 0; --t) {
        uint64_t kt = (k[t / 8] >> (t % 8)) & 1;
        swap ^= kt;
        fe_cswap(x2, x3, swap);
        fe_cswap(z2, z3, swap);
        swap = kt;

        fe A = fe_add(x2, z2);
        fe AA = fe_sq(A);
        fe B = fe_sub(x2, z2);
        fe BB = fe_sq(B);
        fe E = fe_sub(AA, BB);
        fe C = fe_add(x3, z3);
        fe D = fe_sub(x3, z3);
        fe DA = fe_mul(D, A);
        fe CB = fe_mul(C, B);
        fe t0 = fe_add(DA, CB);
        x3 = fe_sq(t0);
        fe t1 = fe_sub(DA, CB);
        z3 = fe_mul(x1, fe_sq(t1));
        x2 = fe_mul(AA, BB);
        fe t2 = fe_mul_small(E, 121665);
        z2 = fe_mul(E, fe_add(AA, t2));
    }
    fe_cswap(x2, x3, swap);
    fe_cswap(z2, z3, swap);
    fe_tobytes(out, fe_mul(x2, fe_invert(z2)));
}

void x25519_base(uint8_t out[32], const uint8_t scalar[32]) {
    uint8_t nine[32] = {9};
    x25519(out, scalar, nine);
}

// ===================================================================== Ed25519

// point in extended coordinates (X:Y:Z:T), x*y = T*Z
struct ge {
    fe X, Y, Z, T;
};

// d = -121665/121666
static fe fe_d() {
    static fe d = [] {
        fe n = fe_neg(fe{{121665, 0, 0, 0, 0}});
        fe dn = fe_invert(fe{{121666, 0, 0, 0, 0}});
        return fe_mul(n, dn);
    }();
    return d;
}

static ge ge_identity() { return ge{fe_zero(), fe_one(), fe_one(), fe_zero()}; }

// unified addition for a=-1 twisted Edwards (add-2008-hwcd-3 variant without
// precomputation: uses d directly).
static ge ge_add(const ge& p, const ge& q) {
    fe A = fe_mul(fe_sub(p.Y, p.X), fe_sub(q.Y, q.X));
    fe B = fe_mul(fe_add(p.Y, p.X), fe_add(q.Y, q.X));
    fe C = fe_mul(fe_mul(p.T, q.T), fe_mul_small(fe_d(), 2));
    fe D = fe_mul(p.Z, fe_mul_small(q.Z, 2));
    fe E = fe_sub(B, A);
    fe F = fe_sub(D, C);
    fe G = fe_add(D, C);
    fe H = fe_add(B, A);
    return ge{fe_mul(E, F), fe_mul(G, H), fe_mul(F, G), fe_mul(E, H)};
}

// dedicated doubling (dbl-2008-hwcd, 4M+4S vs the unified add's 9M)
static ge ge_double(const ge& p) {
    fe A = fe_sq(p.X);
    fe B = fe_sq(p.Y);
    fe C = fe_mul_small(fe_sq(p.Z), 2);
    fe H = fe_add(A, B);
    fe E = fe_sub(H, fe_sq(fe_add(p.X, p.Y)));
    fe G = fe_sub(A, B);
    fe F = fe_add(C, G);
    return ge{fe_mul(E, F), fe_mul(G, H), fe_mul(F, G), fe_mul(E, H)};
}

static ge ge_neg(const ge& p) { return ge{fe_neg(p.X), p.Y, p.Z, fe_neg(p.T)}; }

// scalar (32 bytes LE) * point, plain double-and-add (not constant time for
// verify; signing uses it too — acceptable here, coordinator only verifies).
static ge ge_scalarmult(const uint8_t s[32], const ge& p) {
    ge r = ge_identity();
    for (int i = 255; i >= 0; --i) {
        r = ge_double(r);
        if ((s[i / 8] >> (i % 8)) & 1) r = ge_add(r, p);
    }
    return r;
}

// s*B + k*P with shared doubling and 4-bit fixed windows (Strauss-Shamir;
// variable time — verification only, no secrets involved). ~2x over two
// separate double-and-add ladders; the basepoint window table is cached.
static ge ge_basepoint();

static void ge_window_table(ge out[16], const ge& p) {
    out[0] = ge_identity();
    out[1] = p;
    for (int i = 2; i < 16; ++i) out[i] = ge_add(out[i - 1], p);
}

static ge ge_double_scalarmult_vartime(const uint8_t s[32], const uint8_t k[32], const ge& P) {
    static const ge* TB = [] {
        static ge t[16];
        ge_window_table(t, ge_basepoint());
        return t;
    }();
    ge TP[16];
    ge_window_table(TP, P);
    ge r = ge_identity();
    for (int w = 63; w >= 0; --w) {
        r = ge_double(ge_double(ge_double(ge_double(r))));
        int ns = (s[w / 2] >> ((w & 1) * 4)) & 0xf;
        int nk = (k[w / 2] >> ((w & 1) * 4)) & 0xf;
        if (ns) r = ge_add(r, TB[ns]);
        if (nk) r = ge_add(r, TP[nk]);
    }
    return r;
}

static ge ge_basepoint() {
    static ge B = [] {
        // y = 4/5, x recovered with positive sign... standard base point bytes:
        fe gy = fe_mul(fe{{4, 0, 0, 0, 0}}, fe_invert(fe{{5, 0, 0, 0, 0}}));
        // x^2 = (y^2-1)/(d*y^2+1)
        fe y2 = fe_sq(gy);
        fe u = fe_sub(y2, fe_one());
        fe v = fe_add(fe_mul(fe_d(), y2), fe_one());
        // x = u*v^3 * (u*v^7)^((p-5)/8) — use the standard recover below
        fe v3 = fe_mul(fe_sq(v), v);
        fe v7 = fe_mul(fe_sq(v3), v);
        fe x = fe_mul(fe_mul(u, v3), fe_pow252m3(fe_mul(u, v7)));
        fe vx2 = fe_mul(v, fe_sq(x));
        if (!fe_eq(vx2, u)) {
            // multiply by sqrt(-1) = 2^((p-1)/4)
            fe sqrtm1 = fe_pow(fe{{2, 0, 0, 0, 0}}, [] {
                static uint8_t e[32];
                // (p-1)/4 = (2^255 - 20)/4 = 2^253 - 5
                for (int i = 0; i < 32; ++i) e[i] = 0xff;
                e[0] = 0xfb;
                e[31] = 0x1f;
                return e;
            }(), 254);
            x = fe_mul(x, sqrtm1);
        }
        if (fe_isnegative(x)) x = fe_neg(x);  // base point has even x
        ge b;
        b.X = x;
        b.Y = gy;
        b.Z = fe_one();
        b.T = fe_mul(x, gy);
        return b;
    }();
    return B;
}

static void ge_tobytes(uint8_t out[32], const ge& p) {
    fe zi = fe_invert(p.Z);
    fe x = fe_mul(p.X, zi);
    fe y = fe_mul(p.Y, zi);
    fe_tobytes(out, y);
    if (fe_isnegative(x)) out[31] |= 0x80;
}

static bool ge_frombytes(ge& p, const uint8_t in[32]) {
    fe y = fe_frombytes(in);
    fe y2 = fe_sq(y);
    fe u = fe_sub(y2, fe_one());
    fe v = fe_add(fe_mul(fe_d(), y2), fe_one());
    fe v3 = fe_mul(fe_sq(v), v);
    fe v7 = fe_mul(fe_sq(v3), v);
    fe x = fe_mul(fe_mul(u, v3), fe_pow252m3(fe_mul(u, v7)));
    fe vx2 = fe_mul(v, fe_sq(x));
    if (!fe_eq(vx2, u)) {
        if (fe_eq(vx2, fe_neg(u))) {
            static const fe sqrtm1 = [] {
                uint8_t e[32];
                for (int i = 0; i < 32; ++i) e[i] = 0xff;
                e[0] = 0xfb;
                e[31] = 0x1f;
                return fe_pow(fe{{2, 0, 0, 0, 0}}, e, 254);
            }();
            x = fe_mul(x, sqrtm1);
        } else {
            return false;
        }
    }
    bool sign = (in[31] >> 7) & 1;
    if (fe_iszero(x) && sign) return false;
    if (fe_isnegative(x) != sign) x = fe_neg(x);
    p.X = x;
    p.Y = y;
    p.Z = fe_one();
    p.T = fe_mul(x, y);
    return true;
}

// ----- scalar arithmetic mod L (via BigUint; off the hot path) -----

// eager namespace-scope init (single-threaded .so load): no function-static
// guard on the hot verify path, and no TSAN noise from the guard fast-path
static const BigUint kCurveL = BigUint::from_dec(
    "7237005577332262213973186563042994240857116359379907606001950938285454250989");
static const BigUint& curve_L() { return kCurveL; }

static void sc_reduce_bytes(uint8_t out[32], const uint8_t* in, size_t n) {
    BigUint v = BigUint::from_bytes_le(in, n) % curve_L();
    v.to_bytes_le_fixed(out, 32);
}

// out = (a*b + c) mod L, all 32-byte LE
static void sc_muladd(uint8_t out[32], const uint8_t a[32], const uint8_t b[32],
                      const uint8_t c[32]) {
    BigUint A = BigUint::from_bytes_le(a, 32);
    BigUint B = BigUint::from_bytes_le(b, 32);
    BigUint C = BigUint::from_bytes_le(c, 32);
    BigUint r = (A * B + C) % curve_L();
    r.to_bytes_le_fixed(out, 32);
}

// ----- Ed25519 API -----

void ed25519_keypair_from_seed(uint8_t pk[32], uint8_t sk64[64], const uint8_t seed[32]) {
    auto h = Sha512::hash(seed, 32);
    uint8_t a[32];
    std::memcpy(a, h.data(), 32);
    clamp(a);
    ge A = ge_scalarmult(a, ge_basepoint());
    ge_tobytes(pk, A);
    if (sk64) {
        std::memcpy(sk64, seed, 32);
        std::memcpy(sk64 + 32, pk, 32);
    }
}

void ed25519_sign(uint8_t sig[64], const uint8_t* msg, size_t len, const uint8_t sk64[64]) {
    auto h = Sha512::hash(sk64, 32);  // expand seed
    uint8_t a[32], prefix[32];
    std::memcpy(a, h.data(), 32);
    std::memcpy(prefix, h.data() + 32, 32);
    clamp(a);

    Sha512 hr;
    hr.update(prefix, 32);
    hr.update(msg, len);
    uint8_t rh[64];
    hr.final(rh);
    uint8_t r[32];
    sc_reduce_bytes(r, rh, 64);

    ge R = ge_scalarmult(r, ge_basepoint());
    uint8_t Rb[32];
    ge_tobytes(Rb, R);

    Sha512 hk;
    hk.update(Rb, 32);
    hk.update(sk64 + 32, 32);  // public key
    hk.update(msg, len);
    uint8_t kh[64];
    hk.final(kh);
    uint8_t k[32];
    sc_reduce_bytes(k, kh, 64);

    uint8_t s[32];
    sc_muladd(s, k, a, r);

    std::memcpy(sig, Rb, 32);
    std::memcpy(sig + 32, s, 32);
}

bool ed25519_verify(const uint8_t sig[64], const uint8_t* msg, size_t len, const uint8_t pk[32]) {
    // s < L
    BigUint s = BigUint::from_bytes_le(sig + 32, 32);
    if (!(s < curve_L())) return false;

    ge A;
    if (!ge_frombytes(A, pk)) return false;

    Sha512 hk;
    hk.update(sig, 32);
    hk.update(pk, 32);
    hk.update(msg, len);
    uint8_t kh[64];
    hk.final(kh);
    uint8_t k[32];
    sc_reduce_bytes(k, kh, 64);

    // check [s]B = R + [k]A  <=>  [s]B + [k](-A) == R
    ge R = ge_double_scalarmult_vartime(sig + 32, k, ge_neg(A));
    uint8_t Rb[32];
    ge_tobytes(Rb, R);
    return std::memcmp(Rb, sig, 32) == 0;
}

void ed25519_sk_to_x25519(uint8_t x_sk[32], const uint8_t sk64[64]) {
    auto h = Sha512::hash(sk64, 32);
    std::memcpy(x_sk, h.data(), 32);
    x_sk[0] &= 248;
    x_sk[31] &= 127;
    x_sk[31] |= 64;
}

bool ed25519_pk_to_x25519(uint8_t x_pk[32], const uint8_t pk[32]) {
    ge A;
    if (!ge_frombytes(A, pk)) return false;
    // u = (1+y)/(1-y)
    fe zi = fe_invert(A.Z);
    fe y = fe_mul(A.Y, zi);
    fe num = fe_add(fe_one(), y);
    fe den = fe_sub(fe_one(), y);
    fe u = fe_mul(num, fe_invert(den));
    fe_tobytes(x_pk, u);
    return true;
}

}  // namespace xaynet::crypto
