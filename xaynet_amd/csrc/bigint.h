// Minimal arbitrary-precision unsigned/signed integers and rationals.
//
// Used for: mask-config group orders (up to 173 bytes, reference
// rust/xaynet-core/src/mask/config/mod.rs), the CPU masking oracle, Ed25519
// scalar arithmetic, and the exact-rational Model representation
// (reference Model = Vec<Ratio<BigInt>>, rust/xaynet-core/src/mask/model.rs:25).
//
// Not performance-critical: the GPU/limb-plane paths never touch these.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "common.h"

namespace xaynet {

struct BigUint {
    // little-endian 64-bit limbs, normalized (no trailing zero limbs).
    std::vector<uint64_t> d;

    BigUint() = default;
    explicit BigUint(uint64_t v) {
        if (v) d.push_back(v);
    }

    static BigUint from_bytes_le(const uint8_t* p, size_t n);
    static BigUint from_dec(const std::string& s);
    Bytes to_bytes_le() const;                 // minimal length (empty for 0)
    void to_bytes_le_fixed(uint8_t* out, size_t n) const;  // zero-padded, truncates high
    std::string to_dec() const;

    bool is_zero() const { return d.empty(); }
    size_t bits() const;
    void normalize();

    // comparison: -1, 0, 1
    static int cmp(const BigUint& a, const BigUint& b);
    bool operator<(const BigUint& o) const { return cmp(*this, o) < 0; }
    bool operator<=(const BigUint& o) const { return cmp(*this, o) <= 0; }
    bool operator==(const BigUint& o) const { return d == o.d; }
    bool operator!=(const BigUint& o) const { return d != o.d; }
    bool operator>(const BigUint& o) const { return cmp(*this, o) > 0; }
    bool operator>=(const BigUint& o) const { return cmp(*this, o) >= 0; }

    BigUint operator+(const BigUint& o) const;
    BigUint operator-(const BigUint& o) const;  // requires *this >= o
    BigUint operator*(const BigUint& o) const;
    BigUint operator<<(size_t n) const;
    BigUint operator>>(size_t n) const;
    BigUint operator%(const BigUint& o) const;
    BigUint operator/(const BigUint& o) const;

    // quotient and remainder
    static void divmod(const BigUint& a, const BigUint& b, BigUint& q, BigUint& r);
    static BigUint pow10(unsigned n);
    static BigUint pow2(size_t n);
    static BigUint gcd(BigUint a, BigUint b);
    static BigUint modpow(const BigUint& base, const BigUint& exp, const BigUint& mod);

    uint64_t low_u64() const { return d.empty() ? 0 : d[0]; }
    double to_double() const;  // best-effort (may overflow to inf)
};

struct BigInt {
    bool neg = false;  // sign; zero is always non-negative
    BigUint mag;

    BigInt() = default;
    BigInt(const BigUint& m, bool n = false) : neg(n && !m.is_zero()), mag(m) {}
    explicit BigInt(int64_t v);

    static int cmp(const BigInt& a, const BigInt& b);
    BigInt operator+(const BigInt& o) const;
    BigInt operator-(const BigInt& o) const;
    BigInt operator*(const BigInt& o) const;
    BigInt operator-() const { return BigInt(mag, !neg); }
    bool operator==(const BigInt& o) const { return neg == o.neg && mag == o.mag; }
    bool is_zero() const { return mag.is_zero(); }
    std::string to_dec() const;
};

// Exact rational with reduced representation, denominator > 0
// (mirrors num::rational::Ratio<BigInt> semantics incl. serde layout).
struct Rational {
    BigInt numer;
    BigUint denom;  // always positive; 1 for integers

    Rational() : denom(BigUint(1)) {}
    Rational(BigInt n, BigUint d);  // reduces
    static Rational from_integer(BigInt n) {
        Rational r;
        r.numer = std::move(n);
        r.denom = BigUint(1);
        return r;
    }
    // Exact binary expansion of a finite double (mirrors Ratio::from_float).
    static Rational from_double(double f);

    static int cmp(const Rational& a, const Rational& b);
    Rational operator+(const Rational& o) const;
    Rational operator-(const Rational& o) const;
    Rational operator*(const Rational& o) const;
    Rational operator/(const Rational& o) const;
    bool operator==(const Rational& o) const { return numer == o.numer && denom == o.denom; }

    // truncation toward zero (num::Ratio::to_integer)
    BigInt trunc() const;
    double to_double() const;
    std::string to_string() const;
};

// Miller-Rabin probabilistic primality (deterministic enough for table checks).
bool is_probable_prime(const BigUint& n, int rounds = 40);

}  // namespace xaynet
