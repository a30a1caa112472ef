#include "bigint.h"

#include <algorithm>
#include <cmath>
#include <random>

namespace xaynet {

using u128 = unsigned __int128;

void BigUint::normalize() {
    while (!d.empty() && d.back() == 0) d.pop_back();
}

BigUint BigUint::from_bytes_le(const uint8_t* p, size_t n) {
    BigUint r;
    r.d.resize((n + 7) / 8, 0);
    for (size_t i = 0; i < n; ++i) r.d[i / 8] |= uint64_t(p[i]) << (8 * (i % 8));
    r.normalize();
    return r;
}

Bytes BigUint::to_bytes_le() const {
    Bytes out;
    if (d.empty()) return out;
    out.resize(d.size() * 8, 0);
    for (size_t i = 0; i < d.size(); ++i) store64_le(out.data() + 8 * i, d[i]);
    while (!out.empty() && out.back() == 0) out.pop_back();
    return out;
}

void BigUint::to_bytes_le_fixed(uint8_t* out, size_t n) const {
    std::memset(out, 0, n);
    for (size_t i = 0; i < n; ++i) {
        size_t limb = i / 8;
        if (limb >= d.size()) break;
        out[i] = uint8_t(d[limb] >> (8 * (i % 8)));
    }
}

size_t BigUint::bits() const {
    if (d.empty()) return 0;
    size_t b = (d.size() - 1) * 64;
    uint64_t top = d.back();
    while (top) {
        b += 1;
        top >>= 1;
    }
    return b;
}

int BigUint::cmp(const BigUint& a, const BigUint& b) {
    if (a.d.size() != b.d.size()) return a.d.size() < b.d.size() ? -1 : 1;
    for (size_t i = a.d.size(); i-- > 0;) {
        if (a.d[i] != b.d[i]) return a.d[i] < b.d[i] ? -1 : 1;
    }
    return 0;
}

BigUint BigUint::operator+(const BigUint& o) const {
    BigUint r;
    size_t n = std::max(d.size(), o.d.size());
    r.d.resize(n + 1, 0);
    u128 carry = 0;
    for (size_t i = 0; i < n; ++i) {
        u128 s = carry;
        if (i < d.size()) s += d[i];
        if (i < o.d.size()) s += o.d[i];
        r.d[i] = uint64_t(s);
        carry = s >> 64;
    }
    r.d[n] = uint64_t(carry);
    r.normalize();
    return r;
}

BigUint BigUint::operator-(const BigUint& o) const {
    BigUint r;
    r.d.resize(d.size(), 0);
    unsigned __int128 borrow = 0;
    for (size_t i = 0; i < d.size(); ++i) {
        u128 sub = borrow;
        if (i < o.d.size()) sub += o.d[i];
        u128 cur = d[i];
        if (cur >= sub) {
            r.d[i] = uint64_t(cur - sub);
            borrow = 0;
        } else {
            r.d[i] = uint64_t((u128(1) << 64) + cur - sub);
            borrow = 1;
        }
    }
    if (borrow) throw std::runtime_error("BigUint underflow");
    r.normalize();
    return r;
}

BigUint BigUint::operator*(const BigUint& o) const {
    if (d.empty() || o.d.empty()) return BigUint();
    BigUint r;
    r.d.assign(d.size() + o.d.size(), 0);
    for (size_t i = 0; i < d.size(); ++i) {
        u128 carry = 0;
        for (size_t j = 0; j < o.d.size(); ++j) {
            u128 cur = (u128)d[i] * o.d[j] + r.d[i + j] + carry;
            r.d[i + j] = uint64_t(cur);
            carry = cur >> 64;
        }
        size_t k = i + o.d.size();
        while (carry) {
            u128 cur = (u128)r.d[k] + carry;
            r.d[k] = uint64_t(cur);
            carry = cur >> 64;
            ++k;
        }
    }
    r.normalize();
    return r;
}

BigUint BigUint::operator<<(size_t n) const {
    if (d.empty()) return BigUint();
    size_t limbs = n / 64, bits = n % 64;
    BigUint r;
    r.d.assign(d.size() + limbs + 1, 0);
    for (size_t i = 0; i < d.size(); ++i) {
        r.d[i + limbs] |= bits ? (d[i] << bits) : d[i];
        if (bits) r.d[i + limbs + 1] |= d[i] >> (64 - bits);
    }
    r.normalize();
    return r;
}

BigUint BigUint::operator>>(size_t n) const {
    size_t limbs = n / 64, bits = n % 64;
    if (limbs >= d.size()) return BigUint();
    BigUint r;
    r.d.assign(d.size() - limbs, 0);
    for (size_t i = 0; i < r.d.size(); ++i) {
        r.d[i] = d[i + limbs] >> bits;
        if (bits && i + limbs + 1 < d.size()) r.d[i] |= d[i + limbs + 1] << (64 - bits);
    }
    r.normalize();
    return r;
}

// Simple binary long division (adequate: orders are <= 22 limbs and divmod is
// off the hot path everywhere in this codebase).
void BigUint::divmod(const BigUint& a, const BigUint& b, BigUint& q, BigUint& r) {
    if (b.is_zero()) throw std::runtime_error("division by zero");
    if (cmp(a, b) < 0) {
        q = BigUint();
        r = a;
        return;
    }
    size_t shift = a.bits() - b.bits();
    BigUint cur = b << shift;
    q = BigUint();
    q.d.assign((shift + 64) / 64, 0);
    r = a;
    for (size_t i = shift + 1; i-- > 0;) {
        if (cmp(r, cur) >= 0) {
            r = r - cur;
            q.d[i / 64] |= uint64_t(1) << (i % 64);
        }
        cur = cur >> 1;
    }
    q.normalize();
}

BigUint BigUint::operator%(const BigUint& o) const {
    BigUint q, r;
    divmod(*this, o, q, r);
    return r;
}

BigUint BigUint::operator/(const BigUint& o) const {
    BigUint q, r;
    divmod(*this, o, q, r);
    return q;
}

BigUint BigUint::from_dec(const std::string& s) {
    BigUint r;
    BigUint ten(10);
    for (char c : s) {
        if (c == '_' || c == ',') continue;
        if (c < '0' || c > '9') throw std::runtime_error("bad decimal digit");
        r = r * ten + BigUint(uint64_t(c - '0'));
    }
    return r;
}

std::string BigUint::to_dec() const {
    if (is_zero()) return "0";
    // repeated division by 10^19
    BigUint ten19(10000000000000000000ULL);
    BigUint cur = *this;
    std::vector<uint64_t> chunks;
    while (!cur.is_zero()) {
        BigUint q, r;
        divmod(cur, ten19, q, r);
        chunks.push_back(r.low_u64());
        cur = q;
    }
    std::string out = std::to_string(chunks.back());
    for (size_t i = chunks.size() - 1; i-- > 0;) {
        std::string part = std::to_string(chunks[i]);
        out += std::string(19 - part.size(), '0') + part;
    }
    return out;
}

BigUint BigUint::pow10(unsigned n) {
    BigUint r(1), ten(10);
    for (unsigned i = 0; i < n; ++i) r = r * ten;
    return r;
}

BigUint BigUint::pow2(size_t n) { return BigUint(1) << n; }

BigUint BigUint::gcd(BigUint a, BigUint b) {
    while (!b.is_zero()) {
        BigUint r = a % b;
        a = b;
        b = r;
    }
    return a;
}

BigUint BigUint::modpow(const BigUint& base, const BigUint& exp, const BigUint& mod) {
    BigUint result(1);
    BigUint b = base % mod;
    size_t n = exp.bits();
    for (size_t i = 0; i < n; ++i) {
        if ((exp.d[i / 64] >> (i % 64)) & 1) result = (result * b) % mod;
        b = (b * b) % mod;
    }
    return result;
}

double BigUint::to_double() const {
    double r = 0;
    for (size_t i = d.size(); i-- > 0;) r = r * 18446744073709551616.0 + double(d[i]);
    return r;
}

bool is_probable_prime(const BigUint& n, int rounds) {
    if (n.is_zero()) return false;
    if (n == BigUint(1)) return false;
    if (n == BigUint(2) || n == BigUint(3)) return true;
    if ((n.d[0] & 1) == 0) return false;

    BigUint n1 = n - BigUint(1);
    BigUint d = n1;
    size_t s = 0;
    while ((d.d[0] & 1) == 0) {
        d = d >> 1;
        s += 1;
    }
    std::mt19937_64 rng(0x9e3779b97f4a7c15ULL);  // deterministic witnesses
    for (int i = 0; i < rounds; ++i) {
        // witness in [2, n-2]
        BigUint a;
        a.d.resize(n.d.size());
        for (auto& limb : a.d) limb = rng();
        a.normalize();
        a = a % n1;
        if (BigUint::cmp(a, BigUint(2)) < 0) a = BigUint(2);
        BigUint x = BigUint::modpow(a, d, n);
        if (x == BigUint(1) || x == n1) continue;
        bool composite = true;
        for (size_t r = 1; r < s; ++r) {
            x = (x * x) % n;
            if (x == n1) {
                composite = false;
                break;
            }
        }
        if (composite) return false;
    }
    return true;
}

// ------------------------------------------------------------------ BigInt

BigInt::BigInt(int64_t v) {
    if (v < 0) {
        neg = true;
        mag = BigUint(uint64_t(-(v + 1)) + 1);
    } else {
        mag = BigUint(uint64_t(v));
    }
}

int BigInt::cmp(const BigInt& a, const BigInt& b) {
    if (a.neg != b.neg) return a.neg ? -1 : 1;
    int c = BigUint::cmp(a.mag, b.mag);
    return a.neg ? -c : c;
}

BigInt BigInt::operator+(const BigInt& o) const {
    if (neg == o.neg) return BigInt(mag + o.mag, neg);
    int c = BigUint::cmp(mag, o.mag);
    if (c == 0) return BigInt();
    if (c > 0) return BigInt(mag - o.mag, neg);
    return BigInt(o.mag - mag, o.neg);
}

BigInt BigInt::operator-(const BigInt& o) const { return *this + BigInt(o.mag, !o.neg); }

BigInt BigInt::operator*(const BigInt& o) const { return BigInt(mag * o.mag, neg != o.neg); }

std::string BigInt::to_dec() const { return (neg ? "-" : "") + mag.to_dec(); }

// ---------------------------------------------------------------- Rational

Rational::Rational(BigInt n, BigUint d) {
    if (d.is_zero()) throw std::runtime_error("rational with zero denominator");
    BigUint g = BigUint::gcd(n.mag, d);
    if (!g.is_zero() && !(g == BigUint(1))) {
        n.mag = n.mag / g;
        d = d / g;
    }
    if (n.mag.is_zero()) {
        n.neg = false;
        d = BigUint(1);
    }
    numer = std::move(n);
    denom = std::move(d);
}

Rational Rational::from_double(double f) {
    if (!std::isfinite(f)) throw std::runtime_error("from_double: non-finite");
    if (f == 0.0) return Rational();
    int exp;
    double m = std::frexp(f, &exp);  // f = m * 2^exp, 0.5 <= |m| < 1
    // scale mantissa to an integer: m * 2^53 is integral for doubles
    int64_t mi = int64_t(std::ldexp(m, 53));
    exp -= 53;
    BigInt n(mi);
    if (exp >= 0) return Rational(n * BigInt(BigUint::pow2(size_t(exp))), BigUint(1));
    return Rational(n, BigUint::pow2(size_t(-exp)));
}

int Rational::cmp(const Rational& a, const Rational& b) {
    // a.n/a.d vs b.n/b.d  <=>  a.n*b.d vs b.n*a.d (denoms positive)
    return BigInt::cmp(a.numer * BigInt(b.denom), b.numer * BigInt(a.denom));
}

Rational Rational::operator+(const Rational& o) const {
    return Rational(numer * BigInt(o.denom) + o.numer * BigInt(denom), denom * o.denom);
}

Rational Rational::operator-(const Rational& o) const {
    return Rational(numer * BigInt(o.denom) - o.numer * BigInt(denom), denom * o.denom);
}

Rational Rational::operator*(const Rational& o) const {
    return Rational(numer * o.numer, denom * o.denom);
}

Rational Rational::operator/(const Rational& o) const {
    if (o.numer.is_zero()) throw std::runtime_error("rational division by zero");
    BigInt n(numer.mag * o.denom, false);
    BigUint d = denom * o.numer.mag;
    n.neg = !n.mag.is_zero() && (numer.neg != o.numer.neg);
    return Rational(n, d);
}

BigInt Rational::trunc() const {
    BigUint q = numer.mag / denom;
    return BigInt(q, numer.neg);
}

double Rational::to_double() const {
    // best-effort: shift to keep ~64 bits of precision
    double n = numer.mag.to_double();
    double d = denom.to_double();
    if (std::isinf(n) || std::isinf(d)) {
        // scale both down by 2^k
        size_t nb = numer.mag.bits(), db = denom.bits();
        size_t k = (nb > db ? nb : db) > 900 ? (nb > db ? nb : db) - 900 : 0;
        n = (numer.mag >> k).to_double();
        d = (denom >> k).to_double();
    }
    double r = n / d;
    return numer.neg ? -r : r;
}

std::string Rational::to_string() const { return numer.to_dec() + "/" + denom.to_dec(); }

}  // namespace xaynet
