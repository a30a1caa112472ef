// pybind11 bindings for the xaynet_amd native core (_core).
//
// The Python surface mirrors the reference's bindings/python/xaynet_sdk
// at the top level (see xaynet_amd/sdk); these low-level bindings are the
// building blocks (crypto, masking, wire codecs, coordinator/SDK state
// machines).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "bigint.h"
#include "common.h"
#include "crypto/box.h"
#include "crypto/blake2b.h"
#include "crypto/chacha.h"
#include "crypto/curve25519.h"
#include "crypto/salsa.h"
#include "crypto/poly1305.h"
#include "crypto/sha2.h"

namespace py = pybind11;
using namespace xaynet;
using namespace xaynet::crypto;

static py::bytes to_pybytes(const uint8_t* p, size_t n) {
    return py::bytes(reinterpret_cast<const char*>(p), n);
}

static Bytes from_pybytes(py::bytes b) {
    std::string s = b;
    return Bytes(s.begin(), s.end());
}

void bind_mask(py::module_& m);      // mask_bindings.cpp
void bind_message(py::module_& m);   // message_bindings.cpp
void bind_coordinator(py::module_& m);  // coordinator_bindings.cpp
void bind_sdk(py::module_& m);       // sdk_bindings.cpp
void bind_rest(py::module_& m);      // rest_bindings.cpp

PYBIND11_MODULE(_core, m) {
    m.doc() = "xaynet_amd native core (protocol, crypto, masking, coordinator)";

    // ---- crypto submodule ----
    auto c = m.def_submodule("crypto");

    c.def("sha256", [](py::bytes data) {
        Bytes d = from_pybytes(data);
        auto h = Sha256::hash(d.data(), d.size());
        return to_pybytes(h.data(), 32);
    });
    c.def("sha512", [](py::bytes data) {
        Bytes d = from_pybytes(data);
        auto h = Sha512::hash(d.data(), d.size());
        return to_pybytes(h.data(), 64);
    });
    c.def("blake2b", [](py::bytes data, size_t outlen) {
        Bytes d = from_pybytes(data);
        Bytes out(outlen);
        blake2b(out.data(), outlen, d.data(), d.size());
        return to_pybytes(out.data(), outlen);
    });
    c.def("chacha20_keystream", [](py::bytes seed, size_t n) {
        Bytes s = from_pybytes(seed);
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        ChaChaRng rng(s.data());
        Bytes out(n);
        rng.fill_bytes(out.data(), n);
        return to_pybytes(out.data(), n);
    });

    c.def("x25519_base", [](py::bytes sk) {
        Bytes s = from_pybytes(sk);
        uint8_t pk[32];
        x25519_base(pk, s.data());
        return to_pybytes(pk, 32);
    });
    c.def("x25519", [](py::bytes sk, py::bytes pk) {
        Bytes s = from_pybytes(sk), p = from_pybytes(pk);
        uint8_t out[32];
        x25519(out, s.data(), p.data());
        return to_pybytes(out, 32);
    });

    c.def("sign_keypair_from_seed", [](py::bytes seed) {
        Bytes s = from_pybytes(seed);
        uint8_t pk[32], sk[64];
        ed25519_keypair_from_seed(pk, sk, s.data());
        return py::make_tuple(to_pybytes(pk, 32), to_pybytes(sk, 64));
    });
    c.def("sign_keypair", []() {
        uint8_t seed[32], pk[32], sk[64];
        randombytes(seed, 32);
        ed25519_keypair_from_seed(pk, sk, seed);
        return py::make_tuple(to_pybytes(pk, 32), to_pybytes(sk, 64));
    });
    c.def("sign_detached", [](py::bytes msg, py::bytes sk) {
        Bytes m_ = from_pybytes(msg), s = from_pybytes(sk);
        uint8_t sig[64];
        ed25519_sign(sig, m_.data(), m_.size(), s.data());
        return to_pybytes(sig, 64);
    });
    c.def("verify_detached", [](py::bytes sig, py::bytes msg, py::bytes pk) {
        Bytes g = from_pybytes(sig), m_ = from_pybytes(msg), p = from_pybytes(pk);
        if (g.size() != 64 || p.size() != 32) return false;
        return ed25519_verify(g.data(), m_.data(), m_.size(), p.data());
    });

    c.def("box_keypair", []() {
        uint8_t pk[32], sk[32];
        box_keypair(pk, sk);
        return py::make_tuple(to_pybytes(pk, 32), to_pybytes(sk, 32));
    });
    c.def("box_seed_keypair", [](py::bytes seed) {
        Bytes s = from_pybytes(seed);
        uint8_t pk[32], sk[32];
        box_seed_keypair(pk, sk, s.data());
        return py::make_tuple(to_pybytes(pk, 32), to_pybytes(sk, 32));
    });
    c.def("sealbox_seal", [](py::buffer msg, py::bytes pk) {
        py::buffer_info mi = msg.request();
        Bytes p = from_pybytes(pk);
        PyObject* raw = PyBytes_FromStringAndSize(nullptr, Py_ssize_t(mi.size + SEAL_BYTES));
        if (!raw) throw py::error_already_set();
        py::object holder = py::reinterpret_steal<py::object>(raw);
        auto* dst = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(raw));
        {
            py::gil_scoped_release rel;
            sealbox_seal_into(dst, static_cast<const uint8_t*>(mi.ptr), size_t(mi.size),
                              p.data());
        }
        return holder;
    });
    c.def("sealbox_open", [](py::buffer cipher, py::bytes pk, py::bytes sk) -> py::object {
        // zero-copy ciphertext view + GIL released around the decrypt:
        // update bodies are hundreds of MB and many REST workers decrypt
        // concurrently
        py::buffer_info ci = cipher.request();
        Bytes p = from_pybytes(pk), s = from_pybytes(sk);
        if (size_t(ci.size) < SEAL_BYTES) return py::none();
        // decrypt straight into the final (uninitialized) bytes object:
        // avoids the vector zero-fill pass and the bytes-copy pass, which
        // at multi-hundred-MB update bodies cost as much as the crypto
        PyObject* raw = PyBytes_FromStringAndSize(nullptr, Py_ssize_t(ci.size - SEAL_BYTES));
        if (!raw) throw py::error_already_set();
        py::object holder = py::reinterpret_steal<py::object>(raw);
        auto* dst = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(raw));
        bool ok;
        {
            py::gil_scoped_release rel;
            ok = sealbox_open_into(dst, static_cast<const uint8_t*>(ci.ptr), size_t(ci.size),
                                   p.data(), s.data());
        }
        if (!ok) return py::none();
        return holder;
    });
    c.def("randombytes", [](size_t n) {
        Bytes out(n);
        randombytes(out.data(), n);
        return to_pybytes(out.data(), n);
    });
    c.def("is_eligible", [](py::bytes sig, double threshold) {
        // int(sha256(sig) as LE) / (2^256 - 1) <= threshold
        // (reference rust/xaynet-core/src/crypto/sign.rs:186-200)
        if (threshold < 0.0) return false;
        if (threshold > 1.0) return true;
        Bytes g = from_pybytes(sig);
        auto h = Sha256::hash(g.data(), g.size());
        BigUint numer = BigUint::from_bytes_le(h.data(), 32);
        Bytes ff(32, 0xff);
        BigUint denom = BigUint::from_bytes_le(ff.data(), 32);
        Rational lhs(BigInt(numer), denom);
        Rational rhs = Rational::from_double(threshold);
        return Rational::cmp(lhs, rhs) <= 0;
    });

    // ---- bigint helpers (mostly for tests) ----
    auto b = m.def_submodule("bigint");
    b.def("dec_roundtrip", [](const std::string& s) { return BigUint::from_dec(s).to_dec(); });
    b.def("mod_dec", [](const std::string& a, const std::string& b_) {
        return (BigUint::from_dec(a) % BigUint::from_dec(b_)).to_dec();
    });
    b.def("is_probable_prime", [](const std::string& s) {
        return is_probable_prime(BigUint::from_dec(s));
    });

    bind_mask(m);
    bind_message(m);
    bind_coordinator(m);
    bind_sdk(m);
    bind_rest(m);
}
