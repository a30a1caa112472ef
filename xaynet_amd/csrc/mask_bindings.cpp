// Bindings for mask config / masking / aggregation (CPU oracle path).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "mask/masking.h"

namespace py = pybind11;
using namespace xaynet;
using namespace xaynet::mask;

static MaskConfig cfg_from_ints(int g, int d, int b, int m) {
    uint8_t raw[4] = {uint8_t(g), uint8_t(d), uint8_t(b), uint8_t(m)};
    auto c = MaskConfig::from_bytes(raw);
    if (!c) throw std::runtime_error("invalid mask config");
    return *c;
}

void bind_mask(py::module_& m) {
    auto mm = m.def_submodule("mask");

    py::class_<MaskConfig>(mm, "MaskConfig")
        .def(py::init([](int g, int d, int b, int mo) { return cfg_from_ints(g, d, b, mo); }),
             py::arg("group"), py::arg("dtype"), py::arg("bound"), py::arg("model"))
        .def_property_readonly("group", [](const MaskConfig& c) { return int(c.group); })
        .def_property_readonly("dtype", [](const MaskConfig& c) { return int(c.dtype); })
        .def_property_readonly("bound", [](const MaskConfig& c) { return int(c.bound); })
        .def_property_readonly("model", [](const MaskConfig& c) { return int(c.model); })
        .def_property_readonly("bytes_per_number",
                               [](const MaskConfig& c) { return c.info().bpn; })
        .def_property_readonly("prng_nbytes", [](const MaskConfig& c) { return c.info().prng_nbytes; })
        .def_property_readonly("order", [](const MaskConfig& c) { return c.info().order.to_dec(); })
        .def_property_readonly("order_fits_u64",
                               [](const MaskConfig& c) { return c.info().order_fits_u64; })
        .def_property_readonly("max_nb_models",
                               [](const MaskConfig& c) { return c.info().max_nb_models; })
        .def("__eq__", [](const MaskConfig& a, const MaskConfig& b) { return a == b; })
        .def("to_bytes", [](const MaskConfig& c) {
            uint8_t raw[4];
            c.write_bytes(raw);
            return py::bytes(reinterpret_cast<char*>(raw), 4);
        })
        .def_static("from_bytes", [](py::bytes b) {
            std::string s = b;
            if (s.size() != 4) throw std::runtime_error("config must be 4 bytes");
            auto c = MaskConfig::from_bytes(reinterpret_cast<const uint8_t*>(s.data()));
            if (!c) throw std::runtime_error("invalid mask config bytes");
            return *c;
        });

    py::class_<MaskConfigPair>(mm, "MaskConfigPair")
        .def(py::init([](const MaskConfig& v, const MaskConfig& u) {
                 return MaskConfigPair{v, u};
             }),
             py::arg("vect"), py::arg("unit"))
        .def_readonly("vect", &MaskConfigPair::vect)
        .def_readonly("unit", &MaskConfigPair::unit)
        .def("__eq__", [](const MaskConfigPair& a, const MaskConfigPair& b) { return a == b; });

    py::class_<MaskObject>(mm, "MaskObject")
        .def_property_readonly("count", [](const MaskObject& o) { return o.vect.count; })
        .def_property_readonly("config", [](const MaskObject& o) { return o.config(); })
        .def("is_valid", &MaskObject::is_valid)
        .def("serialize", [](const MaskObject& o) {
            Bytes b = o.serialize();
            return py::bytes(reinterpret_cast<char*>(b.data()), b.size());
        })
        .def_static("deserialize", [](py::bytes data) {
            std::string s = data;
            auto o = MaskObject::deserialize(reinterpret_cast<const uint8_t*>(s.data()), s.size(),
                                             nullptr);
            if (!o) throw std::runtime_error("invalid mask object bytes");
            return *o;
        })
        .def("element", [](const MaskObject& o, size_t i) { return o.vect.element(i).to_dec(); })
        .def_property_readonly("unit_value", [](const MaskObject& o) { return o.unit.value().to_dec(); })
        .def_property_readonly("vect_bytes", [](const MaskObject& o) {
            return py::bytes(reinterpret_cast<const char*>(o.vect.data.data()),
                             o.vect.data.size());
        });

    py::class_<Scalar>(mm, "Scalar")
        .def(py::init<uint64_t, uint64_t>(), py::arg("numer"), py::arg("denom"))
        .def_static("unit", []() { return Scalar(); });

    // The unit (scalar) draw that precedes the vector draws in the mask
    // stream: returns (value_dec, keystream_words_consumed). The GPU K1
    // expander starts its vector draws at that word offset.
    mm.def("unit_draw", [](py::bytes seed, const MaskConfig& cfg) {
        std::string s = seed;
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        MaskPrng prng(reinterpret_cast<const uint8_t*>(s.data()));
        BigUint v = prng.generate_integer(cfg.info());
        return py::make_tuple(v.to_dec(), prng.words_consumed());
    });

    mm.def("derive_mask", [](py::bytes seed, size_t len, const MaskConfigPair& cfg) {
        std::string s = seed;
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        return derive_mask(reinterpret_cast<const uint8_t*>(s.data()), len, cfg);
    });

    // exact-rational oracle path (bypasses the fast typed masker; used by
    // tests to assert bit-equality of the fast path)
    mm.def("mask_model_oracle", [](py::bytes seed, const Scalar& scalar, py::array weights,
                                   const MaskConfigPair& cfg) {
        std::string s = seed;
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        const uint8_t* sp = reinterpret_cast<const uint8_t*>(s.data());
        auto buf = weights.request();
        if (buf.ndim != 1) throw std::runtime_error("weights must be 1-D");
        size_t n = size_t(buf.shape[0]);
        RationalModel m;
        switch (cfg.vect.dtype) {
            case DataType::F32:
                m = model_from_f32(weights.cast<py::array_t<float>>().data(), n);
                break;
            case DataType::F64:
                m = model_from_f64(weights.cast<py::array_t<double>>().data(), n);
                break;
            case DataType::I32:
                m = model_from_i32(weights.cast<py::array_t<int32_t>>().data(), n);
                break;
            case DataType::I64:
                m = model_from_i64(weights.cast<py::array_t<int64_t>>().data(), n);
                break;
        }
        return mask_model(sp, scalar, m, cfg);
    });

    mm.def("mask_model", [](py::bytes seed, const Scalar& scalar, py::array weights,
                            const MaskConfigPair& cfg) {
        std::string s = seed;
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        const uint8_t* sp = reinterpret_cast<const uint8_t*>(s.data());
        auto buf = weights.request();
        if (buf.ndim != 1) throw std::runtime_error("weights must be 1-D");
        size_t n = size_t(buf.shape[0]);
        switch (cfg.vect.dtype) {
            case DataType::F32: {
                auto a = weights.cast<py::array_t<float>>();
                return mask_f32(sp, scalar, a.data(), n, cfg);
            }
            case DataType::F64: {
                auto a = weights.cast<py::array_t<double>>();
                return mask_f64(sp, scalar, a.data(), n, cfg);
            }
            case DataType::I32: {
                auto a = weights.cast<py::array_t<int32_t>>();
                return mask_i32(sp, scalar, a.data(), n, cfg);
            }
            case DataType::I64: {
                auto a = weights.cast<py::array_t<int64_t>>();
                return mask_i64(sp, scalar, a.data(), n, cfg);
            }
        }
        throw std::runtime_error("unreachable");
    });

    py::class_<Aggregation>(mm, "Aggregation")
        .def(py::init<const MaskConfigPair&, size_t>(), py::arg("config"), py::arg("object_size"))
        .def_property_readonly("nb_models", &Aggregation::nb_models)
        .def_property_readonly("object", &Aggregation::object)
        .def("validate_aggregation",
             [](const Aggregation& a, const MaskObject& o) {
                 return int(a.validate_aggregation(o));
             })
        .def("aggregate", &Aggregation::aggregate)
        .def("validate_unmasking",
             [](const Aggregation& a, const MaskObject& o) { return int(a.validate_unmasking(o)); })
        .def("set", &Aggregation::set, py::arg("object"), py::arg("nb_models"))
        .def("unmask", [](const Aggregation& a, const MaskObject& mask_obj) -> py::object {
            RationalModel rm = a.unmask(mask_obj);
            switch (a.config().vect.dtype) {
                case DataType::F32: {
                    auto v = model_to_f32(rm);
                    return py::array_t<float>(py::ssize_t(v.size()), v.data());
                }
                case DataType::F64: {
                    auto v = model_to_f64(rm);
                    return py::array_t<double>(py::ssize_t(v.size()), v.data());
                }
                case DataType::I32: {
                    auto v = model_to_i32(rm);
                    return py::array_t<int32_t>(py::ssize_t(v.size()), v.data());
                }
                case DataType::I64: {
                    auto v = model_to_i64(rm);
                    return py::array_t<int64_t>(py::ssize_t(v.size()), v.data());
                }
            }
            throw std::runtime_error("unreachable");
        });
}
