// Bindings for the participant SDK: C++ state machine + pluggable transport.
#include <pybind11/functional.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "crypto/curve25519.h"
#include "sdk/participant.h"

namespace py = pybind11;
using namespace xaynet;
using namespace xaynet::sdk;

static py::bytes pyb(const Bytes& b) {
    return py::bytes(reinterpret_cast<const char*>(b.data()), b.size());
}

static Bytes frompy(py::bytes b) {
    std::string s = b;
    return Bytes(s.begin(), s.end());
}

// Transport bridge for Python HTTP clients: Python supplies raw response
// bodies (bincode) from GET endpoints and forwards POSTs; the C++ side
// decodes. Callbacks run holding the GIL (the tick caller releases it only
// around pure-C++ work).
class PyTransportClient : public XaynetClient {
  public:
    // get(path: str, pk: bytes|None) -> Optional[bytes]; post(body) -> bool
    PyTransportClient(py::function get, py::function post)
        : get_(std::move(get)), post_(std::move(post)) {}

    std::optional<RoundParameters> get_round_params() override {
        auto body = fetch("params", nullptr);
        if (!body) return std::nullopt;
        return bincode::decode_round_parameters(body->data(), body->size());
    }
    std::optional<SumDict> get_sums() override {
        auto body = fetch("sums", nullptr);
        if (!body) return std::nullopt;
        auto d = bincode::decode_option_sum_dict(body->data(), body->size());
        if (!d || !*d) return std::nullopt;
        return **d;
    }
    std::optional<UpdateSeedDict> get_seeds(const Key32& pk) override {
        auto body = fetch("seeds", &pk);
        if (!body) return std::nullopt;
        auto d = bincode::decode_option_update_seed_dict(body->data(), body->size());
        if (!d || !*d) return std::nullopt;
        return **d;
    }
    std::optional<Bytes> get_model_bincode() override {
        auto body = fetch("model", nullptr);
        if (!body) return std::nullopt;
        // body is Option<Model>; pass through only when Some
        if (body->empty() || (*body)[0] == 0) return std::nullopt;
        return body;
    }
    bool send_message(const Bytes& encrypted) override {
        py::gil_scoped_acquire gil;
        return post_(pyb(encrypted)).cast<bool>();
    }

  private:
    std::optional<Bytes> fetch(const char* path, const Key32* pk) {
        py::gil_scoped_acquire gil;
        py::object r = pk ? get_(path, py::bytes(reinterpret_cast<const char*>(pk->data()), 32))
                          : get_(path, py::none());
        if (r.is_none()) return std::nullopt;
        return frompy(r.cast<py::bytes>());
    }

    py::function get_, post_;
};

void bind_sdk(py::module_& m) {
    auto s = m.def_submodule("sdk");

    py::enum_<Task>(s, "Task")
        .value("None_", Task::None)
        .value("Sum", Task::Sum)
        .value("Update", Task::Update);

    py::class_<XaynetClient, std::shared_ptr<XaynetClient>>(s, "XaynetClient");

    py::class_<InProcessClient, XaynetClient, std::shared_ptr<InProcessClient>>(
        s, "InProcessClient")
        .def(py::init<std::shared_ptr<coord::Coordinator>>());

    py::class_<PyTransportClient, XaynetClient, std::shared_ptr<PyTransportClient>>(
        s, "PyTransportClient")
        .def(py::init<py::function, py::function>(), py::arg("get"), py::arg("post"));

    py::class_<Participant>(s, "Participant")
        .def(py::init([](py::bytes sign_seed, uint64_t scalar_num, uint64_t scalar_den,
                         std::shared_ptr<XaynetClient> client, size_t max_message_size) {
                 Bytes seed = frompy(sign_seed);
                 if (seed.size() != 32) throw std::runtime_error("sign seed must be 32 bytes");
                 PetSettings st;
                 uint8_t pk[32];
                 crypto::ed25519_keypair_from_seed(pk, st.sign_sk, seed.data());
                 std::memcpy(st.sign_pk.data(), pk, 32);
                 st.scalar = mask::Scalar(scalar_num, scalar_den);
                 if (max_message_size) st.max_message_size = max_message_size;
                 return std::make_unique<Participant>(st, std::move(client));
             }),
             py::arg("sign_seed"), py::arg("scalar_num") = 1, py::arg("scalar_den") = 1,
             py::arg("client"), py::arg("max_message_size") = 0)
        .def("tick", &Participant::tick, py::call_guard<py::gil_scoped_release>())
        .def_property_readonly("made_progress", &Participant::made_progress)
        .def_property_readonly("should_set_model", &Participant::should_set_model)
        .def_property_readonly("new_global_model", &Participant::new_global_model)
        .def_property_readonly("model_data_type", &Participant::model_data_type)
        .def_property_readonly("model_length", &Participant::model_length)
        .def_property_readonly("task", &Participant::task)
        .def_property_readonly("phase_id", &Participant::phase_id)
        .def_property_readonly("pk", [](const Participant& p) {
            return py::bytes(reinterpret_cast<const char*>(p.pk().data()), 32);
        })
        .def("set_model",
             [](Participant& p, py::array w) {
                 auto buf = w.request();
                 if (buf.ndim != 1) throw std::runtime_error("model must be 1-D");
                 size_t n = size_t(buf.shape[0]);
                 int num = w.dtype().num();  // by type number, not identity
                 if (num == py::dtype::of<float>().num())
                     p.set_model_f32(static_cast<const float*>(buf.ptr), n);
                 else if (num == py::dtype::of<double>().num())
                     p.set_model_f64(static_cast<const double*>(buf.ptr), n);
                 else if (num == py::dtype::of<int32_t>().num())
                     p.set_model_i32(static_cast<const int32_t*>(buf.ptr), n);
                 else if (num == py::dtype::of<int64_t>().num())
                     p.set_model_i64(static_cast<const int64_t*>(buf.ptr), n);
                 else
                     throw std::runtime_error("model dtype must be f32/f64/i32/i64");
             })
        .def("global_model_bincode",
             [](Participant& p) -> py::object {
                 std::optional<Bytes> m;
                 {
                     // fetching GET /model can move hundreds of MB
                     py::gil_scoped_release rel;
                     m = p.global_model_bincode();
                 }
                 if (!m) return py::none();
                 return pyb(*m);
             })
        .def("save", [](const Participant& p) { return pyb(p.save()); })
        .def_static(
            "restore",
            [](py::bytes state, std::shared_ptr<XaynetClient> client, py::bytes sign_seed,
               uint64_t scalar_num, uint64_t scalar_den, size_t max_message_size) {
                Bytes seed = frompy(sign_seed);
                if (seed.size() != 32) throw std::runtime_error("sign seed must be 32 bytes");
                PetSettings st;
                uint8_t pk[32];
                crypto::ed25519_keypair_from_seed(pk, st.sign_sk, seed.data());
                std::memcpy(st.sign_pk.data(), pk, 32);
                st.scalar = mask::Scalar(scalar_num, scalar_den);
                if (max_message_size) st.max_message_size = max_message_size;
                auto p = Participant::restore(frompy(state), std::move(client), st);
                if (!p) throw std::runtime_error("invalid participant state");
                return p;
            },
            py::arg("state"), py::arg("client"), py::arg("sign_seed"), py::arg("scalar_num") = 1,
            py::arg("scalar_den") = 1, py::arg("max_message_size") = 0)
        .def("set_mask_model_hook",
             [](Participant& p, py::function fn) {
                 // called from tick() with the GIL released -> reacquire.
                 // fn(seed, dtype, raw_weights, n, cfg8) -> wire bytes | None
                 p.set_mask_model_hook([fn](const uint8_t seed[32], int dtype,
                                            const void* data, size_t n,
                                            const std::array<int, 8>& cfg)
                                           -> std::optional<Bytes> {
                     py::gil_scoped_acquire gil;
                     size_t esz = dtype == 0 ? 4 : dtype == 2 ? 4 : 8;
                     py::bytes raw(reinterpret_cast<const char*>(data), n * esz);
                     py::tuple c = py::cast(cfg);
                     py::object r = fn(py::bytes(reinterpret_cast<const char*>(seed), 32),
                                       dtype, raw, n, c);
                     if (r.is_none()) return std::nullopt;
                     std::string s = py::cast<py::bytes>(r);
                     return Bytes(s.begin(), s.end());
                 });
             })
        .def("set_sum2_hook",
             [](Participant& p, py::function fn) {
                 // fn(seeds: list[bytes], length, cfg8) -> wire bytes | None
                 p.set_sum2_hook(
                     [fn](const std::vector<std::array<uint8_t, 32>>& seeds, size_t length,
                          const std::array<int, 8>& cfg) -> std::optional<Bytes> {
                         py::gil_scoped_acquire gil;
                         py::list ls;
                         for (const auto& s : seeds)
                             ls.append(py::bytes(reinterpret_cast<const char*>(s.data()), 32));
                         py::object r = fn(ls, length, py::cast(cfg));
                         if (r.is_none()) return std::nullopt;
                         std::string s = py::cast<py::bytes>(r);
                         return Bytes(s.begin(), s.end());
                     });
             })
        .def("save_reference", [](const Participant& p) { return pyb(p.save_reference()); })
        .def_static(
            "restore_reference",
            [](py::bytes state, std::shared_ptr<XaynetClient> client) {
                auto p = Participant::restore_reference(frompy(state), std::move(client));
                if (!p) throw std::runtime_error("invalid reference participant state");
                return p;
            },
            py::arg("state"), py::arg("client"));

    // decode an Option<Model> bincode body into a numpy array of the given
    // dtype (the app-facing "global model" representation)
    s.def("decode_model", [](py::buffer body, int dtype) -> py::object {
        // zero-copy body view (a 25M-param model body is ~800 MB; copying
        // it into a vector first cost more than the decode itself)
        py::buffer_info bi = body.request();
        const uint8_t* bp = static_cast<const uint8_t*>(bi.ptr);
        size_t blen = size_t(bi.size);
        // decode WITHOUT the GIL: many participant threads decode at once
        if (dtype == 0) {
            std::vector<float> v;
            bool ok;
            {
                py::gil_scoped_release rel;
                ok = bincode::decode_option_model_f32_fast(bp, blen, v);
            }
            if (ok) return py::array_t<float>(py::ssize_t(v.size()), v.data());
        } else if (dtype == 1) {
            std::vector<double> v;
            bool ok;
            {
                py::gil_scoped_release rel;
                ok = bincode::decode_option_model_f64_fast(bp, blen, v);
            }
            if (ok) return py::array_t<double>(py::ssize_t(v.size()), v.data());
        }
        auto m = bincode::decode_option_model(bp, blen);
        if (!m || !*m) return py::none();
        const auto& model = **m;
        switch (dtype) {
            case 0: {
                auto v = mask::model_to_f32(model);
                return py::array_t<float>(py::ssize_t(v.size()), v.data());
            }
            case 1: {
                auto v = mask::model_to_f64(model);
                return py::array_t<double>(py::ssize_t(v.size()), v.data());
            }
            case 2: {
                auto v = mask::model_to_i32(model);
                return py::array_t<int32_t>(py::ssize_t(v.size()), v.data());
            }
            case 3: {
                auto v = mask::model_to_i64(model);
                return py::array_t<int64_t>(py::ssize_t(v.size()), v.data());
            }
        }
        throw std::runtime_error("bad dtype");
    });

    // tests: generic-rational round-trip (decode -> slow re-encode), used to
    // pin the fast typed encoders byte-for-byte against the Rational path
    s.def("reencode_model_slow", [](py::bytes body) {
        Bytes b = frompy(body);
        auto m = bincode::decode_option_model(b.data(), b.size());
        if (!m || !*m) throw std::runtime_error("bad model body");
        return pyb(bincode::encode_option_model(&**m));
    });

    // encode a numpy model into Option<Model> bincode (tests / tools / GPU driver)
    s.def("encode_model_f32", [](py::array_t<float> w) {
        return pyb(bincode::encode_option_model_f32(w.data(), size_t(w.size())));
    });
    s.def("encode_model", [](py::array w) -> py::object {
        auto buf = w.request();
        if (buf.ndim != 1) throw std::runtime_error("model must be 1-D");
        size_t n = size_t(buf.shape[0]);
        // compare by type number, not object identity: arrays that crossed a
        // pickle/process boundary carry equal-but-distinct dtype objects
        int num = w.dtype().num();
        bincode::EncodedModel em;
        {
            // chunked multi-thread encode (byte-identical to the serial
            // encoders) with the GIL released: a 25M-param f32 body is
            // ~800 MB and the serial loop cost seconds on the unmask tail
            py::gil_scoped_release rel;
            if (num == py::dtype::of<float>().num())
                em = bincode::encode_option_model_mt_f32(static_cast<const float*>(buf.ptr), n);
            else if (num == py::dtype::of<double>().num())
                em = bincode::encode_option_model_mt_f64(static_cast<const double*>(buf.ptr), n);
            else if (num == py::dtype::of<int32_t>().num())
                em = bincode::encode_option_model_mt_i32(static_cast<const int32_t*>(buf.ptr), n);
            else if (num == py::dtype::of<int64_t>().num())
                em = bincode::encode_option_model_mt_i64(static_cast<const int64_t*>(buf.ptr), n);
            else
                throw std::runtime_error("model dtype must be f32/f64/i32/i64");
        }
        // assemble straight into the final bytes object (no concat copy)
        PyObject* raw = PyBytes_FromStringAndSize(nullptr, Py_ssize_t(em.total()));
        if (!raw) throw py::error_already_set();
        py::object holder = py::reinterpret_steal<py::object>(raw);
        auto* dst = reinterpret_cast<uint8_t*>(PyBytes_AS_STRING(raw));
        {
            py::gil_scoped_release rel;
            em.assemble(dst);
        }
        return holder;
    });
}
