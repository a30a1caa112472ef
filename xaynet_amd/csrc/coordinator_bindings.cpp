// Bindings for the coordinator (state machine + storage + ingest pipeline).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "coordinator/coordinator.h"
#include "coordinator/metrics.h"
#include "coordinator/redis.h"
#include "coordinator/s3.h"
#include "coordinator/trace.h"

namespace py = pybind11;
using namespace xaynet;
using namespace xaynet::coord;

static py::bytes pyb(const Bytes& b) {
    return py::bytes(reinterpret_cast<const char*>(b.data()), b.size());
}

static Bytes frompy(py::bytes b) {
    std::string s = b;
    return Bytes(s.begin(), s.end());
}

void bind_coordinator(py::module_& m) {
    auto c = m.def_submodule("coordinator");

    py::class_<Settings>(c, "Settings")
        .def(py::init<>())
        .def_readwrite("sum_prob", &Settings::sum_prob)
        .def_readwrite("update_prob", &Settings::update_prob)
        .def_readwrite("model_length", &Settings::model_length)
        .def_readwrite("restore", &Settings::restore)
        .def_readwrite("multipart_max_entries", &Settings::multipart_max_entries)
        .def_readwrite("multipart_max_per_pk_bytes", &Settings::multipart_max_per_pk_bytes)
        .def_readwrite("multipart_max_total_bytes", &Settings::multipart_max_total_bytes)
        .def_readwrite("unmask_timeout_s", &Settings::unmask_timeout_s)
        .def_property(
            "mask_cfg", [](const Settings& s) { return s.mask_cfg; },
            [](Settings& s, const mask::MaskConfigPair& p) { s.mask_cfg = p; })
        .def("set_sum", [](Settings& s, uint64_t cmin, uint64_t cmax, double tmin, double tmax) {
            s.sum = PhaseParams{{cmin, cmax}, {tmin, tmax}};
        })
        .def("set_update", [](Settings& s, uint64_t cmin, uint64_t cmax, double tmin, double tmax) {
            s.update = PhaseParams{{cmin, cmax}, {tmin, tmax}};
        })
        .def("set_sum2", [](Settings& s, uint64_t cmin, uint64_t cmax, double tmin, double tmax) {
            s.sum2 = PhaseParams{{cmin, cmax}, {tmin, tmax}};
        });

    py::enum_<PhaseId>(c, "PhaseId")
        .value("Idle", PhaseId::Idle)
        .value("Sum", PhaseId::Sum)
        .value("Update", PhaseId::Update)
        .value("Sum2", PhaseId::Sum2)
        .value("Unmask", PhaseId::Unmask)
        .value("Failure", PhaseId::Failure)
        .value("Shutdown", PhaseId::Shutdown);

    py::enum_<SumPartAddError>(c, "SumPartAddError")
        .value("Ok", SumPartAddError::Ok)
        .value("AlreadyExists", SumPartAddError::AlreadyExists)
        .value("Storage", SumPartAddError::Storage);
    py::enum_<SeedDictAddError>(c, "SeedDictAddError")
        .value("Ok", SeedDictAddError::Ok)
        .value("LengthMisMatch", SeedDictAddError::LengthMisMatch)
        .value("UnknownSumParticipant", SeedDictAddError::UnknownSumParticipant)
        .value("UpdatePkAlreadySubmitted", SeedDictAddError::UpdatePkAlreadySubmitted)
        .value("UpdatePkAlreadyExistsInUpdateSeedDict",
               SeedDictAddError::UpdatePkAlreadyExistsInUpdateSeedDict)
        .value("Storage", SeedDictAddError::Storage);
    py::enum_<MaskScoreIncrError>(c, "MaskScoreIncrError")
        .value("Ok", MaskScoreIncrError::Ok)
        .value("UnknownSumParticipant", MaskScoreIncrError::UnknownSumParticipant)
        .value("MaskAlreadySubmitted", MaskScoreIncrError::MaskAlreadySubmitted)
        .value("Storage", MaskScoreIncrError::Storage);
    py::enum_<PipelineError>(c, "PipelineError")
        .value("Ok", PipelineError::Ok)
        .value("Decrypt", PipelineError::Decrypt)
        .value("Parsing", PipelineError::Parsing)
        .value("InvalidMessageSignature", PipelineError::InvalidMessageSignature)
        .value("InvalidCoordinatorPublicKey", PipelineError::InvalidCoordinatorPublicKey)
        .value("UnexpectedMessage", PipelineError::UnexpectedMessage)
        .value("NotSumEligible", PipelineError::NotSumEligible)
        .value("NotUpdateEligible", PipelineError::NotUpdateEligible)
        .value("MessageRejected", PipelineError::MessageRejected)
        .value("MessageDiscarded", PipelineError::MessageDiscarded)
        .value("AggregationFailed", PipelineError::AggregationFailed)
        .value("Internal", PipelineError::Internal);

    // ---- metrics (reference metrics/mod.rs; InfluxDB line protocol) ----
    c.def("install_metrics_file", [](const std::string& path) {
        metrics::Recorder::install_file(path);
    });
    c.def("install_metrics_influxdb",
          [](const std::string& host, uint16_t port, const std::string& db) {
              metrics::Recorder::install_influxdb(host, port, db);
          },
          py::arg("host"), py::arg("port"), py::arg("db") = "metrics");
    c.def("metrics_dropped", []() -> size_t {
        auto* r = metrics::Recorder::global();
        return r ? r->dropped() : 0;
    });
    c.def("install_metrics_callback", [](py::function fn) {
        metrics::Recorder::install_sink([fn](const std::string& line) {
            py::gil_scoped_acquire gil;
            fn(py::str(line));
        });
        // the writer thread must not call into Python during interpreter
        // finalization: drain + join it via atexit
        py::module_::import("atexit").attr("register")(
            py::cpp_function([]() {
                py::gil_scoped_release rel;
                metrics::Recorder::uninstall();
            }));
    });
    // ---- tracing spans (reference tracing/phase spans; request->phase) ----
    c.def("install_trace_file", [](const std::string& path) { trace::install_file(path); });
    c.def("install_trace_callback", [](py::function fn) {
        trace::install([fn](const std::string& line) {
            py::gil_scoped_acquire gil;
            fn(py::str(line));
        });
        py::module_::import("atexit").attr("register")(
            py::cpp_function([]() { trace::uninstall(); }));
    });
    c.def("uninstall_trace", []() { trace::uninstall(); });

    c.def("uninstall_metrics", []() {
        // drop outside the GIL: the recorder joins its writer thread, which
        // may be blocked acquiring the GIL for a python callback sink
        py::gil_scoped_release rel;
        metrics::Recorder::uninstall();
    });
    c.def("metrics_flush", []() {
        py::gil_scoped_release rel;
        if (auto* r = metrics::Recorder::global()) r->flush();
    });

    // NOTE: every method releases the GIL — network-backed stores (Redis)
    // block on sockets, and test stubs may be Python threads in-process
    py::class_<CoordinatorStorage, std::shared_ptr<CoordinatorStorage>>(c, "CoordinatorStorage")
        .def("latest_global_model_id",
             [](CoordinatorStorage& s) -> py::object {
                 std::optional<std::string> id;
                 {
                     py::gil_scoped_release rel;
                     id = s.latest_global_model_id();
                 }
                 if (!id) return py::none();
                 return py::str(*id);
             })
        .def("set_latest_global_model_id", &CoordinatorStorage::set_latest_global_model_id,
             py::call_guard<py::gil_scoped_release>())
        .def("is_ready", &CoordinatorStorage::is_ready,
             py::call_guard<py::gil_scoped_release>())
        .def("set_coordinator_state",
             [](CoordinatorStorage& s, py::bytes b) {
                 Bytes v = frompy(b);
                 py::gil_scoped_release rel;
                 return s.set_coordinator_state(v);
             })
        .def("coordinator_state",
             [](CoordinatorStorage& s) -> py::object {
                 std::optional<Bytes> v;
                 {
                     py::gil_scoped_release rel;
                     v = s.coordinator_state();
                 }
                 if (!v) return py::none();
                 return pyb(*v);
             })
        .def("add_sum_participant",
             [](CoordinatorStorage& s, py::bytes pk, py::bytes ephm) {
                 Key32 k{}, e{};
                 Bytes kb = frompy(pk), eb = frompy(ephm);
                 if (kb.size() != 32 || eb.size() != 32)
                     throw std::runtime_error("keys must be 32 bytes");
                 std::memcpy(k.data(), kb.data(), 32);
                 std::memcpy(e.data(), eb.data(), 32);
                 py::gil_scoped_release rel;
                 return s.add_sum_participant(k, e);
             })
        .def("sum_dict",
             [](CoordinatorStorage& s) -> py::object {
                 std::optional<SumDict> d;
                 {
                     py::gil_scoped_release rel;
                     d = s.sum_dict();
                 }
                 if (!d) return py::none();
                 py::dict out;
                 for (const auto& [pk, ephm] : *d)
                     out[py::bytes(reinterpret_cast<const char*>(pk.data()), 32)] =
                         py::bytes(reinterpret_cast<const char*>(ephm.data()), 32);
                 return out;
             })
        .def("add_local_seed_dict",
             [](CoordinatorStorage& s, py::bytes upd,
                const std::vector<std::pair<py::bytes, py::bytes>>& entries) {
                 Key32 u{};
                 Bytes ub = frompy(upd);
                 if (ub.size() != 32) throw std::runtime_error("update pk must be 32 bytes");
                 std::memcpy(u.data(), ub.data(), 32);
                 std::vector<msg::LocalSeedEntry> local;
                 for (const auto& [pk, seed] : entries) {
                     msg::LocalSeedEntry e;
                     Bytes kb = frompy(pk), sb = frompy(seed);
                     if (kb.size() != 32 || sb.size() != 80)
                         throw std::runtime_error("entry must be (32B pk, 80B seed)");
                     std::memcpy(e.pk.data(), kb.data(), 32);
                     std::memcpy(e.seed.data(), sb.data(), 80);
                     local.push_back(e);
                 }
                 py::gil_scoped_release rel;
                 return s.add_local_seed_dict(u, local);
             })
        .def("seed_dict",
             [](CoordinatorStorage& s) -> py::object {
                 std::optional<SeedDict> d;
                 {
                     py::gil_scoped_release rel;
                     d = s.seed_dict();
                 }
                 if (!d) return py::none();
                 py::dict out;
                 for (const auto& [pk, entries] : *d) {
                     py::dict inner;
                     for (const auto& [upk, seed] : entries)
                         inner[py::bytes(reinterpret_cast<const char*>(upk.data()), 32)] =
                             py::bytes(reinterpret_cast<const char*>(seed.data()), 80);
                     out[py::bytes(reinterpret_cast<const char*>(pk.data()), 32)] = inner;
                 }
                 return out;
             })
        .def("incr_mask_score",
             [](CoordinatorStorage& s, py::bytes pk, py::bytes mask) {
                 Key32 k{};
                 Bytes kb = frompy(pk), mb = frompy(mask);
                 if (kb.size() != 32) throw std::runtime_error("pk must be 32 bytes");
                 std::memcpy(k.data(), kb.data(), 32);
                 py::gil_scoped_release rel;
                 return s.incr_mask_score(k, mb);
             })
        .def("best_masks",
             [](CoordinatorStorage& s, size_t n) {
                 std::vector<std::pair<Bytes, uint64_t>> bm;
                 {
                     py::gil_scoped_release rel;
                     bm = s.best_masks(n);
                 }
                 py::list out;
                 for (const auto& [b, score] : bm)
                     out.append(py::make_tuple(pyb(b), score));
                 return out;
             })
        .def("number_of_unique_masks", &CoordinatorStorage::number_of_unique_masks,
             py::call_guard<py::gil_scoped_release>())
        .def("delete_dicts", &CoordinatorStorage::delete_dicts,
             py::call_guard<py::gil_scoped_release>())
        .def("delete_coordinator_data", &CoordinatorStorage::delete_coordinator_data,
             py::call_guard<py::gil_scoped_release>());
    py::class_<ModelStorage, std::shared_ptr<ModelStorage>>(c, "ModelStorage")
        .def("global_model",
             [](ModelStorage& s, const std::string& id) -> py::object {
                 std::optional<Bytes> b;
                 {
                     py::gil_scoped_release rel;
                     b = s.global_model(id);
                 }
                 if (!b) return py::none();
                 return pyb(*b);
             })
        .def("set_global_model",
             [](ModelStorage& s, uint64_t round_id, py::bytes seed, py::bytes model)
                 -> py::object {
                 Key32 k{};
                 Bytes kb = frompy(seed), mb = frompy(model);
                 if (kb.size() != 32) throw std::runtime_error("seed must be 32 bytes");
                 std::memcpy(k.data(), kb.data(), 32);
                 std::optional<std::string> id;
                 {
                     py::gil_scoped_release rel;
                     id = s.set_global_model(round_id, k, mb);
                 }
                 if (!id) return py::none();
                 return py::str(*id);
             })
        .def("is_ready", &ModelStorage::is_ready,
             py::call_guard<py::gil_scoped_release>());

    py::class_<InMemoryCoordinatorStorage, CoordinatorStorage,
               std::shared_ptr<InMemoryCoordinatorStorage>>(c, "InMemoryStorage")
        .def(py::init<>());
    py::class_<InMemoryModelStorage, ModelStorage, std::shared_ptr<InMemoryModelStorage>>(
        c, "InMemoryModels")
        .def(py::init<>());
    py::class_<FileCoordinatorStorage, CoordinatorStorage,
               std::shared_ptr<FileCoordinatorStorage>>(c, "FileStorage")
        .def(py::init<std::string>(), py::arg("dir"));
    py::class_<FaultInjectionStorage, CoordinatorStorage,
               std::shared_ptr<FaultInjectionStorage>>(c, "FaultInjectionStorage")
        .def(py::init<>())
        .def_property(
            "fail_sum_dict",
            [](FaultInjectionStorage& s) { return s.fail_sum_dict.load(); },
            [](FaultInjectionStorage& s, int v) { s.fail_sum_dict = v; })
        .def_property(
            "fail_seed_dict",
            [](FaultInjectionStorage& s) { return s.fail_seed_dict.load(); },
            [](FaultInjectionStorage& s, int v) { s.fail_seed_dict = v; })
        .def_property(
            "fail_state",
            [](FaultInjectionStorage& s) { return s.fail_state.load(); },
            [](FaultInjectionStorage& s, int v) { s.fail_state = v; })
        .def_property(
            "fail_best_masks",
            [](FaultInjectionStorage& s) { return s.fail_best_masks.load(); },
            [](FaultInjectionStorage& s, int v) { s.fail_best_masks = v; })
        .def_property(
            "not_ready",
            [](FaultInjectionStorage& s) { return s.not_ready.load(); },
            [](FaultInjectionStorage& s, int v) { s.not_ready = v; });
    py::class_<FileModelStorage, ModelStorage, std::shared_ptr<FileModelStorage>>(
        c, "FileModels")
        .def(py::init<std::string>(), py::arg("dir"));
    py::class_<RedisCoordinatorStorage, CoordinatorStorage,
               std::shared_ptr<RedisCoordinatorStorage>>(c, "RedisStorage")
        .def(py::init<std::string, uint16_t, double>(), py::arg("host"), py::arg("port"),
             py::arg("timeout_s") = 5.0);
    py::class_<RedisModelStorage, ModelStorage, std::shared_ptr<RedisModelStorage>>(
        c, "RedisModels")
        .def(py::init<std::string, uint16_t, double>(), py::arg("host"), py::arg("port"),
             py::arg("timeout_s") = 5.0);
    py::class_<S3ModelStorage, ModelStorage, std::shared_ptr<S3ModelStorage>>(c, "S3Models")
        .def(py::init<std::string, uint16_t, std::string, double>(), py::arg("host"),
             py::arg("port"), py::arg("bucket") = "global-models",
             py::arg("timeout_s") = 10.0);

    py::class_<Coordinator, std::shared_ptr<Coordinator>>(c, "Coordinator")
        .def(py::init([](const Settings& s, std::shared_ptr<CoordinatorStorage> store,
                         std::shared_ptr<ModelStorage> models, bool staged) {
                 // release the GIL: the ctor's restore path does storage I/O
                 // (Redis GET), which must not starve in-process Python stubs
                 py::gil_scoped_release rel;
                 return std::make_shared<Coordinator>(
                     s, store, models,
                     staged ? AggregationPlane::Staged : AggregationPlane::Cpu);
             }),
             py::arg("settings"), py::arg("store"), py::arg("models"),
             py::arg("staged") = false)
        .def("start", &Coordinator::start, py::call_guard<py::gil_scoped_release>())
        .def("stop", &Coordinator::stop, py::call_guard<py::gil_scoped_release>())
        .def("run_one_phase", &Coordinator::run_one_phase,
             py::call_guard<py::gil_scoped_release>())
        .def_property_readonly("phase", &Coordinator::phase)
        .def_property_readonly("round_id", &Coordinator::round_id)
        .def(
            "handle_encrypted_message",
            [](Coordinator& c, py::buffer data) {
                // zero-copy view: message bodies are up to hundreds of MB
                // and the coordinator only reads them during this call
                py::buffer_info bi = data.request();
                int r;
                {
                    py::gil_scoped_release rel;
                    r = int(c.handle_encrypted_message(
                        static_cast<const uint8_t*>(bi.ptr), size_t(bi.size)));
                }
                return r;
            })
        .def(
            "handle_message_bytes",
            [](Coordinator& c, py::buffer data) {
                py::buffer_info bi = data.request();
                int r;
                {
                    py::gil_scoped_release rel;
                    r = int(c.handle_message_bytes(
                        static_cast<const uint8_t*>(bi.ptr), size_t(bi.size)));
                }
                return r;
            })
        .def("fetch_round_params", [](Coordinator& c) { return pyb(c.fetch_round_params()); })
        .def("fetch_sum_dict", [](Coordinator& c) { return pyb(c.fetch_sum_dict()); })
        .def("fetch_seeds",
             [](Coordinator& c, py::bytes pk) {
                 Bytes k = frompy(pk);
                 if (k.size() != 32) throw std::runtime_error("pk must be 32 bytes");
                 msg::Key32 key;
                 std::memcpy(key.data(), k.data(), 32);
                 return pyb(c.fetch_seeds(key));
             })
        .def("fetch_model", [](Coordinator& c) { return pyb(c.fetch_model()); })
        .def("events_version", &Coordinator::events_version)
        .def("staged_count", &Coordinator::staged_count,
             py::call_guard<py::gil_scoped_release>())
        .def("pop_staged_vect",
             [](Coordinator& c, uintptr_t dst, size_t cap) -> py::object {
                 Bytes unit;
                 size_t n;
                 {
                     py::gil_scoped_release rel;
                     n = c.pop_staged_vect(reinterpret_cast<uint8_t*>(dst), cap, unit);
                 }
                 if (n == 0) return py::none();
                 return py::make_tuple(n, pyb(unit));
             })
        .def("drain_staged_updates",
             [](Coordinator& c) {
                 auto v = c.drain_staged_updates();
                 py::list out;
                 for (const auto& b : v) out.append(pyb(b));
                 return out;
             })
        .def("pending_unmask",
             [](Coordinator& c) -> py::object {
                 Bytes mb;
                 uint64_t nb = 0;
                 if (!c.pending_unmask(mb, nb)) return py::none();
                 return py::make_tuple(pyb(mb), nb);
             })
        .def("supply_unmasked_model",
             [](Coordinator& c, py::bytes model) { c.supply_unmasked_model(frompy(model)); })
        .def("checkpoint_state", [](Coordinator& c) { return pyb(c.checkpoint_state()); })
        .def("restore_state", [](Coordinator& c, py::bytes st) { return c.restore_state(frompy(st)); });
}
