// Bindings for the coordinator (state machine + storage + ingest pipeline).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "coordinator/coordinator.h"
#include "coordinator/metrics.h"

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

    py::class_<CoordinatorStorage, std::shared_ptr<CoordinatorStorage>>(c, "CoordinatorStorage")
        .def("latest_global_model_id",
             [](CoordinatorStorage& s) -> py::object {
                 auto id = s.latest_global_model_id();
                 if (!id) return py::none();
                 return py::str(*id);
             })
        .def("is_ready", &CoordinatorStorage::is_ready);
    py::class_<ModelStorage, std::shared_ptr<ModelStorage>>(c, "ModelStorage")
        .def("global_model",
             [](ModelStorage& s, const std::string& id) -> py::object {
                 auto b = s.global_model(id);
                 if (!b) return py::none();
                 return pyb(*b);
             })
        .def("is_ready", &ModelStorage::is_ready);

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

    py::class_<Coordinator, std::shared_ptr<Coordinator>>(c, "Coordinator")
        .def(py::init([](const Settings& s, std::shared_ptr<CoordinatorStorage> store,
                         std::shared_ptr<ModelStorage> models, bool staged) {
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
            [](Coordinator& c, py::bytes data) {
                Bytes b = frompy(data);
                int r;
                {
                    py::gil_scoped_release rel;
                    r = int(c.handle_encrypted_message(b.data(), b.size()));
                }
                return r;
            })
        .def(
            "handle_message_bytes",
            [](Coordinator& c, py::bytes data) {
                Bytes b = frompy(data);
                int r;
                {
                    py::gil_scoped_release rel;
                    r = int(c.handle_message_bytes(b.data(), b.size()));
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
