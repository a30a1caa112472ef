// Bindings for the native REST server (coordinator HTTP API) and the SDK
// HTTP client (reference rest.rs / xaynet-sdk client.rs parity).
#include <pybind11/pybind11.h>

#include "rest/rest.h"

namespace py = pybind11;
using namespace xaynet;

void bind_rest(py::module_& m) {
    auto r = m.def_submodule("rest");

    py::class_<rest::RestServer>(r, "RestServer")
        .def(py::init<std::shared_ptr<coord::Coordinator>, std::string, uint16_t, int>(),
             py::arg("coordinator"), py::arg("host") = "127.0.0.1", py::arg("port") = 0,
             py::arg("workers") = 4)
        .def(py::init<std::shared_ptr<coord::Coordinator>, std::string, uint16_t, std::string,
                      std::string, std::string>(),
             py::arg("coordinator"), py::arg("host"), py::arg("port"), py::arg("tls_cert"),
             py::arg("tls_key"), py::arg("tls_client_auth") = "")
        .def("start", &rest::RestServer::start, py::call_guard<py::gil_scoped_release>())
        .def("stop", &rest::RestServer::stop, py::call_guard<py::gil_scoped_release>())
        .def_property_readonly("port", &rest::RestServer::port);

    py::class_<rest::HttpXaynetClient, sdk::XaynetClient,
               std::shared_ptr<rest::HttpXaynetClient>>(r, "HttpXaynetClient")
        .def(py::init<std::string, uint16_t, double>(), py::arg("host"), py::arg("port"),
             py::arg("timeout_s") = 30.0);

    py::class_<rest::TlsXaynetClient, sdk::XaynetClient,
               std::shared_ptr<rest::TlsXaynetClient>>(r, "TlsXaynetClient")
        .def(py::init<std::string, uint16_t, std::string, bool, std::string, std::string,
                      double>(),
             py::arg("host"), py::arg("port"), py::arg("ca_file") = "",
             py::arg("insecure") = false, py::arg("cert_file") = "", py::arg("key_file") = "",
             py::arg("timeout_s") = 30.0);

    // raw TLS client (tests / tools)
    py::class_<http::TlsHttpClient>(r, "TlsHttpClient")
        .def(py::init<std::string, uint16_t, std::string, bool, std::string, std::string,
                      double>(),
             py::arg("host"), py::arg("port"), py::arg("ca_file") = "",
             py::arg("insecure") = false, py::arg("cert_file") = "", py::arg("key_file") = "",
             py::arg("timeout_s") = 30.0)
        .def(
            "request",
            [](http::TlsHttpClient& c, const std::string& method, const std::string& pq,
               py::object body) -> py::object {
                Bytes b;
                const Bytes* bp = nullptr;
                if (!body.is_none()) {
                    std::string s = body.cast<py::bytes>();
                    b.assign(s.begin(), s.end());
                    bp = &b;
                }
                int status = 0;
                Bytes out;
                bool ok;
                {
                    py::gil_scoped_release rel;
                    ok = c.request(method, pq, bp, status, out);
                }
                if (!ok) return py::none();
                return py::make_tuple(
                    status, py::bytes(reinterpret_cast<const char*>(out.data()), out.size()));
            },
            py::arg("method"), py::arg("path"), py::arg("body") = py::none());

    // raw HTTP client (tests / tools)
    py::class_<http::HttpClient>(r, "HttpClient")
        .def(py::init<std::string, uint16_t, double>(), py::arg("host"), py::arg("port"),
             py::arg("timeout_s") = 30.0)
        .def(
            "request",
            [](http::HttpClient& c, const std::string& method, const std::string& pq,
               py::object body) -> py::object {
                Bytes b;
                const Bytes* bp = nullptr;
                if (!body.is_none()) {
                    std::string s = body.cast<py::bytes>();
                    b.assign(s.begin(), s.end());
                    bp = &b;
                }
                int status = 0;
                Bytes out;
                bool ok;
                {
                    py::gil_scoped_release rel;
                    ok = c.request(method, pq, bp, status, out);
                }
                if (!ok) return py::none();
                return py::make_tuple(
                    status, py::bytes(reinterpret_cast<const char*>(out.data()), out.size()));
            },
            py::arg("method"), py::arg("path"), py::arg("body") = py::none());
}
