/*
 * binder-amd: Python bindings (binder_amd._native).
 *
 * Exposes the DNS wire codec, the record compiler, and an in-process
 * StubStore+Engine pair so the pytest suite and BASELINE config 1
 * ("single A-record lookup against an in-process stub ZK") can exercise
 * the exact native hot path without sockets.
 */
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../common/json.hpp"
#include "../dns/codec.hpp"
#include "../engine/engine.hpp"
#include "../engine/store.hpp"

namespace py = pybind11;
using namespace bamd;

/* DNS names from the wire are arbitrary bytes; decode lossily for
 * Python (invalid UTF-8 becomes U+FFFD) instead of throwing. */
static py::str pystr(const std::string& s) {
    PyObject* o = PyUnicode_DecodeUTF8(s.data(), (Py_ssize_t)s.size(),
                                       "replace");
    if (o == nullptr) {
        PyErr_Clear();
        return py::str("");
    }
    return py::reinterpret_steal<py::str>(o);
}

static py::dict recordToDict(const dns::Record& r) {
    py::dict d;
    d["name"] = pystr(r.name);
    d["type"] = dns::typeName(r.type);
    d["class"] = r.rclass;
    d["ttl"] = r.ttl;
    switch (r.type) {
    case dns::TYPE_A:
    case dns::TYPE_AAAA:
        d["address"] = r.addrString();
        break;
    case dns::TYPE_SRV:
        d["target"] = pystr(r.target);
        d["port"] = r.port;
        d["priority"] = r.priority;
        d["weight"] = r.weight;
        break;
    case dns::TYPE_SOA:
        d["mname"] = pystr(r.soa.mname);
        d["rname"] = pystr(r.soa.rname);
        d["serial"] = r.soa.serial;
        d["minimum"] = r.soa.minimum;
        break;
    case dns::TYPE_OPT:
        d["udp_size"] = r.rclass;
        break;
    default:
        d["target"] = pystr(r.target);
        break;
    }
    return d;
}

static dns::Record recordFromDict(const py::dict& d) {
    std::string type = d["type"].cast<std::string>();
    std::string name = d.contains("name") ? d["name"].cast<std::string>() : "";
    uint32_t ttl = d.contains("ttl") ? d["ttl"].cast<uint32_t>() : 0;
    if (type == "A")
        return dns::Record::A(name, d["address"].cast<std::string>(), ttl);
    if (type == "AAAA")
        return dns::Record::AAAA(name, d["address"].cast<std::string>(), ttl);
    if (type == "SRV") {
        auto r = dns::Record::SRV(name, d["target"].cast<std::string>(),
                                  d["port"].cast<uint16_t>(), ttl);
        if (d.contains("priority"))
            r.priority = d["priority"].cast<uint16_t>();
        if (d.contains("weight")) r.weight = d["weight"].cast<uint16_t>();
        return r;
    }
    if (type == "PTR")
        return dns::Record::PTR(name, d["target"].cast<std::string>(), ttl);
    if (type == "CNAME")
        return dns::Record::CNAME(name, d["target"].cast<std::string>(), ttl);
    if (type == "TXT")
        return dns::Record::TXT(name, d["target"].cast<std::string>(), ttl);
    if (type == "OPT")
        return dns::Record::OPT(d.contains("udp_size")
                                    ? d["udp_size"].cast<uint16_t>()
                                    : 1400);
    if (type == "SOA") {
        dns::SoaData soa;
        if (d.contains("mname")) soa.mname = d["mname"].cast<std::string>();
        if (d.contains("rname")) soa.rname = d["rname"].cast<std::string>();
        if (d.contains("minimum"))
            soa.minimum = d["minimum"].cast<uint32_t>();
        return dns::Record::SOA(name, soa, ttl);
    }
    throw std::runtime_error("unsupported record type: " + type);
}

static py::dict messageToDict(const dns::Message& m) {
    py::dict d;
    d["id"] = m.header.id;
    d["qr"] = m.header.qr;
    d["opcode"] = m.header.opcode;
    d["aa"] = m.header.aa;
    d["tc"] = m.header.tc;
    d["rd"] = m.header.rd;
    d["ra"] = m.header.ra;
    d["rcode"] = dns::rcodeName(m.header.rcode);
    py::list qs;
    for (const auto& q : m.questions) {
        py::dict qd;
        qd["name"] = pystr(q.name);
        qd["type"] = dns::typeName(q.qtype);
        qd["class"] = q.qclass;
        qs.append(qd);
    }
    d["questions"] = qs;
    auto section = [](const std::vector<dns::Record>& rs) {
        py::list out;
        for (const auto& r : rs) out.append(recordToDict(r));
        return out;
    };
    d["answers"] = section(m.answers);
    d["authorities"] = section(m.authorities);
    d["additionals"] = section(m.additionals);
    return d;
}

static dns::Message messageFromDict(const py::dict& d) {
    dns::Message m;
    if (d.contains("id")) m.header.id = d["id"].cast<uint16_t>();
    if (d.contains("qr")) m.header.qr = d["qr"].cast<bool>();
    if (d.contains("aa")) m.header.aa = d["aa"].cast<bool>();
    if (d.contains("tc")) m.header.tc = d["tc"].cast<bool>();
    if (d.contains("rd")) m.header.rd = d["rd"].cast<bool>();
    if (d.contains("ra")) m.header.ra = d["ra"].cast<bool>();
    if (d.contains("rcode")) {
        std::string rc = d["rcode"].cast<std::string>();
        for (uint8_t i = 0; i < 16; ++i)
            if (rc == dns::rcodeName(i)) m.header.rcode = i;
    }
    if (d.contains("questions")) {
        for (auto item : d["questions"].cast<py::list>()) {
            auto qd = item.cast<py::dict>();
            dns::Question q;
            q.name = qd["name"].cast<std::string>();
            std::string t = qd.contains("type")
                                ? qd["type"].cast<std::string>()
                                : "A";
            q.qtype = dns::typeFromName(t);
            m.questions.push_back(std::move(q));
        }
    }
    auto section = [&](const char* key, std::vector<dns::Record>& out) {
        if (!d.contains(key)) return;
        for (auto item : d[key].cast<py::list>())
            out.push_back(recordFromDict(item.cast<py::dict>()));
    };
    section("answers", m.answers);
    section("authorities", m.authorities);
    section("additionals", m.additionals);
    return m;
}

/*
 * In-process store + engine: the full native resolution path minus
 * sockets.
 */
class PyStubEngine {
  public:
    PyStubEngine(const std::string& dnsDomain,
                 const std::string& datacenterName, bool recursion) {
        EngineConfig cfg;
        cfg.dnsDomain = dnsDomain;
        cfg.datacenterName = datacenterName;
        cfg.recursionEnabled = recursion;
        engine_ = std::make_unique<Engine>(cfg, &store_);
    }

    void put(const std::string& domain, const std::string& jsonText) {
        auto parsed = Json::parse(jsonText);
        if (!parsed) return;  // parity: unparseable payloads ignored
        store_.put(domain, *parsed);
    }
    void remove(const std::string& domain) { store_.remove(domain); }
    void setReady(bool r) { store_.setReady(r); }

    /* Wire-in, wire-out. Returns (response_bytes, action_str). */
    py::tuple queryWire(py::bytes wire, size_t maxSize) {
        std::string_view sv = std::string_view(wire);
        auto q = dns::Message::decode((const uint8_t*)sv.data(), sv.size());
        if (!q) return py::make_tuple(py::bytes(), "drop");
        dns::Message resp;
        QueryResult qr = engine_->handle(*q, resp);
        if (qr.action == QueryResult::Action::Recurse)
            return py::make_tuple(py::bytes(), "recurse");
        auto out = resp.encode(maxSize);
        return py::make_tuple(
            py::bytes((const char*)out.data(), out.size()), "respond");
    }

    /* Microbenchmark: run the wire query N times in a tight C++ loop
     * (BASELINE config 1 without Python call overhead). Returns
     * queries/second. */
    double benchWire(py::bytes wire, size_t iters, size_t maxSize) {
        std::string_view sv = std::string_view(wire);
        std::vector<uint8_t> out;
        struct timespec t0, t1;
        clock_gettime(CLOCK_MONOTONIC, &t0);
        for (size_t i = 0; i < iters; ++i) {
            auto q = dns::Message::decode((const uint8_t*)sv.data(),
                                          sv.size());
            if (!q) return 0;
            dns::Message resp;
            engine_->handle(*q, resp);
            resp.encodeInto(out, maxSize);
        }
        clock_gettime(CLOCK_MONOTONIC, &t1);
        double secs = (double)(t1.tv_sec - t0.tv_sec) +
                      (double)(t1.tv_nsec - t0.tv_nsec) / 1e9;
        return secs > 0 ? (double)iters / secs : 0;
    }

    /* Convenience: query by name/type, response as dict. */
    py::dict query(const std::string& name, const std::string& type,
                   bool rd) {
        dns::Message q;
        q.header.id = 0x1234;
        q.header.rd = rd;
        dns::Question question;
        question.name = name;
        question.qtype = dns::typeFromName(type);
        q.questions.push_back(question);
        dns::Message resp;
        QueryResult qr = engine_->handle(q, resp);
        py::dict out = messageToDict(resp);
        out["action"] = qr.action == QueryResult::Action::Recurse
                            ? "recurse"
                            : "respond";
        out["log_name"] = qr.logName;
        return out;
    }

  private:
    StubStore store_;
    std::unique_ptr<Engine> engine_;
};

PYBIND11_MODULE(_native, m) {
    m.doc() = "binder-amd native core bindings";

    m.def("encode_message", [](const py::dict& d, size_t maxSize) {
        auto out = messageFromDict(d).encode(maxSize);
        return py::bytes((const char*)out.data(), out.size());
    }, py::arg("msg"), py::arg("max_size") = 0);

    m.def("decode_message", [](py::bytes wire) -> py::object {
        std::string_view sv = std::string_view(wire);
        auto m2 = dns::Message::decode((const uint8_t*)sv.data(), sv.size());
        if (!m2) return py::none();
        return messageToDict(*m2);
    });

    m.def("compile_record", [](const std::string& jsonText) {
        auto parsed = Json::parse(jsonText);
        py::dict d;
        if (!parsed) {
            d["ignored"] = true;
            return d;
        }
        CompiledRecord r = compileRecord(*parsed);
        d["ignored"] = !r.hasData;
        d["has_data"] = r.hasData;
        d["valid"] = r.valid;
        d["type"] = r.typeName;
        d["address"] = r.address;
        d["ttl"] = r.ttl;
        if (r.memberTtlOverride)
            d["member_ttl"] = *r.memberTtlOverride;
        d["srvce"] = r.srvce;
        d["proto"] = r.proto;
        d["port"] = r.defaultPort;
        d["ports"] = r.ports;
        return d;
    });

    m.def("domain_to_path", &domainToPath);
    m.def("path_to_domain", &pathToDomain);
    m.def("url_hostname", &urlHostname);

    m.def("json_roundtrip", [](const std::string& text) -> py::object {
        auto v = Json::parse(text);
        if (!v) return py::none();
        return py::str(v->dump());
    });

    py::class_<PyStubEngine>(m, "StubEngine")
        .def(py::init<const std::string&, const std::string&, bool>(),
             py::arg("dns_domain"), py::arg("datacenter_name") = "",
             py::arg("recursion") = false)
        .def("put", &PyStubEngine::put)
        .def("remove", &PyStubEngine::remove)
        .def("set_ready", &PyStubEngine::setReady)
        .def("query_wire", &PyStubEngine::queryWire, py::arg("wire"),
             py::arg("max_size") = 0)
        .def("bench_wire", &PyStubEngine::benchWire, py::arg("wire"),
             py::arg("iters"), py::arg("max_size") = 512)
        .def("query", &PyStubEngine::query, py::arg("name"),
             py::arg("type") = "A", py::arg("rd") = false);
}
