#include "engine.hpp"

#include <algorithm>

namespace bamd {

using namespace dns;

bool isSuffix(const std::string& suffix, const std::string& str) {
    size_t idx = str.rfind(suffix);
    return idx != std::string::npos && idx + suffix.size() == str.size();
}

std::string stripSuffix(const std::string& suffix, const std::string& str) {
    if (isSuffix(suffix, str))
        return str.substr(0, str.size() - suffix.size()) + "...";
    return str;
}

static bool validNameChars(const std::string& s) {
    for (char c : s) {
        bool ok = (c >= 'a' && c <= 'z') || (c >= '0' && c <= '9') ||
                  c == '_' || c == '.' || c == '-';
        if (!ok) return false;
    }
    return true;
}

/*
 * Parse the SRV shape "_svc._proto.rest" (server.js:141 regex
 * /^(_[^_.]*)[.](_[^_.]*)[.](.*)/ — note each label is '_' followed by
 * zero or more chars that are neither '_' nor '.').
 */
static bool srvShape(const std::string& domain, std::string& service,
                     std::string& protocol, std::string& rest) {
    size_t d1 = domain.find('.');
    if (d1 == std::string::npos || d1 == 0) return false;
    std::string l1 = domain.substr(0, d1);
    size_t d2 = domain.find('.', d1 + 1);
    if (d2 == std::string::npos) return false;
    std::string l2 = domain.substr(d1 + 1, d2 - d1 - 1);
    auto labelOk = [](const std::string& l) {
        if (l.empty() || l[0] != '_') return false;
        for (size_t i = 1; i < l.size(); ++i)
            if (l[i] == '_') return false;  // '.' excluded by split
        return true;
    };
    if (!labelOk(l1) || !labelOk(l2)) return false;
    service = l1;
    protocol = l2;
    rest = domain.substr(d2 + 1);  // may be empty (regex (.*) allows it)
    return true;
}

QueryResult Engine::handle(const Message& query, Message& resp) {
    QueryResult qr;
    resp = Message();
    resp.header.id = query.header.id;
    resp.header.qr = true;
    resp.header.opcode = query.header.opcode;
    resp.header.rd = query.header.rd;
    resp.header.ra = false;  // reference clears RA (server.js:137, 68)
    resp.header.aa = true;
    resp.questions = query.questions;

    if (query.questions.empty()) {
        resp.header.rcode = RCODE_FORMERR;
        return qr;
    }
    if (query.header.opcode != 0) {
        /* only standard QUERY is implemented (IQUERY/STATUS/NOTIFY/
         * UPDATE are not) */
        resp.header.rcode = RCODE_NOTIMP;
        return qr;
    }
    const Question& q = query.questions[0];
    qr.logName = q.name;
    bool rd = query.header.rd;

    /* Dispatch (server.js:491-506): only A, SRV, PTR are served. */
    switch (q.qtype) {
    case TYPE_A:
    case TYPE_SRV:
        resolve(q, rd, resp, qr);
        break;
    case TYPE_PTR:
        resolvePtr(q, rd, resp, qr);
        break;
    default:
        resp.header.rcode = RCODE_NOTIMP;
        break;
    }
    return qr;
}

void Engine::addSoaAuthority(Message& resp, const std::string& name,
                             uint32_t ttl) {
    SoaData soa;
    soa.mname = cfg_.dnsDomain;
    soa.rname = "hostmaster." +
                (cfg_.dnsDomain.empty() ? name : cfg_.dnsDomain);
    soa.minimum = ttl;
    resp.authorities.push_back(Record::SOA(name, std::move(soa), ttl));
}

void Engine::resolvePtr(const Question& q, bool rd, Message& resp,
                        QueryResult& qr) {
    const std::string& domain = q.name;

    /* Split and reverse; require ...in-addr.arpa (server.js:70-78). */
    std::vector<std::string> parts;
    size_t start = 0;
    while (start <= domain.size()) {
        size_t dot = domain.find('.', start);
        if (dot == std::string::npos) {
            parts.push_back(domain.substr(start));
            break;
        }
        parts.push_back(domain.substr(start, dot - start));
        start = dot + 1;
    }
    std::reverse(parts.begin(), parts.end());
    if (parts.size() < 2 || parts[0] != "arpa" || parts[1] != "in-addr") {
        resp.header.rcode = RCODE_REFUSED;
        return;
    }
    std::string ip;
    for (size_t i = 2; i < parts.size(); ++i) {
        if (!ip.empty()) ip.push_back('.');
        ip += parts[i];
    }
    qr.logName = ip;

    if (!store_->ready()) {
        resp.header.rcode = RCODE_SERVFAIL;  // 'eserver'
        return;
    }

    const StoreNode* node = store_->reverseLookup(ip);
    if (node == nullptr) {
        if (cfg_.recursionEnabled && rd) {
            qr.action = QueryResult::Action::Recurse;
            return;
        }
        resp.header.rcode = RCODE_REFUSED;
        return;
    }

    resp.answers.push_back(
        Record::PTR(domain, node->domain(), node->rec().ttl));
}

void Engine::resolve(const Question& q, bool rd, Message& resp,
                     QueryResult& qr) {
    std::string domain = q.name;

    std::string service, protocol;
    bool haveSrv = false;
    {
        std::string svc, proto, rest;
        bool match = srvShape(domain, svc, proto, rest);
        if (q.qtype == TYPE_SRV ||
            (q.qtype == TYPE_ANY && match)) {
            if (!match || rest.empty()) {
                resp.header.rcode = RCODE_REFUSED;
                return;
            }
            service = svc;
            protocol = proto;
            domain = rest;
            haveSrv = true;
            qr.srvLabel = service + "." + protocol;
        }
    }

    /*
     * dnsDomain suffix enforcement — case-sensitive, pre-lowercase, as in
     * the reference (server.js:156-176 run before :207).
     */
    if (!cfg_.dnsDomain.empty()) {
        std::string dotSuffix = "." + cfg_.dnsDomain;
        if (!isSuffix(dotSuffix, domain)) {
            resp.header.rcode = RCODE_REFUSED;
            return;
        }
        std::string strippedBare =
            domain.substr(0, domain.size() - dotSuffix.size());
        qr.logName = strippedBare + "...";
        /* Intended doubled-suffix rejection (see header comment). */
        std::string dcsuff = cfg_.dnsDomain + "." + cfg_.datacenterName;
        if (isSuffix(cfg_.dnsDomain, strippedBare) ||
            isSuffix(dcsuff, strippedBare)) {
            resp.header.rcode = RCODE_REFUSED;
            return;
        }
    }

    if (!store_->ready()) {
        resp.header.rcode = RCODE_SERVFAIL;  // 'eserver'
        return;
    }

    if (domain.empty()) {
        resp.header.rcode = RCODE_REFUSED;
        return;
    }

    toLowerAscii(domain);
    if (!validNameChars(domain)) {
        resp.header.rcode = RCODE_REFUSED;
        return;
    }

    const StoreNode* node = store_->lookup(domain);
    if (node == nullptr) {
        if (cfg_.recursionEnabled && rd) {
            qr.action = QueryResult::Action::Recurse;
            return;
        }
        /* REFUSED, not NXDOMAIN: clients must fall through to their next
         * resolver (policy comment server.js:227-241). */
        resp.header.rcode = RCODE_REFUSED;
        return;
    }

    const CompiledRecord& rec = node->rec();
    if (!rec.hasData || !rec.valid) {
        resp.header.rcode = RCODE_SERVFAIL;
        return;
    }

    uint32_t ttl = rec.ttl;

    if (haveSrv && rec.type != RecType::Service) {
        /* SRV on a non-service name we own: NODATA + SOA authority for
         * negative caching (server.js:276-292). */
        resp.header.rcode = RCODE_NOERROR;
        addSoaAuthority(resp, domain, ttl);
        return;
    }

    switch (rec.type) {
    case RecType::Database:
    case RecType::Host:
    case RecType::DbHost:
    case RecType::LoadBalancer:
    case RecType::MorayHost:
    case RecType::RedisHost:
    case RecType::OpsHost:
    case RecType::RrHost:
        if (!rec.address.empty())
            resp.answers.push_back(Record::A(domain, rec.address, ttl));
        break;

    case RecType::Service: {
        if (haveSrv &&
            (service != rec.srvce || protocol != rec.proto)) {
            /* Wrong service/proto labels on a name we own: NXDOMAIN
             * (server.js:334-345). */
            resp.header.rcode = RCODE_NXDOMAIN;
            return;
        }
        resp.header.rcode = RCODE_NOERROR;

        std::vector<const StoreNode*> kids;
        for (const StoreNode* kid : node->children()) {
            const CompiledRecord& kr = kid->rec();
            if (kr.hasData && recTypeServesUnderService(kr.type))
                kids.push_back(kid);
        }
        /* Fisher-Yates shuffle (server.js:40-53, 361). */
        for (size_t i = kids.size(); i > 1;) {
            --i;
            size_t j = rng_() % (i + 1);
            std::swap(kids[i], kids[j]);
        }

        for (const StoreNode* kid : kids) {
            const CompiledRecord& kr = kid->rec();
            if (!kr.valid) {
                /* Partial answers + SERVFAIL, like the reference's loop
                 * break (server.js:366-376). */
                resp.header.rcode = RCODE_SERVFAIL;
                break;
            }
            if (kr.address.empty()) continue;

            std::vector<uint16_t> ports = kr.ports;
            if (ports.empty())
                ports.push_back(rec.hasDefaultPort ? rec.defaultPort : 0);

            uint32_t rttl = kr.memberTtlOverride.value_or(ttl);

            if (haveSrv) {
                std::string nm = kid->name() + "." + domain;
                for (uint16_t p : ports)
                    resp.answers.push_back(
                        Record::SRV(q.name, nm, p, ttl));
                resp.additionals.push_back(
                    Record::A(nm, kr.address, rttl));
            } else {
                /* Plain A for a service: min of service/member TTLs
                 * (server.js:403-415). */
                if (ttl < rttl) rttl = ttl;
                resp.answers.push_back(
                    Record::A(domain, kr.address, rttl));
            }
        }
        break;
    }

    default:
        /* Unknown record type: reference logs and responds NOERROR with
         * no answers (server.js:419-424). */
        break;
    }
}

}  // namespace bamd
