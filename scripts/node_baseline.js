/*
 * Reference-architecture Node.js baseline server (measurement aid).
 *
 * The actual reference (TritonDataCenter/binder) cannot run offline —
 * its npm deps (mname, zkstream, ...) are not vendored and there is no
 * registry access. This single-file server reproduces the reference's
 * per-query work on its hot path so the Node-vs-native comparison is
 * apples-to-apples-shaped:
 *   - single-threaded dgram UDP server (the reference's model);
 *   - DNS wire parse/encode in JS buffers (what mname does);
 *   - records held as raw parsed-JSON objects walked PER QUERY with
 *     the same TTL-precedence/validity/member-filter/shuffle logic as
 *     lib/server.js:249-424 (the reference does not precompile);
 *   - same REFUSED-on-miss policy.
 * If anything, this flatters the baseline: mname does more validation
 * and event plumbing than this file.
 *
 * usage: node node_baseline.js <port> <tree.json>
 */
'use strict';

const dgram = require('dgram');
const fs = require('fs');

const port = parseInt(process.argv[2], 10) || 1053;
const treeFile = process.argv[3];

/* tree: domain -> record object (raw JSON, like tn_data) */
const tree = JSON.parse(fs.readFileSync(treeFile, 'utf8'));
/* children index: domain -> [childDomain...] */
const children = {};
for (const domain of Object.keys(tree)) {
    const idx = domain.indexOf('.');
    if (idx < 0)
        continue;
    const parent = domain.slice(idx + 1);
    if (!(parent in children))
        children[parent] = [];
    children[parent].push(domain);
}

function shuffle(arr) {
    let i = arr.length;
    while (--i > 0) {
        const j = Math.floor(Math.random() * (i + 1));
        const tmp = arr[i];
        arr[i] = arr[j];
        arr[j] = tmp;
    }
    return arr;
}

function parseName(buf, pos) {
    const labels = [];
    while (true) {
        const len = buf[pos];
        if (len === 0) {
            pos += 1;
            break;
        }
        if ((len & 0xc0) !== 0)
            return null;
        labels.push(buf.toString('ascii', pos + 1, pos + 1 + len));
        pos += 1 + len;
        if (pos >= buf.length)
            return null;
    }
    return { name: labels.join('.'), end: pos };
}

function writeName(buf, pos, name) {
    if (name.length > 0) {
        for (const label of name.split('.')) {
            buf[pos] = label.length;
            buf.write(label, pos + 1, 'ascii');
            pos += 1 + label.length;
        }
    }
    buf[pos++] = 0;
    return pos;
}

function ipToBytes(ip) {
    const parts = ip.split('.');
    return [parts[0] | 0, parts[1] | 0, parts[2] | 0, parts[3] | 0];
}

const HOSTISH = {
    db_host: true, host: true, load_balancer: true, moray_host: true,
    redis_host: true, ops_host: true, rr_host: true
};
const SERVICE_MEMBER = {
    load_balancer: true, moray_host: true, ops_host: true,
    rr_host: true, redis_host: true
};

const server = dgram.createSocket('udp4');

server.on('message', (msg, rinfo) => {
    if (msg.length < 17)
        return;
    const id = msg.readUInt16BE(0);
    const q = parseName(msg, 12);
    if (q === null)
        return;
    const qtype = msg.readUInt16BE(q.end);
    let domain = q.name.toLowerCase();

    /* response skeleton: copy header+question */
    const out = Buffer.alloc(512);
    msg.copy(out, 0, 0, q.end + 4);
    out.writeUInt16BE(0x8400, 2);   /* QR|AA */
    out.writeUInt16BE(1, 4);        /* qd */
    let ancount = 0;
    let pos = q.end + 4;
    let rcode = 0;

    function addA(name, ip, ttl) {
        pos = writeName(out, pos, name);
        out.writeUInt16BE(1, pos);          /* A */
        out.writeUInt16BE(1, pos + 2);      /* IN */
        out.writeUInt32BE(ttl, pos + 4);
        out.writeUInt16BE(4, pos + 8);
        const b = ipToBytes(ip);
        out[pos + 10] = b[0];
        out[pos + 11] = b[1];
        out[pos + 12] = b[2];
        out[pos + 13] = b[3];
        pos += 14;
        ancount++;
    }

    function addSrv(name, target, p, ttl) {
        pos = writeName(out, pos, name);
        out.writeUInt16BE(33, pos);
        out.writeUInt16BE(1, pos + 2);
        out.writeUInt32BE(ttl, pos + 4);
        const lenAt = pos + 8;
        pos += 10;
        out.writeUInt16BE(0, pos);          /* prio */
        out.writeUInt16BE(10, pos + 2);     /* weight */
        out.writeUInt16BE(p, pos + 4);
        pos += 6;
        const start = pos;
        pos = writeName(out, pos, target);
        out.writeUInt16BE(6 + (pos - start), lenAt);
        ancount++;
    }

    /* SRV shape handling (server.js:140-154) */
    let service, protocol;
    if (qtype === 33) {
        const m = domain.match(/^(_[^_.]*)[.](_[^_.]*)[.](.*)/);
        if (!m || m[3].length < 1) {
            rcode = 5;
        } else {
            service = m[1];
            protocol = m[2];
            domain = m[3];
        }
    }

    if (rcode === 0) {
        /* per-query raw-object walk, like lib/server.js:249-424 */
        const record = tree[domain];
        if (record === undefined || record === null ||
            typeof (record.type) !== 'string' ||
            typeof (record[record.type]) !== 'object' ||
            record[record.type] === null) {
            rcode = record === undefined ? 5 : 2;
        } else {
            let ttl = 30;
            if (record.ttl !== undefined)
                ttl = record.ttl;
            if (record[record.type].ttl !== undefined)
                ttl = record[record.type].ttl;

            if (HOSTISH[record.type]) {
                addA(domain, record[record.type].address, ttl);
            } else if (record.type === 'database') {
                const u = record.database.primary;
                const host = u.split('@').pop().split(':')[0];
                addA(domain, host, ttl);
            } else if (record.type === 'service') {
                let s = record.service;
                if (typeof (s.service) === 'object')
                    s = s.service;
                if (s.ttl !== undefined)
                    ttl = s.ttl;
                if (service !== undefined &&
                    (service !== s.srvce || protocol !== s.proto)) {
                    rcode = 3;
                } else {
                    const kids = shuffle((children[domain] || [])
                        .filter(function (kd) {
                            const kr = tree[kd];
                            return (kr && SERVICE_MEMBER[kr.type]);
                        }));
                    for (const kd of kids) {
                        const kr = tree[kd];
                        const a = kr[kr.type].address;
                        if (a === null)
                            continue;
                        let rttl = ttl;
                        if (kr.ttl !== undefined)
                            rttl = kr.ttl;
                        if (kr[kr.type].ttl !== undefined)
                            rttl = kr[kr.type].ttl;
                        if (service !== undefined) {
                            const ports = kr[kr.type].ports || [s.port];
                            for (const p of ports)
                                addSrv(q.name, kd, p, ttl);
                        } else {
                            addA(domain, a, Math.min(ttl, rttl));
                        }
                    }
                }
            }
        }
    }

    out.writeUInt16BE(id, 0);
    if (rcode !== 0)
        out[3] = (out[3] & 0xf0) | rcode;
    out.writeUInt16BE(ancount, 6);
    server.send(out.slice(0, pos), rinfo.port, rinfo.address);
});

server.bind(port, '127.0.0.1', () => {
    console.log(JSON.stringify({ listening: port }));
});
