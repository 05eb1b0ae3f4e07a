/*
 * mname-client shim: a real (small) DNS client over UDP with the
 * surface node-cueball's resolver uses (lib/resolver.js:385-392,
 * :1210-1377): new DnsClient({concurrency}), lookup(opts, cb) with
 * opts {domain, type, timeout, resolvers, errorThreshold}, message
 * objects exposing getAnswers()/getAuthority()/getAdditionals() with
 * records {type, name, target, port, ttl}, TimeoutError by name, and
 * rcode errors carrying .code ('NXDOMAIN' etc.) aggregated into a
 * MultiError when every resolver fails.
 */

'use strict';

const dgram = require('dgram');
const util = require('util');
const mod_verror = require('verror');

const QTYPE = { A: 1, AAAA: 28, SRV: 33, CNAME: 5, SOA: 6, OPT: 41,
    DNAME: 39 };
const QTYPE_REV = {};
Object.keys(QTYPE).forEach(function (k) { QTYPE_REV[QTYPE[k]] = k; });

const RCODES = ['NOERROR', 'FORMERR', 'SERVFAIL', 'NXDOMAIN', 'NOTIMP',
    'REFUSED'];

function TimeoutError(server, domain) {
	mod_verror.VError.call(this, 'DNS request to %s for %s timed out',
	    server, domain);
}
util.inherits(TimeoutError, mod_verror.VError);
TimeoutError.prototype.name = 'TimeoutError';

function RcodeError(server, domain, code) {
	mod_verror.VError.call(this, 'DNS server %s returned %s for %s',
	    server, code, domain);
	this.code = code;
}
util.inherits(RcodeError, mod_verror.VError);
RcodeError.prototype.name = 'RcodeError';

/* -------- wire codec -------- */

function encodeName(name, buf, off) {
	const labels = name.split('.').filter(function (l) {
		return (l.length > 0);
	});
	labels.forEach(function (l) {
		buf.writeUInt8(l.length, off++);
		buf.write(l, off, 'ascii');
		off += l.length;
	});
	buf.writeUInt8(0, off++);
	return (off);
}

function encodeQuery(id, domain, type) {
	const buf = Buffer.alloc(512);
	buf.writeUInt16BE(id, 0);
	buf.writeUInt16BE(0x0100, 2);	/* RD */
	buf.writeUInt16BE(1, 4);	/* qdcount */
	buf.writeUInt16BE(0, 6);
	buf.writeUInt16BE(0, 8);
	buf.writeUInt16BE(1, 10);	/* arcount: EDNS0 OPT */
	var off = encodeName(domain, buf, 12);
	buf.writeUInt16BE(QTYPE[type], off); off += 2;
	buf.writeUInt16BE(1, off); off += 2;	/* IN */
	/* EDNS0 OPT RR: root name, type 41, class = udp payload 1400 */
	buf.writeUInt8(0, off++);
	buf.writeUInt16BE(41, off); off += 2;
	buf.writeUInt16BE(1400, off); off += 2;
	buf.writeUInt32BE(0, off); off += 4;
	buf.writeUInt16BE(0, off); off += 2;
	return (buf.slice(0, off));
}

function parseName(buf, off) {
	const labels = [];
	var jumped = false;
	var next = off;
	var guard = 0;
	while (guard++ < 128) {
		const len = buf.readUInt8(off);
		if (len === 0) {
			off += 1;
			break;
		}
		if ((len & 0xc0) === 0xc0) {
			const ptr = buf.readUInt16BE(off) & 0x3fff;
			if (!jumped)
				next = off + 2;
			jumped = true;
			off = ptr;
			continue;
		}
		labels.push(buf.toString('ascii', off + 1, off + 1 + len));
		off += 1 + len;
	}
	if (!jumped)
		next = off;
	return ({ name: labels.join('.'), off: next });
}

function parseRR(buf, off) {
	const n = parseName(buf, off);
	off = n.off;
	const type = buf.readUInt16BE(off); off += 2;
	off += 2;	/* class */
	const ttl = buf.readUInt32BE(off); off += 4;
	const rdlen = buf.readUInt16BE(off); off += 2;
	const rr = { name: n.name, type: QTYPE_REV[type] || type,
	    ttl: ttl };
	if (rr.type === 'A' && rdlen === 4) {
		rr.target = buf[off] + '.' + buf[off + 1] + '.' +
		    buf[off + 2] + '.' + buf[off + 3];
	} else if (rr.type === 'AAAA' && rdlen === 16) {
		const parts = [];
		for (var i = 0; i < 16; i += 2)
			parts.push(buf.readUInt16BE(off + i).toString(16));
		rr.target = parts.join(':');
	} else if (rr.type === 'SRV') {
		rr.priority = buf.readUInt16BE(off);
		rr.weight = buf.readUInt16BE(off + 2);
		rr.port = buf.readUInt16BE(off + 4);
		rr.target = parseName(buf, off + 6).name;
	} else if (rr.type === 'CNAME' || rr.type === 'DNAME') {
		rr.target = parseName(buf, off).name;
	}
	/* SOA/OPT/others: type+ttl are all the resolver reads */
	return ({ rr: rr, off: off + rdlen });
}

function Message(buf) {
	this.id = buf.readUInt16BE(0);
	const flags = buf.readUInt16BE(2);
	this.rcode = RCODES[flags & 0x0f] || ('RCODE' + (flags & 0x0f));
	this.truncated = (flags & 0x0200) !== 0;
	const qd = buf.readUInt16BE(4);
	const an = buf.readUInt16BE(6);
	const ns = buf.readUInt16BE(8);
	const ar = buf.readUInt16BE(10);
	var off = 12;
	for (var i = 0; i < qd; ++i) {
		off = parseName(buf, off).off + 4;
	}
	this.m_answers = [];
	this.m_authority = [];
	this.m_additionals = [];
	var p;
	for (i = 0; i < an; ++i) {
		p = parseRR(buf, off);
		this.m_answers.push(p.rr);
		off = p.off;
	}
	for (i = 0; i < ns; ++i) {
		p = parseRR(buf, off);
		this.m_authority.push(p.rr);
		off = p.off;
	}
	for (i = 0; i < ar; ++i) {
		p = parseRR(buf, off);
		if (p.rr.type !== 'OPT')
			this.m_additionals.push(p.rr);
		off = p.off;
	}
}

Message.prototype.getAnswers = function () { return (this.m_answers); };
Message.prototype.getAuthority = function () { return (this.m_authority); };
Message.prototype.getAdditionals = function () {
	return (this.m_additionals);
};

/* -------- client -------- */

function DnsClient(opts) {
	opts = opts || {};
	this.dc_concurrency = opts.concurrency || 3;
}

DnsClient.prototype.lookup = function (opts, cb) {
	const domain = opts.domain;
	const type = opts.type;
	const timeout = opts.timeout || 3000;
	var resolvers = (opts.resolvers || []).slice();
	const errs = [];
	const self = this;
	var inflight = 0;
	var idx = 0;
	var done = false;

	function finish(err, msg) {
		if (done)
			return;
		done = true;
		cb(err, msg);
	}

	function maybeNext() {
		if (done)
			return;
		while (inflight < self.dc_concurrency &&
		    idx < resolvers.length) {
			queryOne(resolvers[idx++]);
		}
		if (inflight === 0 && idx >= resolvers.length) {
			if (errs.length === 1)
				finish(errs[0]);
			else
				finish(new mod_verror.MultiError(errs));
		}
	}

	function queryOne(server) {
		inflight++;
		var host = server, port = 53;
		const m = server.match(/^(.*):(\d+)$/);
		if (m) {
			host = m[1];
			port = parseInt(m[2], 10);
		}
		const id = Math.floor(Math.random() * 65536);
		const q = encodeQuery(id, domain, type);
		const sock = dgram.createSocket(
		    host.indexOf(':') !== -1 ? 'udp6' : 'udp4');
		var timer = setTimeout(function () {
			sock.close();
			inflight--;
			errs.push(new TimeoutError(server, domain));
			maybeNext();
		}, timeout);
		sock.on('error', function (e) {
			clearTimeout(timer);
			try { sock.close(); } catch (e2) { }
			inflight--;
			errs.push(e);
			maybeNext();
		});
		sock.on('message', function (data) {
			var msg;
			try {
				msg = new Message(data);
			} catch (e) {
				return;	/* garbled: wait for timeout */
			}
			if (msg.id !== id)
				return;	/* spoof/stale: ignore */
			clearTimeout(timer);
			sock.close();
			inflight--;
			if (msg.rcode !== 'NOERROR') {
				errs.push(new RcodeError(server, domain,
				    msg.rcode));
				maybeNext();
				return;
			}
			finish(null, msg);
		});
		sock.send(q, 0, q.length, port, host);
	}

	maybeNext();
};

/* DnsMessage(parsedPacket): wrap an already-parsed packet object (the
 * shape mname's Protocol produces; the reference's dns tests build
 * these by hand) in the message interface the resolver consumes. */

const TYPE_NAMES = QTYPE_REV;

function convertRecord(rr) {
	const out = { name: rr.name,
	    type: TYPE_NAMES[rr.rtype] || rr.rtype,
	    ttl: rr.rttl };
	const rd = rr.rdata || {};
	if (out.type === 'SRV') {
		out.priority = rd.priority;
		out.weight = rd.weight;
		out.port = rd.port;
		out.target = rd.target;
	} else if (out.type === 'A' || out.type === 'AAAA' ||
	    out.type === 'CNAME' || out.type === 'DNAME') {
		out.target = rd.target;
	} else if (out.type === 'SOA') {
		out.ttl = rr.rttl;
	}
	return (out);
}

function DnsMessage(parsed) {
	this.id = (parsed.header && parsed.header.id) || 0;
	const flags = (parsed.header && parsed.header.flags) || {};
	const rc = flags.rcode === undefined ? 0 : flags.rcode;
	this.rcode = RCODES[rc] || ('RCODE' + rc);
	this.m_answers = (parsed.answer || []).map(convertRecord);
	this.m_authority = (parsed.authority || []).map(convertRecord);
	this.m_additionals = (parsed.additional || []).map(convertRecord);
}

DnsMessage.prototype.getAnswers = function () {
	return (this.m_answers);
};
DnsMessage.prototype.getAuthority = function () {
	return (this.m_authority);
};
DnsMessage.prototype.getAdditionals = function () {
	return (this.m_additionals);
};
DnsMessage.prototype.toError = function () {
	if (this.rcode === 'NOERROR')
		return (null);
	return (new RcodeError('(parsed)', '(message)', this.rcode));
};

module.exports = { DnsClient: DnsClient, DnsMessage: DnsMessage };
