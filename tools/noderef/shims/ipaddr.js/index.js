/*
 * ipaddr.js shim: parse() for IPv4/IPv6 with toNormalizedString(),
 * used by node-cueball's srvKey (lib/resolver.js:1157-1171) and the
 * static resolver's address validation.
 */

'use strict';

function IPv4(octets) {
	this.octets = octets;
}
IPv4.prototype.kind = function () { return ('ipv4'); };
IPv4.prototype.toString = function () {
	return (this.octets.join('.'));
};
IPv4.prototype.toNormalizedString = IPv4.prototype.toString;

function IPv6(parts) {
	this.parts = parts;
}
IPv6.prototype.kind = function () { return ('ipv6'); };
IPv6.prototype.toNormalizedString = function () {
	return (this.parts.map(function (p) {
		return (p.toString(16));
	}).join(':'));
};
IPv6.prototype.toString = IPv6.prototype.toNormalizedString;
IPv6.prototype.isIPv4MappedAddress = function () {
	return (this.parts[0] === 0 && this.parts[1] === 0 &&
	    this.parts[2] === 0 && this.parts[3] === 0 &&
	    this.parts[4] === 0 && this.parts[5] === 0xffff);
};

function parseV4(str) {
	const m = str.split('.');
	if (m.length !== 4)
		return (null);
	const oct = [];
	for (var i = 0; i < 4; ++i) {
		if (!/^[0-9]{1,3}$/.test(m[i]))
			return (null);
		const v = parseInt(m[i], 10);
		if (v > 255)
			return (null);
		oct.push(v);
	}
	return (new IPv4(oct));
}

function parseV6(str) {
	var s = str;
	/* embedded v4 tail: convert to two hex groups */
	const v4m = s.match(/:(\d+\.\d+\.\d+\.\d+)$/);
	if (v4m) {
		const v4 = parseV4(v4m[1]);
		if (v4 === null)
			return (null);
		s = s.slice(0, s.length - v4m[1].length) +
		    ((v4.octets[0] << 8) | v4.octets[1]).toString(16) + ':' +
		    ((v4.octets[2] << 8) | v4.octets[3]).toString(16);
	}
	var head = s, tail = '';
	const dc = s.indexOf('::');
	if (dc !== -1) {
		if (s.indexOf('::', dc + 1) !== -1)
			return (null);
		head = s.slice(0, dc);
		tail = s.slice(dc + 2);
	}
	const hp = head === '' ? [] : head.split(':');
	const tp = tail === '' ? [] : tail.split(':');
	if (dc === -1 && hp.length !== 8)
		return (null);
	if (dc !== -1 && hp.length + tp.length > 7)
		return (null);
	const parts = [];
	function push(arr) {
		for (var i = 0; i < arr.length; ++i) {
			if (!/^[0-9a-fA-F]{1,4}$/.test(arr[i]))
				return (false);
			parts.push(parseInt(arr[i], 16));
		}
		return (true);
	}
	if (!push(hp))
		return (null);
	if (dc !== -1) {
		const fill = 8 - hp.length - tp.length;
		for (var i = 0; i < fill; ++i)
			parts.push(0);
	}
	if (!push(tp))
		return (null);
	if (parts.length !== 8)
		return (null);
	return (new IPv6(parts));
}

function parse(str) {
	var r = null;
	if (str.indexOf(':') !== -1)
		r = parseV6(str);
	else
		r = parseV4(str);
	if (r === null)
		throw (new Error('ipaddr: the address has neither IPv6 nor ' +
		    'IPv4 format: ' + str));
	return (r);
}

function isValid(str) {
	try {
		parse(str);
		return (true);
	} catch (e) {
		return (false);
	}
}

module.exports = {
	parse: parse,
	isValid: isValid,
	IPv4: IPv4,
	IPv6: IPv6
};
