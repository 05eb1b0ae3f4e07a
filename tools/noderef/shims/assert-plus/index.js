/*
 * assert-plus shim: type assertions with optional* variants, for
 * running the reference offline.  Only the checks node-cueball uses.
 */

'use strict';

const assert = require('assert');

function fail(name, type, val) {
	throw (new assert.AssertionError({
		message: (name || 'value') + ' (' + type + ') is required, ' +
		    'got ' + String(val),
		actual: typeof (val),
		expected: type
	}));
}

function mk(type, check) {
	const f = function (val, name) {
		if (!check(val))
			fail(name, type, val);
	};
	const opt = function (val, name) {
		if (val !== undefined && val !== null && !check(val))
			fail(name, 'optional ' + type, val);
	};
	return ({ f: f, opt: opt });
}

const UUID_RE =
    /^[0-9a-f]{8}-[0-9a-f]{4}-[0-9a-f]{4}-[0-9a-f]{4}-[0-9a-f]{12}$/i;

const checks = {
	object: function (v) {
		return (typeof (v) === 'object' && v !== null);
	},
	func: function (v) { return (typeof (v) === 'function'); },
	string: function (v) { return (typeof (v) === 'string'); },
	number: function (v) { return (typeof (v) === 'number'); },
	finite: function (v) {
		return (typeof (v) === 'number' && isFinite(v));
	},
	bool: function (v) { return (typeof (v) === 'boolean'); },
	array: function (v) { return (Array.isArray(v)); },
	buffer: function (v) { return (Buffer.isBuffer(v)); },
	date: function (v) { return (v instanceof Date); },
	regexp: function (v) { return (v instanceof RegExp); },
	uuid: function (v) {
		return (typeof (v) === 'string' && UUID_RE.test(v));
	},
	arrayOfString: function (v) {
		return (Array.isArray(v) && v.every(function (e) {
			return (typeof (e) === 'string');
		}));
	},
	arrayOfObject: function (v) {
		return (Array.isArray(v) && v.every(function (e) {
			return (typeof (e) === 'object' && e !== null);
		}));
	},
	arrayOfNumber: function (v) {
		return (Array.isArray(v) && v.every(function (e) {
			return (typeof (e) === 'number');
		}));
	},
	arrayOfFunc: function (v) {
		return (Array.isArray(v) && v.every(function (e) {
			return (typeof (e) === 'function');
		}));
	}
};

const out = {};
Object.keys(checks).forEach(function (t) {
	const pair = mk(t, checks[t]);
	out[t] = pair.f;
	out['optional' + t.charAt(0).toUpperCase() + t.slice(1)] = pair.opt;
});

out.ok = function (val, msg) {
	if (!val) {
		throw (new assert.AssertionError({
			message: msg || 'assertion failed', actual: val,
			expected: true, operator: '=='
		}));
	}
};
out.equal = assert.equal.bind(assert);
out.strictEqual = assert.strictEqual.bind(assert);
out.notStrictEqual = assert.notStrictEqual.bind(assert);
out.deepEqual = assert.deepEqual.bind(assert);
out.fail = function (msg) {
	throw (new assert.AssertionError({ message: msg || 'fail' }));
};
out.AssertionError = assert.AssertionError;

module.exports = out;
