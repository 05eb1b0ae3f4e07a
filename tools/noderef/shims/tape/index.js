/*
 * tape shim: serial async test runner with the assertion surface the
 * reference's suite uses.  Lets the unmodified reference run its own
 * tests under the offline dependency shims, which validates that the
 * shims (mooremachine in particular) implement the semantics the
 * reference depends on.
 */

'use strict';

const EventEmitter = require('events').EventEmitter;
const util = require('util');
const deepEqual = require('assert').deepStrictEqual;
const looseDeepEqual = require('assert').deepEqual;

var queue = [];
var running = false;
var failures = 0;
var assertions = 0;
var testsRun = 0;

function Test(name, cb) {
	EventEmitter.call(this);
	this.t_name = name;
	this.t_cb = cb;
	this.t_ended = false;
}
util.inherits(Test, EventEmitter);

function report(t, ok, msg, extra) {
	assertions++;
	if (ok) {
		if (process.env.TAPE_VERBOSE)
			console.log('ok %d - %s', assertions, msg || '');
		return;
	}
	failures++;
	console.log('not ok %d - %s%s', assertions, msg || 'assertion',
	    extra ? ' (' + extra + ')' : '');
	console.log('  in test: %s', t.t_name);
	const e = new Error('trace');
	console.log(e.stack.split('\n').slice(2, 6).join('\n'));
}

function fmt(v) {
	try {
		return (util.inspect(v, { depth: 3 }));
	} catch (e) {
		return (String(v));
	}
}

Test.prototype.ok = function (v, msg) {
	report(this, !!v, msg || 'ok', v === undefined ? 'undefined' : null);
};
Test.prototype.notOk = function (v, msg) {
	report(this, !v, msg || 'notOk');
};
Test.prototype.equal = function (a, b, msg) {
	/* tape's equal is strict (===) */
	report(this, a === b, msg || 'equal',
	    a === b ? null : fmt(a) + ' !== ' + fmt(b));
};
Test.prototype.strictEqual = Test.prototype.equal;
Test.prototype.is = Test.prototype.equal;
Test.prototype.notEqual = function (a, b, msg) {
	report(this, a !== b, msg || 'notEqual');
};
Test.prototype.notStrictEqual = Test.prototype.notEqual;
Test.prototype.deepEqual = function (a, b, msg) {
	var ok = true, why = null;
	try {
		looseDeepEqual(a, b);
	} catch (e) {
		ok = false;
		why = fmt(a) + ' !~ ' + fmt(b);
	}
	report(this, ok, msg || 'deepEqual', why);
};
Test.prototype.same = Test.prototype.deepEqual;
Test.prototype.ifError = function (err, msg) {
	report(this, !err, msg || 'ifError',
	    err ? String(err && err.message || err) : null);
};
Test.prototype.error = Test.prototype.ifError;
Test.prototype.fail = function (msg) {
	report(this, false, msg || 'fail');
};
Test.prototype.pass = function (msg) {
	report(this, true, msg || 'pass');
};
Test.prototype.throws = function (fn, expected, msg) {
	var threw = false;
	try {
		fn();
	} catch (e) {
		threw = true;
	}
	report(this, threw, msg || 'throws');
};
Test.prototype.doesNotThrow = function (fn, msg) {
	var threw = null;
	try {
		fn();
	} catch (e) {
		threw = e;
	}
	report(this, threw === null, msg || 'doesNotThrow',
	    threw ? String(threw) : null);
};
Test.prototype.comment = function (msg) {
	console.log('# ' + msg);
};
Test.prototype.end = function () {
	if (this.t_ended)
		return;
	this.t_ended = true;
	this.emit('end');
	setImmediate(runNext);
};

function runNext() {
	if (queue.length === 0) {
		running = false;
		console.log('1..%d', assertions);
		console.log('# tests %d', testsRun);
		console.log('# pass  %d', assertions - failures);
		if (failures) {
			console.log('# fail  %d', failures);
			process.exitCode = 1;
		} else {
			console.log('# ok');
		}
		return;
	}
	const t = queue.shift();
	testsRun++;
	if (process.env.TAPE_VERBOSE)
		console.log('# %s', t.t_name);
	try {
		t.t_cb(t);
	} catch (e) {
		failures++;
		console.log('not ok - test "%s" threw: %s', t.t_name,
		    e.stack || e);
		if (!t.t_ended)
			t.end();
	}
}

function test(name, cb) {
	queue.push(new Test(name, cb));
	if (!running) {
		running = true;
		setImmediate(runNext);
	}
}

test.test = test;
test.only = function () {
	throw (new Error('tape shim: .only not supported'));
};
test.skip = function (name) {
	console.log('# SKIP ' + name);
};

module.exports = test;
