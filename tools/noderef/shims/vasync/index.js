/*
 * vasync shim: forEachParallel / forEachPipeline, the only entry
 * points node-cueball uses (teardown fan-in paths).
 */

'use strict';

function forEachParallel(opts, cb) {
	const inputs = opts.inputs;
	const func = opts.func;
	const results = {
		operations: [],
		successes: [],
		ndone: 0,
		nerrors: 0
	};
	var firstErr = null;
	if (inputs.length === 0) {
		setImmediate(function () { cb(null, results); });
		return (results);
	}
	var remaining = inputs.length;
	inputs.forEach(function (input, i) {
		const op = { input: input, status: 'pending',
		    err: null, result: null };
		results.operations[i] = op;
		func(input, function (err, res) {
			op.status = err ? 'fail' : 'ok';
			op.err = err || null;
			op.result = res;
			results.ndone++;
			if (err) {
				results.nerrors++;
				if (firstErr === null)
					firstErr = err;
			} else {
				results.successes.push(res);
			}
			if (--remaining === 0)
				cb(firstErr, results);
		});
	});
	return (results);
}

function forEachPipeline(opts, cb) {
	const inputs = opts.inputs.slice();
	const func = opts.func;
	const results = {
		operations: [], successes: [], ndone: 0, nerrors: 0
	};
	var i = 0;
	function next(err) {
		if (err || i >= inputs.length) {
			cb(err || null, results);
			return;
		}
		const input = inputs[i++];
		func(input, function (e, res) {
			results.ndone++;
			if (e)
				results.nerrors++;
			else
				results.successes.push(res);
			next(e);
		});
	}
	next(null);
	return (results);
}

function barrier() {
	const EventEmitter = require('events').EventEmitter;
	const b = new EventEmitter();
	b.pending = {};
	var count = 0;
	b.start = function (name) {
		if (!b.pending[name]) {
			b.pending[name] = true;
			count++;
		}
	};
	b.done = function (name) {
		if (b.pending[name]) {
			delete b.pending[name];
			if (--count === 0) {
				setImmediate(function () {
					if (count === 0)
						b.emit('drain');
				});
			}
		}
	};
	return (b);
}

module.exports = {
	forEachParallel: forEachParallel,
	forEachPipeline: forEachPipeline,
	barrier: barrier
};
