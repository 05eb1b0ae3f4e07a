/*
 * verror shim: VError with cause chains and printf-style messages,
 * enough for node-cueball's errors.js and resolver/pool call sites.
 * Supported constructor forms:
 *   VError()
 *   VError(fmt, ...args)
 *   VError(cause, fmt, ...args)
 *   VError({cause, constructorOpt, name}, fmt, ...args)
 */

'use strict';

const util = require('util');

function parseArgs(args) {
	var opts = {};
	var fmtArgs = args;
	if (args.length > 0) {
		const a0 = args[0];
		if (a0 instanceof Error) {
			opts = { cause: a0 };
			fmtArgs = args.slice(1);
		} else if (typeof (a0) === 'object' && a0 !== null) {
			opts = a0;
			fmtArgs = args.slice(1);
		}
	}
	var msg = '';
	if (fmtArgs.length > 0)
		msg = util.format.apply(util, fmtArgs);
	return ({ opts: opts, message: msg });
}

function VError() {
	const p = parseArgs(Array.prototype.slice.call(arguments));
	var msg = p.message;
	if (p.opts.cause && p.opts.cause.message)
		msg = msg + ': ' + p.opts.cause.message;
	Error.call(this, msg);
	this.message = msg;
	this.jse_cause = p.opts.cause;
	this.jse_shortmsg = p.message;
	if (Error.captureStackTrace) {
		Error.captureStackTrace(this,
		    p.opts.constructorOpt || VError);
	}
}
util.inherits(VError, Error);
VError.prototype.name = 'VError';

VError.prototype.cause = function () {
	return (this.jse_cause);
};

VError.prototype.toString = function () {
	return ((this.name || 'VError') +
	    (this.message ? ': ' + this.message : ''));
};

VError.cause = function (err) {
	return (err instanceof VError ? (err.jse_cause || null) : null);
};

VError.hasCauseWithName = function (err, name) {
	return (VError.findCauseByName(err, name) !== null);
};

VError.findCauseByName = function (err, name) {
	var e = err;
	while (e) {
		if (e.name === name)
			return (e);
		e = (typeof (e.cause) === 'function') ?
		    e.cause() : e.jse_cause;
	}
	return (null);
};

VError.fullStack = function (err) {
	var out = err.stack || String(err);
	var c = (typeof (err.cause) === 'function') ?
	    err.cause() : err.jse_cause;
	while (c) {
		out += '\ncaused by: ' + (c.stack || String(c));
		c = (typeof (c.cause) === 'function') ?
		    c.cause() : c.jse_cause;
	}
	return (out);
};

function MultiError(errs) {
	VError.call(this, errs && errs[0],
	    'first of %d error%s', errs.length,
	    errs.length === 1 ? '' : 's');
	this.ase_errors = errs;
}
util.inherits(MultiError, VError);
MultiError.prototype.name = 'MultiError';
MultiError.prototype.errors = function () {
	return (this.ase_errors);
};

function WError() {
	VError.apply(this, arguments);
}
util.inherits(WError, VError);
WError.prototype.name = 'WError';

module.exports = VError;
module.exports.VError = VError;
module.exports.MultiError = MultiError;
module.exports.WError = WError;
