/*
 * bunyan shim: child loggers + leveled logging.  Default level 'warn'
 * (overridable with LOG_LEVEL) writing one-line JSON to stderr, so the
 * benchmark is not perturbed by log formatting, matching how the
 * reference is benchmarked in production (bunyan at info/warn).
 */

'use strict';

const LEVELS = { trace: 10, debug: 20, info: 30, warn: 40, error: 50,
    fatal: 60 };

function levelOf(v) {
	if (typeof (v) === 'number')
		return (v);
	return (LEVELS[v] || 40);
}

function Logger(fields, level) {
	this.l_fields = fields || {};
	this.l_level = level;
}

Logger.prototype.child = function (fields) {
	const merged = {};
	const self = this;
	Object.keys(this.l_fields).forEach(function (k) {
		merged[k] = self.l_fields[k];
	});
	if (fields) {
		Object.keys(fields).forEach(function (k) {
			merged[k] = fields[k];
		});
	}
	return (new Logger(merged, this.l_level));
};

Logger.prototype.level = function (v) {
	if (v === undefined)
		return (this.l_level);
	this.l_level = levelOf(v);
	return (this.l_level);
};

function logAt(lvlName, lvlNum) {
	return (function () {
		if (lvlNum < this.l_level)
			return (false);
		const rec = { level: lvlNum, name: lvlName,
		    time: new Date().toISOString() };
		const self = this;
		Object.keys(this.l_fields).forEach(function (k) {
			rec[k] = self.l_fields[k];
		});
		var args = Array.prototype.slice.call(arguments);
		if (args.length > 0 && typeof (args[0]) === 'object' &&
		    args[0] !== null) {
			const extra = args.shift();
			Object.keys(extra).forEach(function (k) {
				if (extra[k] instanceof Error)
					rec[k] = String(extra[k]);
				else
					rec[k] = extra[k];
			});
		}
		rec.msg = require('util').format.apply(null, args);
		try {
			process.stderr.write(JSON.stringify(rec) + '\n');
		} catch (e) {
			/* circular structures etc.: drop the record */
		}
		return (true);
	});
}

Object.keys(LEVELS).forEach(function (name) {
	Logger.prototype[name] = logAt(name, LEVELS[name]);
});

function createLogger(opts) {
	opts = opts || {};
	const level = levelOf(process.env.LOG_LEVEL || opts.level || 'warn');
	const fields = {};
	Object.keys(opts).forEach(function (k) {
		if (k !== 'level' && k !== 'stream' && k !== 'streams' &&
		    k !== 'serializers')
			fields[k] = opts[k];
	});
	return (new Logger(fields, level));
}

module.exports = {
	createLogger: createLogger,
	TRACE: 10, DEBUG: 20, INFO: 30, WARN: 40, ERROR: 50, FATAL: 60
};
