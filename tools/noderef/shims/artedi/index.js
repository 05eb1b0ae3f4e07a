/*
 * artedi shim: Prometheus-style collector, just the counter surface
 * node-cueball's utils.js uses (createCollector, collector.counter,
 * collector.getCollector, counter.increment).
 */

'use strict';

function labelKey(labels) {
	return (Object.keys(labels || {}).sort().map(function (k) {
		return (k + '=' + labels[k]);
	}).join(','));
}

function Counter(opts) {
	this.c_name = opts.name;
	this.c_help = opts.help;
	this.c_counts = {};
}

Counter.prototype.increment = function (labels) {
	const k = labelKey(labels);
	this.c_counts[k] = (this.c_counts[k] || 0) + 1;
};

Counter.prototype.add = function (n, labels) {
	const k = labelKey(labels);
	this.c_counts[k] = (this.c_counts[k] || 0) + n;
};

function Collector(opts) {
	this.col_labels = (opts && opts.labels) || {};
	this.col_collectors = {};
}

Collector.prototype.counter = function (opts) {
	if (this.col_collectors[opts.name] === undefined)
		this.col_collectors[opts.name] = new Counter(opts);
	return (this.col_collectors[opts.name]);
};

Collector.prototype.getCollector = function (name) {
	return (this.col_collectors[name]);
};

Collector.prototype.collect = function (fmt, cb) {
	var out = '';
	const self = this;
	Object.keys(this.col_collectors).forEach(function (name) {
		const c = self.col_collectors[name];
		out += '# HELP ' + name + ' ' + c.c_help + '\n';
		out += '# TYPE ' + name + ' counter\n';
		Object.keys(c.c_counts).forEach(function (k) {
			out += name + '{' + k + '} ' + c.c_counts[k] + '\n';
		});
	});
	setImmediate(function () { cb(null, out); });
};

module.exports = {
	createCollector: function (opts) {
		return (new Collector(opts));
	},
	FMT_PROM: 'prometheus-0.0.4'
};
