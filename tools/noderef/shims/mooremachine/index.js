/*
 * Minimal Moore-FSM runtime, API-compatible with the subset of the
 * `mooremachine` npm package that node-cueball uses.  Written from the
 * semantics documented in SURVEY.md §2.2 and pinned by the rebuild's
 * FSM test suite (cueball_amd/fsm.py is the Python twin):
 *
 *  - state-entry functions `state_<name>(S)`; dotted sub-states are
 *    properties on the parent entry function (state_stopping.backends);
 *  - the S handle scopes listeners/timers to the state: everything
 *    registered through it is disconnected on state exit;
 *  - S.gotoState on an exited scope is a no-op (the transition it
 *    wanted is obsolete);
 *  - gotoState during the entry function is deferred until the entry
 *    function returns;
 *  - 'stateChanged' events are emitted asynchronously, in order;
 *  - validTransitions is asserted on the *next* transition.
 *
 * This file exists so the reference implementation can be executed
 * head-to-head against the rebuild without network access to npm.
 */

'use strict';

const EventEmitter = require('events').EventEmitter;
const util = require('util');

function Scope(fsm) {
	this.sc_fsm = fsm;
	this.sc_listeners = [];
	this.sc_timers = [];
	this.sc_active = true;
	this.sc_valid = undefined;
}

Scope.prototype.on = function (emitter, evt, cb) {
	if (!this.sc_active)
		throw (new Error('S.on() used on exited state scope'));
	emitter.on(evt, cb);
	this.sc_listeners.push([emitter, evt, cb]);
};

Scope.prototype.timeout = function (ms, cb) {
	if (!this.sc_active)
		throw (new Error('S.timeout() used on exited state scope'));
	const self = this;
	const h = setTimeout(function () {
		if (self.sc_active)
			cb();
	}, ms);
	this.sc_timers.push(h);
	return (h);
};

Scope.prototype.interval = function (ms, cb) {
	if (!this.sc_active)
		throw (new Error('S.interval() used on exited state scope'));
	const self = this;
	const h = setInterval(function () {
		if (self.sc_active)
			cb();
	}, ms);
	this.sc_timers.push(h);
	return (h);
};

Scope.prototype.immediate = function (cb) {
	if (!this.sc_active)
		throw (new Error('S.immediate() used on exited state scope'));
	const self = this;
	const h = setImmediate(function () {
		if (self.sc_active)
			cb();
	});
	this.sc_timers.push(h);
	return (h);
};

Scope.prototype.callback = function (cb) {
	const self = this;
	return (function () {
		if (self.sc_active)
			return (cb.apply(this, arguments));
		return (undefined);
	});
};

Scope.prototype.validTransitions = function (states) {
	this.sc_fsm.fsm_valid = states;
};

Scope.prototype.gotoState = function (state) {
	if (!this.sc_active) {
		/* stale handler in the same synchronous cascade */
		return;
	}
	this.sc_fsm.gotoState(state);
};

Scope.prototype.gotoStateOn = function (emitter, evt, state) {
	const self = this;
	this.on(emitter, evt, function () {
		self.gotoState(state);
	});
};

Scope.prototype.gotoStateTimeout = function (ms, state) {
	const self = this;
	this.timeout(ms, function () {
		self.gotoState(state);
	});
};

Scope.prototype.dispose = function () {
	this.sc_active = false;
	var i;
	for (i = 0; i < this.sc_listeners.length; ++i) {
		const l = this.sc_listeners[i];
		l[0].removeListener(l[1], l[2]);
	}
	this.sc_listeners = [];
	for (i = 0; i < this.sc_timers.length; ++i) {
		const t = this.sc_timers[i];
		clearTimeout(t);
		clearInterval(t);
		clearImmediate(t);
	}
	this.sc_timers = [];
};

function FSM(initialState) {
	EventEmitter.call(this);
	this.fsm_state = undefined;
	this.fsm_scope = undefined;
	this.fsm_valid = undefined;
	this.fsm_entering = false;
	this.fsm_pending = undefined;
	this.fsm_history = [];
	this.fsm_emitQueue = [];
	this.fsm_emitScheduled = false;
	this.gotoState(initialState);
}
util.inherits(FSM, EventEmitter);

FSM.prototype.getState = function () {
	return (this.fsm_state);
};

FSM.prototype.isInState = function (state) {
	const cur = this.fsm_state;
	if (cur === undefined)
		return (false);
	return (cur === state || cur.indexOf(state + '.') === 0);
};

FSM.prototype.allStateEvent = function () {
	/* accepted but unused by cueball */
};

FSM.prototype._entryFor = function (state) {
	const parts = state.split('.');
	var f = this['state_' + parts[0]];
	for (var i = 1; i < parts.length && f !== undefined; ++i)
		f = f[parts[i]];
	if (typeof (f) !== 'function') {
		throw (new Error(this.constructor.name +
		    ' has no state-entry function for "' + state + '"'));
	}
	return (f);
};

FSM.prototype.gotoState = function (state) {
	const valid = this.fsm_valid;
	if (valid !== undefined && valid.indexOf(state) === -1) {
		throw (new Error(this.constructor.name +
		    ': invalid transition "' + this.fsm_state + '" -> "' +
		    state + '" (valid: ' + valid.join(',') + ')'));
	}
	if (this.fsm_entering) {
		if (this.fsm_pending !== undefined &&
		    this.fsm_pending !== state) {
			throw (new Error(this.constructor.name +
			    ': conflicting deferred transitions "' +
			    this.fsm_pending + '" and "' + state + '"'));
		}
		this.fsm_pending = state;
		return;
	}
	var next = state;
	while (next !== undefined) {
		const target = next;
		next = undefined;
		if (this.fsm_scope !== undefined)
			this.fsm_scope.dispose();
		this.fsm_valid = undefined;
		this.fsm_state = target;
		this.fsm_history.push(target);
		if (this.fsm_history.length > 8)
			this.fsm_history.shift();
		const scope = new Scope(this);
		this.fsm_scope = scope;
		const entry = this._entryFor(target);
		this.fsm_entering = true;
		try {
			entry.call(this, scope);
		} finally {
			this.fsm_entering = false;
		}
		this._queueStateChanged(target);
		if (this.fsm_pending !== undefined) {
			const pend = this.fsm_pending;
			this.fsm_pending = undefined;
			const v = this.fsm_valid;
			if (v !== undefined && v.indexOf(pend) === -1) {
				throw (new Error(this.constructor.name +
				    ': invalid transition "' + target +
				    '" -> "' + pend + '"'));
			}
			next = pend;
		}
	}
};

FSM.prototype._queueStateChanged = function (state) {
	this.fsm_emitQueue.push(state);
	if (!this.fsm_emitScheduled) {
		this.fsm_emitScheduled = true;
		const self = this;
		setImmediate(function () {
			self.fsm_emitScheduled = false;
			while (self.fsm_emitQueue.length > 0) {
				const st = self.fsm_emitQueue.shift();
				self.emit('stateChanged', st);
			}
		});
	}
};

module.exports = { FSM: FSM };
