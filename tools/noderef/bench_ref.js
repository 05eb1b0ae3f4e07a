/*
 * Head-to-head benchmark harness: drives the UNMODIFIED node-cueball
 * reference (/root/reference/lib, or $CUEBALL_REF) through the same
 * scenarios as this repo's bench.py, with the same parameters, so the
 * two can be compared on the same host.
 *
 * Usage:
 *   NODE_PATH=tools/noderef/shims node tools/noderef/bench_ref.js \
 *       --config headline --steps 10 --warmup 2
 *
 * Prints one JSON line in the same shape as bench.py's output.
 * Scenarios mirror bench.py:
 *   headline - 8 TCP backends, spares=8/max=16, 16 concurrent claimers,
 *              20000 claim/release ops per step
 *   static1  - 1 backend, spares=1/max=2, serial claim/release
 *   codel    - pool max=4, targetClaimDelay=5ms, N queued claims
 *   dns      - DNS-SRV resolver against a local mock DNS, 4 backends,
 *              spares=2/max=8 (uses the shim mname-client UDP path)
 *   agent    - HttpAgent keep-alive over 8 local HTTP backends
 *   cset     - ConnectionSet under backend churn (6 backends)
 */

'use strict';

const net = require('net');
const http = require('http');
const dgram = require('dgram');
const path = require('path');

const REF = process.env.CUEBALL_REF || '/root/reference';
const cueball = require(path.join(REF, 'lib', 'index.js'));

/* ------------------------------------------------------------------ */

function parseArgs() {
	const args = { config: 'headline', steps: 10, warmup: 2,
	    claimsPerStep: 20000, concurrency: 16, spares: 8, maximum: 16,
	    agentConcurrency: 1000, churnInterval: 5.0 };
	const argv = process.argv.slice(2);
	for (var i = 0; i < argv.length; ++i) {
		const k = argv[i].replace(/^--/, '');
		const map = { 'config': 'config', 'steps': 'steps',
		    'warmup': 'warmup', 'claims-per-step': 'claimsPerStep',
		    'concurrency': 'concurrency', 'spares': 'spares',
		    'maximum': 'maximum',
		    'agent-concurrency': 'agentConcurrency',
		    'churn-interval': 'churnInterval' };
		if (map[k] === undefined)
			throw (new Error('unknown flag ' + argv[i]));
		const v = argv[++i];
		args[map[k]] = (map[k] === 'config') ? v : Number(v);
	}
	return (args);
}

function startBackends(n, cb) {
	const servers = [];
	var left = n;
	for (var i = 0; i < n; ++i) {
		const srv = net.createServer(function (c) {
			c.on('error', function () { });
			c.pipe(c);
		});
		servers.push(srv);
		srv.listen(0, '127.0.0.1', function () {
			if (--left === 0)
				cb(servers);
		});
	}
}

function makePool(servers, opts) {
	const res = new cueball.StaticIpResolver({
		backends: servers.map(function (s) {
			return ({ address: '127.0.0.1',
			    port: s.address().port });
		})
	});
	const poolOpts = {
		domain: 'bench.local',
		constructor: function (backend) {
			const sock = net.connect(backend.port,
			    backend.address);
			sock.on('error', function () { });
			return (sock);
		},
		resolver: res,
		recovery: { default: { timeout: 2000, retries: 3,
		    delay: 100, maxDelay: 2000 } },
		spares: opts.spares,
		maximum: opts.maximum
	};
	if (opts.targetClaimDelay !== undefined)
		poolOpts.targetClaimDelay = opts.targetClaimDelay;
	const pool = new cueball.ConnectionPool(poolOpts);
	res.start();
	return ({ pool: pool, res: res });
}

function waitForIdle(pool, want, cb) {
	const t0 = Date.now();
	(function check() {
		if (pool.getStats().idleConnections >= want) {
			cb(null);
			return;
		}
		if (Date.now() - t0 > 15000) {
			cb(new Error('pool never reached ' + want +
			    ' idle connections: ' +
			    JSON.stringify(pool.getStats())));
			return;
		}
		setTimeout(check, 10);
	})();
}

function percentile(sorted, p) {
	if (sorted.length === 0)
		return (null);
	const i = Math.min(sorted.length - 1,
	    Math.floor(sorted.length * p));
	return (sorted[i]);
}

/*
 * C concurrent claim/release chains; mirrors bench.py's ClaimDriver.
 */
function ClaimDriver(pool, concurrency, latencies) {
	this.pool = pool;
	this.concurrency = concurrency;
	this.latencies = latencies;
	this.ops = 0;
	this.target = 0;
	this.active = 0;
	this.doneCb = null;
}

ClaimDriver.prototype.runStep = function (nOps, cb) {
	this.target = nOps;
	this.ops = 0;
	this.doneCb = cb;
	for (var i = 0; i < this.concurrency; ++i) {
		this.active++;
		this.claimOnce();
	}
};

ClaimDriver.prototype.claimOnce = function () {
	const self = this;
	const t0 = process.hrtime();
	this.pool.claim({}, function (err, hdl, conn) {
		if (!err)
			hdl.release();
		const d = process.hrtime(t0);
		self.latencies.push(d[0] * 1000 + d[1] / 1e6);
		self.ops++;
		if (self.ops + self.active - 1 < self.target) {
			self.claimOnce();
		} else {
			self.active--;
			if (self.active === 0) {
				const cb = self.doneCb;
				self.doneCb = null;
				cb();
			}
		}
	});
};

/*
 * Multi-worker barrier: when forked by bench_ref_multi.js
 * (BENCH_WORKER=1), wait for every worker to finish warmup before the
 * timed region starts — the analog of bench.py's gloo barrier.
 */
function barrier(cb) {
	if (process.env.BENCH_WORKER !== '1' ||
	    process.send === undefined) {
		cb();
		return;
	}
	process.once('message', function onMsg(m) {
		if (m && m.type === 'go') {
			cb();
			return;
		}
		process.once('message', onMsg);
	});
	process.send({ type: 'ready' });
}

function runSteps(driver, args, results, done) {
	var i = 0;
	function warm() {
		if (i++ < args.warmup) {
			driver.runStep(args.claimsPerStep, warm);
			return;
		}
		barrier(function () {
			driver.latencies.length = 0;
			const t0 = process.hrtime();
			var j = 0;
			function step() {
				if (j++ < args.steps) {
					driver.runStep(args.claimsPerStep,
					    step);
					return;
				}
				const d = process.hrtime(t0);
				results.elapsed = d[0] + d[1] / 1e9;
				results.ops = args.steps *
				    args.claimsPerStep;
				done();
			}
			step();
		});
	}
	warm();
}

function finishLatencies(results, latencies) {
	const sorted = latencies.slice().sort(function (a, b) {
		return (a - b);
	});
	results.lat_p50 = percentile(sorted, 0.50);
	results.lat_p99 = percentile(sorted, 0.99);
}

/* ------------------------------------------------------------------ */

function scenarioHeadline(args, results, done) {
	startBackends(8, function (servers) {
		const pr = makePool(servers, { spares: args.spares,
		    maximum: args.maximum });
		waitForIdle(pr.pool, Math.min(args.spares, 8),
		    function (err) {
			if (err)
				throw (err);
			const lat = [];
			const driver = new ClaimDriver(pr.pool,
			    args.concurrency, lat);
			runSteps(driver, args, results, function () {
				finishLatencies(results, lat);
				pr.pool.stop();
				servers.forEach(function (s) {
					s.close();
				});
				done();
			});
		});
	});
}

function scenarioStatic1(args, results, done) {
	startBackends(1, function (servers) {
		const pr = makePool(servers, { spares: 1, maximum: 2 });
		waitForIdle(pr.pool, 1, function (err) {
			if (err)
				throw (err);
			const lat = [];
			const driver = new ClaimDriver(pr.pool, 1, lat);
			runSteps(driver, args, results, function () {
				finishLatencies(results, lat);
				pr.pool.stop();
				servers.forEach(function (s) {
					s.close();
				});
				done();
			});
		});
	});
}

function scenarioCodel(args, results, done) {
	startBackends(4, function (servers) {
		const pr = makePool(servers, { spares: 4, maximum: 4,
		    targetClaimDelay: 5 });
		waitForIdle(pr.pool, 4, function (err) {
			if (err)
				throw (err);
			const n = args.claimsPerStep;
			var ok = 0, shed = 0;
			const lat = [];
			const t0 = process.hrtime();
			function fin() {
				if (ok + shed !== n)
					return;
				const d = process.hrtime(t0);
				results.elapsed = d[0] + d[1] / 1e9;
				results.ops = n;
				results.shed = shed;
				finishLatencies(results, lat);
				pr.pool.stop();
				servers.forEach(function (s) {
					s.close();
				});
				done();
			}
			for (var i = 0; i < n; ++i) {
				(function () {
					const c0 = process.hrtime();
					pr.pool.claim({},
					    function (err2, hdl) {
						const dd =
						    process.hrtime(c0);
						lat.push(dd[0] * 1000 +
						    dd[1] / 1e6);
						if (err2) {
							shed++;
						} else {
							ok++;
							setTimeout(
							    function () {
								hdl.
								release();
							}, 0.5);
						}
						fin();
					});
				})();
			}
		});
	});
}

/*
 * Minimal mock DNS server speaking enough of the protocol for the shim
 * mname-client and the reference resolver: answers SRV for the bench
 * service with 4 targets (+ A additionals), and A for b<N> names.
 */
function MockDns(ports, cb) {
	const sock = dgram.createSocket('udp4');
	this.sock = sock;

	function nameAt(buf, off) {
		const labels = [];
		while (buf[off] !== 0) {
			const len = buf[off];
			labels.push(buf.toString('ascii', off + 1,
			    off + 1 + len));
			off += 1 + len;
		}
		return ({ name: labels.join('.'), off: off + 1 });
	}

	function encName(name, buf, off) {
		name.split('.').forEach(function (l) {
			buf.writeUInt8(l.length, off++);
			buf.write(l, off, 'ascii');
			off += l.length;
		});
		buf.writeUInt8(0, off++);
		return (off);
	}

	sock.on('message', function (q, rinfo) {
		const id = q.readUInt16BE(0);
		const qn = nameAt(q, 12);
		const qtype = q.readUInt16BE(qn.off);
		const resp = Buffer.alloc(4096);
		resp.writeUInt16BE(id, 0);
		resp.writeUInt16BE(0x8180, 2);	/* QR RD RA NOERROR */
		resp.writeUInt16BE(1, 4);
		var ancount = 0, arcount = 0;
		var off = encName(qn.name, resp, 12);
		resp.writeUInt16BE(qtype, off); off += 2;
		resp.writeUInt16BE(1, off); off += 2;
		if (qtype === 33 && qn.name.indexOf('_bench') === 0) {
			/* SRV + A additionals */
			var aoff = off;
			ports.forEach(function (p, i) {
				aoff = encName(qn.name, resp, aoff);
				resp.writeUInt16BE(33, aoff); aoff += 2;
				resp.writeUInt16BE(1, aoff); aoff += 2;
				resp.writeUInt32BE(60, aoff); aoff += 4;
				const tgt = 'b' + i + '.svc.bench';
				const rdl = 6 + tgt.length + 2;
				resp.writeUInt16BE(rdl, aoff); aoff += 2;
				resp.writeUInt16BE(10, aoff); aoff += 2;
				resp.writeUInt16BE(10, aoff); aoff += 2;
				resp.writeUInt16BE(p, aoff); aoff += 2;
				aoff = encName(tgt, resp, aoff);
				ancount++;
			});
			off = aoff;
		} else if (qtype === 1 && /^b\d\.svc\.bench$/.
		    test(qn.name)) {
			off = encName(qn.name, resp, off);
			resp.writeUInt16BE(1, off); off += 2;
			resp.writeUInt16BE(1, off); off += 2;
			resp.writeUInt32BE(60, off); off += 4;
			resp.writeUInt16BE(4, off); off += 2;
			resp.writeUInt8(127, off++);
			resp.writeUInt8(0, off++);
			resp.writeUInt8(0, off++);
			resp.writeUInt8(1, off++);
			ancount++;
		} else if (qtype === 28) {
			/* AAAA: NODATA */
		}
		resp.writeUInt16BE(ancount, 6);
		resp.writeUInt16BE(0, 8);
		resp.writeUInt16BE(arcount, 10);
		sock.send(resp.slice(0, off), 0, off, rinfo.port,
		    rinfo.address);
	});
	/*
	 * The reference resolver only accepts bare IPs in `resolvers`
	 * (port 53 implied, lib/resolver.js:465-474), so the mock must
	 * listen on 127.0.0.1:53 (we run as root on the bench box).
	 */
	sock.on('error', function (e) {
		console.error('mock DNS bind failed: ' + e.message);
		process.exit(1);
	});
	sock.bind(53, '127.0.0.1', function () {
		cb(sock.address().port);
	});
}

function scenarioDns(args, results, done) {
	startBackends(4, function (servers) {
		const ports = servers.map(function (s) {
			return (s.address().port);
		});
		new MockDns(ports, function (dnsPort) {
			const res = new cueball.DNSResolver({
				domain: 'svc.bench',
				service: '_bench._tcp',
				resolvers: ['127.0.0.1'],
				recovery: { default: { timeout: 2000,
				    retries: 3, delay: 100,
				    maxDelay: 2000 } }
			});
			const pool = new cueball.ConnectionPool({
				domain: 'svc.bench',
				constructor: function (backend) {
					const s = net.connect(backend.port,
					    backend.address);
					s.on('error', function () { });
					return (s);
				},
				resolver: res,
				recovery: { default: { timeout: 2000,
				    retries: 3, delay: 100,
				    maxDelay: 2000 } },
				spares: 2,
				maximum: 8
			});
			res.start();
			waitForIdle(pool, 2, function (err) {
				if (err)
					throw (err);
				const lat = [];
				const driver = new ClaimDriver(pool,
				    args.concurrency, lat);
				runSteps(driver, args, results,
				    function () {
					finishLatencies(results, lat);
					pool.stop();
					res.stop();
					servers.forEach(function (s) {
						s.close();
					});
					done();
				});
			});
		});
	});
}

function scenarioAgent(args, results, done) {
	const servers = [];
	var left = 8;
	for (var i = 0; i < 8; ++i) {
		const srv = http.createServer(function (req, resp) {
			resp.writeHead(200,
			    { 'content-type': 'text/plain' });
			resp.end('ok');
		});
		servers.push(srv);
		srv.listen(0, '127.0.0.1', function () {
			if (--left === 0)
				ready();
		});
	}
	function ready() {
		const port0 = servers[0].address().port;
		const agent = new cueball.HttpAgent({
			defaultPort: port0,
			recovery: { default: { timeout: 2000, retries: 3,
			    delay: 100, maxDelay: 2000 } },
			spares: 8,
			maximum: 32,
			resolvers: ['127.0.0.1']
		});
		/*
		 * The reference agent creates a pool per host via its
		 * resolver factory; to use our 8 local backends inject a
		 * pre-created pool the same way the rebuild's bench does
		 * (createPool with a static resolver listing all 8).
		 */
		const res = new cueball.StaticIpResolver({
			backends: servers.map(function (s) {
				return ({ address: '127.0.0.1',
				    port: s.address().port });
			})
		});
		agent.createPool('svc.bench', { resolver: res });
		/* external resolvers are started by the caller */
		res.start();
		const pool = agent.getPool('svc.bench');
		waitForIdle(pool, 8, function (err) {
			if (err)
				throw (err);
			const lat = [];
			const conc = args.agentConcurrency;
			function oneGet(cb) {
				const t0 = process.hrtime();
				const req = http.request({
					host: 'svc.bench',
					path: '/x',
					agent: agent
				}, function (resp) {
					resp.resume();
					resp.on('end', function () {
						const d =
						    process.hrtime(t0);
						lat.push(d[0] * 1000 +
						    d[1] / 1e6);
						cb();
					});
				});
				req.on('error', function (e) {
					cb();
				});
				req.end();
			}
			function step(cb) {
				var pending = conc;
				for (var j = 0; j < conc; ++j) {
					oneGet(function () {
						if (--pending === 0)
							cb();
					});
				}
			}
			var w = 0;
			function warm() {
				if (w++ < args.warmup) {
					step(warm);
					return;
				}
				lat.length = 0;
				const t0 = process.hrtime();
				var s = 0;
				function run() {
					if (s++ < args.steps) {
						step(run);
						return;
					}
					const d = process.hrtime(t0);
					results.elapsed = d[0] +
					    d[1] / 1e9;
					results.ops = args.steps * conc;
					finishLatencies(results, lat);
					agent.stop(function () {
						servers.forEach(
						    function (sv) {
							sv.close();
						});
						done();
					});
				}
				run();
			}
			warm();
		});
	}
}

function scenarioCset(args, results, done) {
	startBackends(6, function (servers) {
		/*
		 * DummyResolver equivalent: a bare emitter wrapped by the
		 * reference Resolver FSM contract — the reference accepts
		 * any object with start/stop/count/on; emit added/removed
		 * by hand for churn.
		 */
		const EventEmitter = require('events').EventEmitter;
		const res = new EventEmitter();
		const resBackends = {};
		res.start = function () { };
		res.stop = function () { };
		res.count = function () {
			return (Object.keys(resBackends).length);
		};
		res.list = function () {
			const out = {};
			Object.keys(resBackends).forEach(function (k) {
				out[k] = resBackends[k];
			});
			return (out);
		};
		res.getLastError = function () { return (undefined); };
		res.isInState = function (st) {
			return (st === 'running');
		};
		res.getState = function () { return ('running'); };
		res.addBackend = function (k, be) {
			resBackends[k] = be;
			res.emit('added', k, be);
		};
		res.removeBackend = function (k) {
			delete resBackends[k];
			res.emit('removed', k);
		};

		const cset = new cueball.ConnectionSet({
			constructor: function (backend) {
				const s = net.connect(backend.port,
				    backend.address);
				s.on('error', function () { });
				return (s);
			},
			resolver: res,
			recovery: { default: { timeout: 2000, retries: 3,
			    delay: 100, maxDelay: 2000 } },
			target: 6,
			maximum: 8
		});
		const live = {};
		var opsN = 0;
		cset.on('added', function (ck, conn, hdl) {
			live[ck] = { conn: conn, hdl: hdl };
		});
		cset.on('removed', function (ck, conn, hdl) {
			delete live[ck];
			hdl.release();
		});
		servers.forEach(function (s, i) {
			res.addBackend('b' + i, {
				name: 'b' + i,
				address: '127.0.0.1',
				port: s.address().port
			});
		});

		const t0 = Date.now();
		function waitLive() {
			if (Object.keys(live).length >= 6 ||
			    Date.now() - t0 > 10000) {
				startLoad();
				return;
			}
			setTimeout(waitLive, 10);
		}
		function startLoad() {
			var stop = false;
			const buf = Buffer.alloc(64, 0x78);
			function loadTick() {
				if (stop)
					return;
				Object.keys(live).forEach(function (ck) {
					const c = live[ck].conn;
					if (!c.destroyed) {
						try {
							c.write(buf);
							opsN++;
						} catch (e) { }
					}
				});
				setImmediate(loadTick);
			}
			var ci = 0;
			function churnTick() {
				if (stop)
					return;
				setTimeout(function () {
					if (stop)
						return;
					for (var j = 0; j < 2; ++j) {
						res.removeBackend(
						    'b' + ((ci + j) % 6));
					}
					setTimeout(function () {
						if (stop)
							return;
						for (var j = 0; j < 2;
						    ++j) {
							const k =
							    (ci + j) % 6;
							res.addBackend(
							    'b' + k, {
							name: 'b' + k,
							address:
							    '127.0.0.1',
							port: servers[k].
							    address().port
							});
						}
						ci = (ci + 2) % 6;
						churnTick();
					}, args.churnInterval * 1000);
				}, args.churnInterval * 1000);
			}
			const tl0 = process.hrtime();
			loadTick();
			churnTick();
			setTimeout(function () {
				stop = true;
				const d = process.hrtime(tl0);
				results.elapsed = d[0] + d[1] / 1e9;
				results.ops = opsN;
				results.lat_p50 = null;
				results.lat_p99 = null;
				cset.stop();
				servers.forEach(function (s) {
					s.close();
				});
				setTimeout(done, 200);
			}, args.steps * 1000);
		}
		waitLive();
	});
}

/* ------------------------------------------------------------------ */

const SCENARIOS = {
	headline: scenarioHeadline,
	static1: scenarioStatic1,
	codel: scenarioCodel,
	dns: scenarioDns,
	agent: scenarioAgent,
	cset: scenarioCset
};

function main() {
	const args = parseArgs();
	const results = {};
	const fn = SCENARIOS[args.config];
	if (fn === undefined)
		throw (new Error('unknown config ' + args.config));
	fn(args, results, function () {
		if (process.env.BENCH_WORKER === '1' &&
		    process.send !== undefined) {
			process.send({ type: 'result',
			    elapsed: results.elapsed, ops: results.ops,
			    lat_p50: results.lat_p50,
			    lat_p99: results.lat_p99,
			    shed: results.shed });
			setTimeout(function () { process.exit(0); }, 200);
			return;
		}
		const value = results.ops / results.elapsed;
		const out = {
			metric: 'pool claims/sec (8-backend synthetic ' +
			    'TCP set)',
			impl: 'node-cueball reference @ ' + REF,
			node: process.version,
			value: Math.round(value * 10) / 10,
			unit: 'claims/s',
			n_gpus: 1,
			steps: args.steps,
			warmup: args.warmup,
			ms_per_step: Math.round(results.elapsed /
			    args.steps * 1e6) / 1e3,
			higher_is_better: true,
			dtype: 'n/a',
			data: 'synthetic',
			config: {
				model: 'ConnectionPool claim/release ' +
				    '(config: ' + args.config + ')',
				global_batch: args.claimsPerStep,
				parallelism: '1 worker process',
				backends: 8,
				spares: args.spares,
				maximum: args.maximum,
				concurrency: args.concurrency,
				claim_latency_p50_ms: results.lat_p50 ===
				    null ? null :
				    Math.round(results.lat_p50 * 1e4) / 1e4,
				claim_latency_p99_ms: results.lat_p99 ===
				    null ? null :
				    Math.round(results.lat_p99 * 1e4) / 1e4
			}
		};
		if (results.shed !== undefined)
			out.config.shed_claims = results.shed;
		console.log(JSON.stringify(out));
		setTimeout(function () { process.exit(0); }, 200);
	});
}

main();
