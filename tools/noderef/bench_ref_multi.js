/*
 * Multi-worker launcher for bench_ref.js: forks N workers, holds them
 * at a barrier after warmup, releases them together, and aggregates
 * exactly like bench.py's distributed path (ops summed, elapsed is the
 * MAX over workers, latencies from worker 0).
 *
 * Usage:
 *   NODE_PATH=tools/noderef/shims node tools/noderef/bench_ref_multi.js \
 *       --workers 4 --config headline --steps 10 --warmup 2
 */

'use strict';

const fork = require('child_process').fork;
const path = require('path');

function main() {
	const argv = process.argv.slice(2);
	var workers = 1;
	const passthru = [];
	var config = 'headline', steps = 10, warmup = 2;
	var claimsPerStep = 20000;
	for (var i = 0; i < argv.length; ++i) {
		if (argv[i] === '--workers') {
			workers = Number(argv[++i]);
			continue;
		}
		passthru.push(argv[i]);
		const k = argv[i], v = argv[i + 1];
		if (k === '--config')
			config = v;
		if (k === '--steps')
			steps = Number(v);
		if (k === '--warmup')
			warmup = Number(v);
		if (k === '--claims-per-step')
			claimsPerStep = Number(v);
	}

	const script = path.join(__dirname, 'bench_ref.js');
	const procs = [];
	const resultsBy = {};
	var readyCount = 0;
	var resultCount = 0;

	function finish() {
		var ops = 0, elapsed = 0;
		for (var w = 0; w < workers; ++w) {
			ops += resultsBy[w].ops;
			if (resultsBy[w].elapsed > elapsed)
				elapsed = resultsBy[w].elapsed;
		}
		const out = {
			metric: 'pool claims/sec (8-backend synthetic ' +
			    'TCP set)',
			impl: 'node-cueball reference',
			node: process.version,
			value: Math.round(ops / elapsed * 10) / 10,
			unit: 'claims/s',
			n_gpus: workers,
			steps: steps,
			warmup: warmup,
			ms_per_step: Math.round(elapsed / steps * 1e6) /
			    1e3,
			higher_is_better: true,
			scaling: 'weak',
			dtype: 'n/a',
			data: 'synthetic',
			config: {
				model: 'ConnectionPool claim/release ' +
				    '(config: ' + config + ')',
				global_batch: claimsPerStep * workers,
				parallelism: workers +
				    ' worker processes',
				claim_latency_p50_ms: resultsBy[0].lat_p50,
				claim_latency_p99_ms: resultsBy[0].lat_p99
			}
		};
		console.log(JSON.stringify(out));
		process.exit(0);
	}

	for (var w = 0; w < workers; ++w) {
		(function (idx) {
			const p = fork(script, passthru, {
				env: Object.assign({}, process.env,
				    { BENCH_WORKER: '1' }),
				stdio: ['inherit', 'inherit', 'inherit',
				    'ipc']
			});
			procs.push(p);
			p.on('message', function (m) {
				if (m.type === 'ready') {
					if (++readyCount === workers) {
						procs.forEach(
						    function (q) {
							q.send({
							    type: 'go' });
						});
					}
				} else if (m.type === 'result') {
					resultsBy[idx] = m;
					if (++resultCount === workers)
						finish();
				}
			});
			p.on('exit', function (code) {
				if (resultsBy[idx] === undefined) {
					console.error('worker ' + idx +
					    ' died (code ' + code + ')');
					process.exit(1);
				}
			});
		})(w);
	}
}

main();
