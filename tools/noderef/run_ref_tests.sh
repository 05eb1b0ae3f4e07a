#!/bin/bash
# Run the unmodified reference's own tape test suite under the offline
# dependency shims.  This validates that the shims (mooremachine's FSM
# semantics in particular) faithfully implement what the reference
# depends on — which is what makes the head-to-head benchmark a fair
# comparison.
#
# Covered: utils, pool, cset, codel, resolver_for, resolver_static
# (510 assertions).  Skipped: dns.test.js (needs the `mname` DNS
# *server* package to synthesize wire responses), agent/monitor
# (need restify/kang/sshpk HTTP stacks).  Those code paths are
# exercised on the benchmark side by bench_ref.js's dns/agent
# scenarios instead.
#
# Usage: bash tools/noderef/run_ref_tests.sh [reference-dir]

set -e
HERE="$(cd "$(dirname "$0")" && pwd)"
REF="${1:-/root/reference}"
if [ ! -d "$REF/lib" ]; then
    WORK="$(mktemp -d)"
    tar xzf "$HERE/reference-cueball-2.10.3.tgz" -C "$WORK"
    REF="$WORK/reference"
    echo "note: using vendored reference sources (no test/ dir there)" >&2
fi
if [ ! -d "$REF/test" ]; then
    echo "reference test suite not available at $REF/test" >&2
    exit 2
fi

STAGE="$(mktemp -d)"
cp -r "$REF/lib" "$REF/test" "$STAGE/"
export NODE_PATH="$HERE/shims"

fails=0
for f in utils codel resolver_for resolver_static pool cset; do
    echo "== $f.test.js =="
    if ! (cd "$STAGE" && timeout 600 node "test/$f.test.js" | tail -4)
    then
        fails=$((fails+1))
    fi
done
rm -rf "$STAGE"
if [ "$fails" -ne 0 ]; then
    echo "REFERENCE SUITE FAILURES: $fails file(s)"
    exit 1
fi
echo "reference suite green under shims"
