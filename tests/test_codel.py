"""CoDel AQM behavior on a virtual clock.

Mirrors the reference's statistical approach (test/codel.test.js): feed a
synthetic overload pattern and check that drops begin only after the
sojourn time stays above target for a control interval, that the drop
rate accelerates (interval/sqrt(count)), and that get_max_idle()
tightens under sustained overload.
"""

import math

from cueball_amd.codel import CODEL_INTERVAL, ControlledDelay
from cueball_amd.testing import advance
from conftest import run_vt


def _ms(loop):
    return loop.time() * 1000.0


def test_no_drops_below_target():
    async def body(loop):
        cd = ControlledDelay(100, loop=loop)
        for _ in range(100):
            start = _ms(loop) - 50  # sojourn 50ms < target 100ms
            assert cd.overloaded(start) is False
            await advance(loop, 0.01)

    run_vt(lambda loop: body(loop))


def test_drop_after_interval_above_target():
    async def body(loop):
        cd = ControlledDelay(10, loop=loop)
        drops = 0
        # every dequeue sees a sojourn of 50ms (> 10ms target)
        for _ in range(100):
            start = _ms(loop) - 50
            if cd.overloaded(start):
                drops += 1
            await advance(loop, 0.01)  # 10ms per dequeue
        # first interval (100ms) must pass before the first drop
        assert drops > 0
        assert drops < 100

    run_vt(lambda loop: body(loop))


def test_drop_rate_accelerates():
    async def body(loop):
        cd = ControlledDelay(10, loop=loop)
        drop_times = []
        for _ in range(2000):
            start = _ms(loop) - 100
            if cd.overloaded(start):
                drop_times.append(_ms(loop))
            await advance(loop, 0.002)
        assert len(drop_times) > 3
        gaps = [b - a for a, b in zip(drop_times, drop_times[1:])]
        # gaps shrink as count rises
        assert gaps[-1] < gaps[0]
        # first re-drop comes ~interval/sqrt(1) after entering dropping
        assert abs(gaps[0] - CODEL_INTERVAL) < 25
        assert gaps[-1] < CODEL_INTERVAL / 2

    run_vt(lambda loop: body(loop))


def test_recovers_when_below_target():
    async def body(loop):
        cd = ControlledDelay(10, loop=loop)
        for _ in range(50):
            cd.overloaded(_ms(loop) - 100)
            await advance(loop, 0.01)
        assert cd.cd_dropping is True
        # sojourn drops below target -> dropping mode ends
        assert cd.overloaded(_ms(loop) - 1) is False
        assert cd.cd_dropping is False

    run_vt(lambda loop: body(loop))


def test_get_max_idle_bounds():
    async def body(loop):
        cd = ControlledDelay(100, loop=loop)
        cd.empty()
        # healthy: high bound (10x)
        assert cd.get_max_idle() == 1000
        # not empty for > bound: tighten to 3x
        await advance(loop, 2.0)
        assert cd.get_max_idle() == 300

    run_vt(lambda loop: body(loop))


def test_native_and_pure_controllers_agree():
    """Drop-for-drop agreement between the native and pure CoDel
    implementations over a scripted overload pattern (this caught the
    drop-next-not-advanced reference quirk during the C port)."""
    from cueball_amd.codel import (ControlledDelay,
                                   PurePythonControlledDelay)
    from cueball_amd.testing import VirtualLoop, advance

    async def run(loop, cls):
        cd = cls(5.0, loop=loop)
        pattern = []
        # overload, recover, overload again
        for i in range(150):
            sojourn = 50.0 if (i < 60 or i >= 100) else 1.0
            start = loop.time() * 1000.0 - sojourn
            pattern.append(bool(cd.overloaded(start)))
            if i == 80:
                cd.empty()
            await advance(loop, 0.01)
        return pattern, cd.cd_count, cd.cd_dropping

    loop = VirtualLoop()
    try:
        nat = loop.run_until_complete(run(loop, ControlledDelay))
    finally:
        loop.close()
    loop = VirtualLoop()
    try:
        pure = loop.run_until_complete(
            run(loop, PurePythonControlledDelay))
    finally:
        loop.close()
    assert nat == pure
    assert any(nat[0]), "pattern should include drops"
