"""Hang-detection watchdog: fires on a missed heartbeat, stays quiet
with beats, rearms."""

import time


def test_watchdog_fires_and_rearms(capsys):
    from quintnet_amd.utils import Watchdog

    with Watchdog(timeout_s=0.3) as wd:
        for _ in range(4):
            wd.beat()
            time.sleep(0.05)
        assert not wd.fired
        time.sleep(0.8)  # miss heartbeats
        assert wd.fired


def test_watchdog_quiet_with_beats():
    from quintnet_amd.utils import Watchdog

    with Watchdog(timeout_s=0.5) as wd:
        for _ in range(8):
            wd.beat()
            time.sleep(0.05)
        assert not wd.fired
