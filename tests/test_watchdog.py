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


def test_watchdog_kill_on_hang(tmp_path):
    """kill_on_hang exits the process (code 42) so an elastic agent can
    restart it instead of waiting out the collective timeout."""
    import subprocess
    import sys

    import os

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = tmp_path / "hang.py"
    script.write_text(
        "import sys, time\n"
        "sys.path.insert(0, %r)\n"
        "from quintnet_amd.utils import Watchdog\n"
        "Watchdog(timeout_s=0.4, kill_on_hang=True).start()\n"
        "time.sleep(30)\n" % root
    )
    r = subprocess.run([sys.executable, str(script)], cwd=root,
                       capture_output=True, text=True, timeout=20)
    assert r.returncode == 42, (r.returncode, r.stderr[-500:])
    assert "no heartbeat" in r.stderr
