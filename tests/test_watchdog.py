"""Watchdog failure-detection tests."""

import os
import subprocess
import sys
import time

from yet_another_mobilenet_series_amd.parallel.distributed import Watchdog

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_watchdog_quiet_when_kicked():
    wd = Watchdog(0.5)
    with wd:
        for _ in range(4):
            time.sleep(0.1)
            wd.kick()
    # process still alive — nothing to assert beyond not exiting


def test_watchdog_disabled_at_zero():
    wd = Watchdog(0)
    wd.start()
    assert wd._thread is None
    wd.kick()  # no-op, no crash
    wd.stop()


def test_watchdog_aborts_stalled_process():
    """A stalled step loop exits with code 124 within the timeout window."""
    code = (
        "import sys, time; sys.path.insert(0, %r)\n"
        "from yet_another_mobilenet_series_amd.parallel.distributed "
        "import Watchdog\n"
        "wd = Watchdog(1.0).start()\n"
        "time.sleep(30)\n"  # stall: never kicks
        "print('should not get here')\n" % REPO)
    t0 = time.time()
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       timeout=25, text=True)
    assert r.returncode == Watchdog.EXIT_CODE, (r.returncode, r.stdout,
                                                r.stderr)
    assert time.time() - t0 < 15
    assert "watchdog" in r.stdout
