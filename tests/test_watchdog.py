import time

from atomo_amd.utils.watchdog import StepWatchdog


def test_watchdog_fires_on_stall():
    fired = []
    wd = StepWatchdog(timeout_s=0.2, action="warn", on_expire=fired.append,
                      poll_s=0.05).start()
    time.sleep(0.5)
    wd.stop()
    assert wd.fired >= 1
    assert fired and fired[0] > 0.2


def test_watchdog_quiet_with_heartbeats():
    wd = StepWatchdog(timeout_s=0.3, action="warn", poll_s=0.05).start()
    for _ in range(10):
        wd.step()
        time.sleep(0.05)
    wd.stop()
    assert wd.fired == 0
