import time

import pytest

from simple_tip_amd.core.timer import Timer


def test_basic_timing():
    t = Timer()
    with t:
        time.sleep(0.05)
    assert 0.04 <= t.get() <= 1.0


def test_accumulates():
    t = Timer()
    with t:
        time.sleep(0.02)
    with t:
        time.sleep(0.02)
    assert t.get() >= 0.03


def test_double_start_raises():
    t = Timer(start=True)
    with pytest.raises(RuntimeError):
        t.start()


def test_stop_without_start_raises():
    t = Timer()
    with pytest.raises(RuntimeError):
        t.stop()


def test_get_while_running_warns():
    t = Timer(start=True)
    with pytest.warns(RuntimeWarning):
        t.get()
    t.stop()


def test_decorator():
    t = Timer()

    @t.timed
    def f():
        time.sleep(0.02)
        return 42

    assert f() == 42
    assert t.get() >= 0.01
