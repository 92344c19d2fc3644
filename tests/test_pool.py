"""Fault-tolerant mining pool: ordering, retry-on-failure, crash recovery."""
import os

import pytest

from tosem2021_amd.parallel.pool import TaskFailed, run_tasks


def _square(x):
    return x * x


def _fail_once(sentinel):
    if not os.path.exists(sentinel):
        open(sentinel, "w").close()
        raise RuntimeError("transient failure")
    return "recovered"


def _always_fail():
    raise RuntimeError("permanent")


def _crash_once(sentinel):
    if not os.path.exists(sentinel):
        open(sentinel, "w").close()
        os._exit(13)  # hard-kill the worker process
    return "survived"


@pytest.mark.timeout(300)
def test_ordered_results():
    assert run_tasks([(_square, (i,)) for i in range(8)], workers=3) == \
        [i * i for i in range(8)]


@pytest.mark.timeout(300)
def test_retry_on_exception(tmp_path):
    s = str(tmp_path / "sentinel")
    assert run_tasks([(_fail_once, (s,))], workers=1) == ["recovered"]


@pytest.mark.timeout(300)
def test_permanent_failure_raises():
    with pytest.raises(TaskFailed):
        run_tasks([(_always_fail, ())], workers=1, retries=1)


@pytest.mark.timeout(300)
def test_worker_crash_recovery(tmp_path):
    s = str(tmp_path / "crash_sentinel")
    out = run_tasks([(_crash_once, (s,)), (_square, (5,))], workers=2,
                    retries=3)
    assert out[0] == "survived"
    assert out[1] == 25
