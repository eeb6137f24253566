"""Fiber runtime tests (≙ reference bthread unittests).

The scenario bodies are C++ (fibers must never run Python code); pytest
drives them through the bindings and checks the results.
"""
import pytest

import brpc_amd as b

f = b.core.fiber


def test_workers_started():
    assert f.concurrency() >= 1


def test_start_join_many():
    assert f.start_join_test(100, 1000) == 100_000


def test_start_join_single():
    assert f.start_join_test(1, 1) == 1


def test_urgent_start_preempts():
    # start_urgent from a worker switches to the child immediately; the
    # parent is requeued and MAY legally be stolen by an idle worker and
    # resume in parallel with the child (same semantics as the reference),
    # so a single run can observe parent-first. Retry: systematic failure
    # means the urgent path is broken; occasional steal-races are fine.
    assert any(f.urgent_test() for _ in range(10))


def test_usleep():
    measured = f.usleep_test(20_000)
    assert 18_000 <= measured < 1_000_000


def test_butex_wake():
    assert f.butex_wake_test()


def test_butex_timeout():
    assert f.butex_timeout_test()


def test_mutex_mutual_exclusion():
    # 16 fibers x 2000 increments of an unsynchronized counter under a
    # FiberMutex with forced yields inside the critical section.
    assert f.mutex_test(16, 2000) == 32_000


def test_countdown_event():
    assert f.countdown_test(50)


def test_timer_add_delete():
    assert f.timer_test()


def test_counters_move():
    created = f.count_created()
    f.start_join_test(10, 1)
    assert f.count_created() >= created + 10


def test_fiber_local_storage():
    # fiber_key (≙ bthread_key): per-fiber slots + exit-time destructors
    assert f.key_test()


def test_semaphore():
    """FiberSemaphore (≙ reference bthread/semaphore): blocked acquirers
    park as fibers; release(n) wakes exactly that much capacity."""
    assert f.semaphore_test()


def test_rwlock():
    """FiberRWLock (≙ reference bthread/rwlock): writers exclusive (exact
    counter under racy read-modify-write; rc==2 means corruption), readers
    overlap in at least one of a few attempts (a starved CI box can
    serialize them)."""
    rcs = []
    for _ in range(5):
        rc = f.rwlock_test(8, 2, 300)
        assert rc != 2, "rwlock correctness violated"
        rcs.append(rc)
        if rc == 0:
            break
    if 0 not in rcs:
        pytest.skip("no reader overlap observed — starved box; correctness held (%s)" % rcs)


def test_gpu_wait_parks_and_wakes():
    """fiber/gpu_wait.h: a fiber blocked on a GPU ticket parks on a butex
    (yielding its worker pthread) and a wake callback — in production a
    hipLaunchHostFunc marker on the same stream — resumes it. Exercised
    here with a fake wake hook; the HIP-side integration is covered by
    tests/test_gpu.py on a real MI355X."""
    assert f.gpu_wait_test()


def test_fd_wait_epoll_integrated():
    """fiber_fd_wait is epoll-backed (≙ reference bthread_fd_wait): a
    blocked waiter parks on a butex and wakes when the fd turns readable —
    well under the old 500µs poll granularity — and timeouts fire."""
    lat_us = f.fd_wait_test()
    assert lat_us >= 0, lat_us
    assert lat_us < 100_000, lat_us  # generous bound for a loaded CI box


def test_stack_size_classes():
    """FIBER_ATTR_SMALL/NORMAL/LARGE pooled stack classes (≙ reference
    BTHREAD_ATTR_* stack sizes, bthread/stack_inl.h)."""
    assert f.stack_class_test()


def test_fiber_interrupt_and_stop():
    """fiber_interrupt cuts a sleep short with EINTR; fiber_stop makes
    sleeps return ESTOP immediately (≙ reference bthread_interrupt /
    bthread_stop, bthread/bthread.h)."""
    ok, err = f.interrupt_test()
    assert ok, err


def test_execution_queue_urgent_lane():
    """execute_urgent jumps queued normal tasks (≙ reference
    TASK_OPTIONS_URGENT, bthread/execution_queue.h:78)."""
    ok, err = f.execution_queue_urgent_test()
    assert ok, err
