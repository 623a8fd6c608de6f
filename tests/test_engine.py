"""C++ ThreadedEngine tests (reference tests/cpp/engine/threaded_engine_test.cc
ported to the python binding): ordering under read/write dependencies,
concurrency of independent ops, exception propagation, stress."""
import threading
import time

import pytest

from mxnet_amd import _engine


@pytest.fixture
def eng():
    e = _engine.get()
    e.wait_for_all()
    return e


def test_write_ordering(eng):
    v = eng.new_variable()
    out = []
    for i in range(50):
        eng.push((lambda i=i: out.append(i)), (), (v,))
    eng.wait_for_var(v)
    assert out == list(range(50))


def test_read_write_interleave(eng):
    v = eng.new_variable()
    log = []
    lock = threading.Lock()

    def w(tag):
        with lock:
            log.append(('w', tag))

    def r(tag):
        with lock:
            log.append(('r', tag))

    eng.push(lambda: w(0), (), (v,))
    eng.push(lambda: r(0), (v,), ())
    eng.push(lambda: r(1), (v,), ())
    eng.push(lambda: w(1), (), (v,))
    eng.push(lambda: r(2), (v,), ())
    eng.wait_for_var(v)
    # write0 before both reads; both reads before write1; read2 last
    assert log[0] == ('w', 0)
    assert set(log[1:3]) == {('r', 0), ('r', 1)}
    assert log[3] == ('w', 1)
    assert log[4] == ('r', 2)


def test_independent_parallelism(eng):
    """Two chains on different vars overlap (wall < serial time)."""
    v1, v2 = eng.new_variable(), eng.new_variable()
    t0 = time.time()
    for _ in range(4):
        eng.push(lambda: time.sleep(0.08), (), (v1,))
        eng.push(lambda: time.sleep(0.08), (), (v2,))
    eng.wait_for_all()
    # serial would be 0.64s; two parallel chains ~0.32s. The sleeps
    # release the GIL so workers genuinely overlap.
    assert time.time() - t0 < 0.55


def test_exception_propagation(eng):
    v = eng.new_variable()

    def boom():
        raise ValueError('boom from engine op')

    eng.push(boom, (), (v,))
    with pytest.raises(RuntimeError, match='boom'):
        eng.wait_for_var(v)
    # the global flag also fires once, then clears
    with pytest.raises(RuntimeError, match='boom'):
        eng.wait_for_all()
    # engine still alive afterwards
    out = []
    eng.push(lambda: out.append(1), (), (v,))
    eng.wait_for_var(v)
    assert out == [1]


def test_diamond_dependency(eng):
    a, b, c = (eng.new_variable() for _ in range(3))
    log = []
    lock = threading.Lock()

    def add(x):
        with lock:
            log.append(x)

    eng.push(lambda: add('src'), (), (a,))
    eng.push(lambda: add('l'), (a,), (b,))
    eng.push(lambda: add('r'), (a,), (c,))
    eng.push(lambda: add('sink'), (b, c), (a,))
    eng.wait_for_all()
    assert log[0] == 'src'
    assert set(log[1:3]) == {'l', 'r'}
    assert log[3] == 'sink'


def test_stress_many_vars(eng):
    vs = [eng.new_variable() for _ in range(64)]
    counters = [0] * 64
    for round_ in range(20):
        for i, v in enumerate(vs):
            def bump(i=i):
                counters[i] += 1
            eng.push(bump, (vs[(i + 1) % 64],), (v,))
    eng.wait_for_all()
    assert counters == [20] * 64


def test_engine_fork_safety():
    """Engine worker pool survives fork (reference initialize.cc:71-83
    pthread_atfork stop/restart): child gets a working engine, parent
    keeps processing."""
    import multiprocessing as mp
    from mxnet_amd import engine
    eng = engine.get()
    v = eng.new_variable()
    res = []
    eng.push(lambda: res.append(1), mutable_vars=(v,))
    eng.wait_for_all()

    def child(q):
        e = engine.get()
        w = e.new_variable()
        out = []
        e.push(lambda: out.append(42), mutable_vars=(w,))
        e.wait_for_all()
        q.put(out[0])

    ctx = mp.get_context('fork')
    q = ctx.Queue()
    p = ctx.Process(target=child, args=(q,))
    p.start()
    p.join(timeout=60)
    assert p.exitcode == 0
    assert q.get(timeout=10) == 42
    eng.push(lambda: res.append(2), mutable_vars=(v,))
    eng.wait_for_all()
    assert res == [1, 2]
