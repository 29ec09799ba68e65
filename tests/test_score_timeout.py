"""Algorithm-selection (coll_score) and timeout behavior.
Reference parity: test/gtest/coll_score/ (tuning-string parse, range
override, fallback) and test/gtest/core/test_timeout.cc."""

import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TUNE_WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core
from ucc_amd.testing import LocalJob

job = LocalJob(2)
c = core()
smap = c.score_map_str(job.teams[0])
assert "shm" in smap, smap
# the tuning string must have zeroed shm's allreduce score and boosted tcp
lines = [l for l in smap.splitlines() if l.startswith("allreduce:")]
assert lines, smap
for l in lines:
    if "@shm/slotted" in l:
        assert l.rstrip().endswith(":0"), l
# collective still completes via fallback (tcp)
arrs = [np.ones(1000, np.float32), np.full(1000, 2.0, np.float32)]
outs = job.allreduce_np(arrs)
for o in outs:
    np.testing.assert_allclose(o, np.full(1000, 3.0, np.float32))
print("TUNE_OK")
""" % (REPO,)


def test_tuning_string_override():
    env = dict(os.environ)
    env["UCC_TUNE"] = "allreduce:@slotted:0"
    p = subprocess.run([sys.executable, "-c", TUNE_WORKER], env=env,
                       capture_output=True, text=True, timeout=120)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "TUNE_OK" in p.stdout


def test_score_map_contents():
    from ucc_amd import core
    from ucc_amd.testing import LocalJob

    job = LocalJob(2)
    smap = core().score_map_str(job.teams[0])
    # all 16 coll types present for host memory
    for coll in ("allreduce", "allgather", "allgatherv", "alltoall",
                 "alltoallv", "barrier", "bcast", "fanin", "fanout",
                 "gather", "gatherv", "reduce", "reduce_scatter",
                 "reduce_scatterv", "scatter", "scatterv"):
        assert any(line.startswith(coll + ":") for line in
                   smap.splitlines()), f"{coll} missing:\n{smap}"


def test_timeout():
    """A collective that cannot complete (peer never posts) must fail
    with UCC_ERR_TIMED_OUT once its timeout elapses."""
    from ucc_amd import core, dtypes
    from ucc_amd.testing import LocalJob

    job = LocalJob(2)
    c = core()
    a = np.ones(4096, np.float32)
    o = np.zeros(4096, np.float32)
    req = c.coll_init(job.teams[0], "allreduce", src=a.ctypes.data,
                      dst=o.ctypes.data, count=4096, dt=dtypes.FLOAT32,
                      timeout=0.3)
    req.post()
    # rank 1 never posts; progress rank 0 only
    st = c.INPROGRESS
    for _ in range(2_000_000):
        st = req.test()
        if st != c.INPROGRESS:
            break
        job.ctxs[0].progress()
    assert st < 0, f"expected timeout error, got {st}"


def test_generic_datatype_native():
    """Native unit test: generic user dtype create/size/reduce-cb +
    2-rank allreduce through the C API (build/test_generic_dt)."""
    binary = os.path.join(REPO, "build", "test_generic_dt")
    if not os.path.exists(binary):
        pytest.skip("native test binary not built (run make)")
    p = subprocess.run([binary], capture_output=True, text=True,
                       timeout=120)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-500:])
    assert p.returncode == 0 and "GENERIC_DT_OK" in p.stdout


def test_native_obj_size():
    """Native object-size regression guard (reference test_obj_size.cc
    role): progress-engine objects stay cache-friendly."""
    binary = os.path.join(REPO, "build", "test_obj_size")
    if not os.path.exists(binary):
        pytest.skip("native test binary not built")
    p = subprocess.run([binary], capture_output=True, text=True,
                       timeout=60)
    sys.stdout.write(p.stdout[-500:])
    assert p.returncode == 0 and "OBJ_SIZE_OK" in p.stdout


def test_native_mt():
    """Native THREAD_MULTIPLE driver (also the `make tsan` payload):
    dedicated progress threads + two teams driven concurrently."""
    binary = os.path.join(REPO, "build", "test_mt")
    if not os.path.exists(binary):
        pytest.skip("native test binary not built")
    p = subprocess.run([binary], capture_output=True, text=True,
                       timeout=120)
    sys.stdout.write(p.stdout[-500:])
    assert p.returncode == 0 and "MT_OK" in p.stdout


def test_thread_multiple_progress():
    """THREAD_MULTIPLE: a second thread pumping ucc_context_progress
    concurrently with the posting thread (lock-free progress queue path,
    reference MT progress-queue coverage)."""
    import threading

    import numpy as np

    from ucc_amd import core, dtypes

    c = core()
    n = 2
    libs = [c.Lib(thread_mode="multiple") for _ in range(n)]
    ctxs = [c.Context(lib) for lib in libs]
    oob = c.LocalOob(n)
    teams = [c.team_create_post(ctxs[r], local_oob=oob, rank=r)
             for r in range(n)]
    while True:
        sts = [c.team_create_test(t) for t in teams]
        assert all(s >= 0 for s in sts)
        if all(s == c.OK for s in sts):
            break

    stop = threading.Event()

    def pump(ctx):
        while not stop.is_set():
            ctx.progress()

    threads = [threading.Thread(target=pump, args=(ctx,)) for ctx in ctxs]
    for t in threads:
        t.start()
    try:
        for it in range(50):
            arrs = [np.full(2048, float(r + it), np.float32)
                    for r in range(n)]
            outs = [np.zeros(2048, np.float32) for _ in range(n)]
            reqs = [c.coll_init(teams[r], "allreduce",
                                src=arrs[r].ctypes.data,
                                dst=outs[r].ctypes.data, count=2048,
                                dt=dtypes.FLOAT32) for r in range(n)]
            for r in reqs:
                r.post()
            import time
            deadline = time.time() + 30
            while any(r.test() == c.INPROGRESS for r in reqs):
                assert time.time() < deadline, "MT collective stuck"
                for ctx in ctxs:
                    ctx.progress()
            expected = sum(float(r + it) for r in range(n))
            for o in outs:
                np.testing.assert_allclose(o, expected)
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=10)


def test_team_split():
    """ucc_team_create_from_parent: odd ranks form a sub-team; evens
    observe. Collective on the sub-team validates membership/reindexing
    (reference core/ucc_team.c team split)."""
    import numpy as np

    from ucc_amd import core, dtypes
    from ucc_amd.testing import LocalJob

    job = LocalJob(5)
    c = core()
    n = job.n
    subs = []
    for r in range(n):
        included = 1 if r % 2 == 1 else 0
        subs.append(c.team_create_from_parent(job.teams[r], r, included))
    for _ in range(200000):
        sts = [c.team_create_test(t) for t in subs]
        assert all(s >= 0 for s in sts)
        if all(s == c.OK for s in sts):
            break
    else:
        raise TimeoutError("split did not converge")
    members = [r for r in range(n) if r % 2 == 1]  # [1, 3]
    arrs = {r: np.full(512, float(r), np.float32) for r in members}
    outs = {r: np.zeros(512, np.float32) for r in members}
    reqs = [c.coll_init(subs[r], "allreduce", src=arrs[r].ctypes.data,
                        dst=outs[r].ctypes.data, count=512,
                        dt=dtypes.FLOAT32) for r in members]
    for q in reqs:
        q.post()
    for _ in range(2000000):
        if all(q.test() != c.INPROGRESS for q in reqs):
            break
        for ctx in job.ctxs:
            ctx.progress()
    expected = np.full(512, float(sum(members)), np.float32)
    for r in members:
        np.testing.assert_allclose(outs[r], expected)


def test_active_set_bcast():
    """Active-set bcast: {start,stride,size} subset over tl/tcp with a
    user tag; non-members do not participate (reference active_set
    gtest coverage)."""
    import numpy as np

    from ucc_amd import core, dtypes
    from ucc_amd.testing import LocalJob

    job = LocalJob(6)
    c = core()
    members = [1, 3, 5]  # start=1, stride=2, size=3
    bufs = {r: np.zeros(777, np.float64) for r in members}
    bufs[1][:] = np.arange(777, dtype=np.float64)
    reqs = [c.coll_init(job.teams[r], "bcast", src=bufs[r].ctypes.data,
                        dst=0, count=777, dt=dtypes.FLOAT64, root=1,
                        active_set=(1, 2, 3), tag=7)
            for r in members]
    for q in reqs:
        q.post()
    for _ in range(2000000):
        if all(q.test() != c.INPROGRESS for q in reqs):
            break
        for ctx in job.ctxs:
            ctx.progress()
    for r in members:
        np.testing.assert_array_equal(bufs[r],
                                      np.arange(777, dtype=np.float64))
    # team-wide collective still consistent after the subset coll
    arrs = [np.full(100, float(r), np.float32) for r in range(job.n)]
    outs = job.allreduce_np(arrs)
    for o in outs:
        np.testing.assert_allclose(o, np.full(100, 15.0, np.float32))


def test_profiler_and_coll_trace(tmp_path):
    """Aux subsystems: UCC_PROFILE_MODE event recorder output and
    UCC_COLL_TRACE selection print (reference utils/profile +
    UCC_COLL_TRACE parity)."""
    prof = tmp_path / "prof.log"
    worker = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd.testing import LocalJob\n"
        "j = LocalJob(2)\n"
        "j.allreduce_np([np.ones(64, np.float32)] * 2)\n"
        "print('WORKER_OK')\n" % (REPO,))
    env = dict(os.environ)
    env["UCC_PROFILE_MODE"] = "log"
    env["UCC_PROFILE_FILE"] = str(prof)
    env["UCC_COLL_TRACE"] = "1"
    env["UCC_LOG_LEVEL"] = "info"
    p = subprocess.run([sys.executable, "-c", worker], env=env,
                       capture_output=True, text=True, timeout=120)
    assert p.returncode == 0 and "WORKER_OK" in p.stdout
    out = p.stdout + p.stderr
    assert "allreduce" in out and "->" in out, out[-1500:]  # coll trace
    txt = prof.read_text()
    assert "new allreduce" in txt and "free finalize" in txt, txt[:500]


def test_ucc_info_tool():
    """ucc_info: version, config dump, simulated score map."""
    binary = os.path.join(REPO, "build", "ucc_info")
    if not os.path.exists(binary):
        pytest.skip("ucc_info not built")
    p = subprocess.run([binary, "-c", "-s"], capture_output=True,
                       text=True, timeout=120)
    assert p.returncode == 0
    assert "transports: self shm tcp cdna4" in p.stdout
    assert "UCC_LOG_LEVEL" in p.stdout
    assert "allreduce:host:" in p.stdout


def test_concurrent_multi_team():
    """THREAD_MULTIPLE with TWO teams over the same contexts: one thread
    per team drives independent collectives concurrently, stressing
    team-tag isolation in the shm slot segments and the MT progress
    path. Reference MT multi-team coverage (test_mt.cc role)."""
    import threading

    import numpy as np

    from ucc_amd import core, dtypes

    c = core()
    n = 4
    libs = [c.Lib(thread_mode="multiple") for _ in range(n)]
    ctxs = [c.Context(lib) for lib in libs]
    oob1 = c.LocalOob(n)
    oob2 = c.LocalOob(n)
    teams1 = [c.team_create_post(ctxs[r], local_oob=oob1, rank=r)
              for r in range(n)]
    while True:
        sts = [c.team_create_test(t) for t in teams1]
        assert all(s >= 0 for s in sts)
        if all(s == c.OK for s in sts):
            break
    teams2 = [c.team_create_post(ctxs[r], local_oob=oob2, rank=r)
              for r in range(n)]
    while True:
        sts = [c.team_create_test(t) for t in teams2]
        assert all(s >= 0 for s in sts)
        if all(s == c.OK for s in sts):
            break

    errors = []

    def drive(teams, coll, iters, seed):
        try:
            rng = np.random.default_rng(seed)
            for it in range(iters):
                if coll == "allreduce":
                    arrs = [rng.standard_normal(1500).astype(np.float32)
                            for _ in range(n)]
                    outs = [np.zeros(1500, np.float32) for _ in range(n)]
                    reqs = [c.coll_init(teams[r], "allreduce",
                                        src=arrs[r].ctypes.data,
                                        dst=outs[r].ctypes.data,
                                        count=1500, dt=dtypes.FLOAT32)
                            for r in range(n)]
                else:
                    root = it % n
                    arrs = [np.zeros(900, np.float64) for _ in range(n)]
                    arrs[root][:] = rng.standard_normal(900)
                    outs = [arrs[root].copy()]
                    reqs = [c.coll_init(teams[r], "bcast",
                                        src=arrs[r].ctypes.data, dst=0,
                                        count=900, dt=dtypes.FLOAT64,
                                        root=root) for r in range(n)]
                for rq in reqs:
                    rq.post()
                spins = 0
                while any(rq.test() == c.INPROGRESS for rq in reqs):
                    for ctx in ctxs:
                        ctx.progress()
                    spins += 1
                    assert spins < 40_000_000
                if coll == "allreduce":
                    exp = np.sum(arrs, axis=0)
                    for o in outs:
                        np.testing.assert_allclose(o, exp, rtol=1e-5,
                                                   atol=1e-4)
                else:
                    for a in arrs:
                        np.testing.assert_array_equal(a, outs[0])
        except Exception as e:  # pragma: no cover
            errors.append(e)

    t1 = threading.Thread(target=drive, args=(teams1, "allreduce", 25, 5))
    t2 = threading.Thread(target=drive, args=(teams2, "bcast", 25, 9))
    t1.start()
    t2.start()
    t1.join(180)
    t2.join(180)
    assert not t1.is_alive() and not t2.is_alive()
    assert not errors, errors


def test_tuning_string_malformed():
    """Malformed UCC_TUNE fragments must not crash team creation; valid
    fragments in the same string still apply."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "from ucc_amd import core\n"
        "from ucc_amd.testing import LocalJob\n"
        "job = LocalJob(2)\n"
        "smap = core().score_map_str(job.teams[0])\n"
        "assert 'allreduce' in smap\n"
        "print('MALFORMED_OK')\n" % (REPO,))
    for tune in ("garbage", "allreduce", ":::", "allreduce:@x:notanum",
                 "nosuchcoll:@a:5,allreduce:@slotted:7", ","):
        env = dict(os.environ)
        env["UCC_TUNE"] = tune
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=120)
        assert p.returncode == 0 and "MALFORMED_OK" in p.stdout, (
            tune, p.stdout[-500:], p.stderr[-1000:])


def test_oob_failure_surfaces_cleanly():
    """An OOB allgather that fails mid-bootstrap must surface a negative
    status from ucc_team_create_test — no crash, no hang. Two ranks run
    in threads with a barrier-synchronized blocking OOB (the py OOB is
    synchronous); both ranks' round-2 allgather raises."""
    import threading

    from ucc_amd import core

    c = core()
    libs = [c.Lib(thread_mode="multiple") for _ in range(2)]
    ctxs = [c.Context(l) for l in libs]
    blobs = {}
    barrier = threading.Barrier(2, timeout=60)
    lock = threading.Lock()

    def mk(rank):
        state = {"next": 0}

        def ag(data: bytes):
            rnd = state["next"]
            state["next"] += 1
            if rnd >= 1:  # team create uses exactly 2 rounds; fail the
                raise RuntimeError("injected OOB failure")  # second
            with lock:
                blobs.setdefault(rnd, {})[rank] = data
            barrier.wait()  # both contributions present
            with lock:
                out = [blobs[rnd][0], blobs[rnd][1]]
            barrier.wait()  # both read before next round reuses dict
            return out

        return ag

    results = {}

    def driver(rank):
        try:
            team = c.team_create_post(ctxs[rank], py_allgather=mk(rank),
                                      rank=rank, n_ranks=2)
            st = c.INPROGRESS
            for _ in range(500_000):
                st = c.team_create_test(team)
                if st != c.INPROGRESS:
                    break
            results[rank] = st
        except Exception as e:
            results[rank] = repr(e)

    th = [threading.Thread(target=driver, args=(r,)) for r in range(2)]
    for x in th:
        x.start()
    for x in th:
        x.join(120)
    assert all(not x.is_alive() for x in th)
    assert len(results) == 2, results
    for r, st in results.items():
        assert isinstance(st, int) and st < 0, results  # error surfaced


def test_config_file_ini(tmp_path):
    """UCC_CONFIG_FILE ini values apply (env still wins over file)."""
    ini = tmp_path / "ucc.conf"
    ini.write_text(
        "# comment\n"
        "UCC_TL_TCP_BRUCK_MAX = 1234\n"
        "UCC_TL_CDNA4_CHUNK_SIZE = 8m\n")
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "from ucc_amd import core\n"
        "from ucc_amd.testing import LocalJob\n"
        "job = LocalJob(2)\n"
        "smap = core().score_map_str(job.teams[0])\n"
        "line = [l for l in smap.splitlines()\n"
            "        if 'bruck' in l and l.startswith('alltoall:')][0]\n"
        "assert '0-1234' in line, line  # ini-driven range\n"
        "print('INI_OK')\n" % (REPO,))
    env = dict(os.environ)
    env["UCC_CONFIG_FILE"] = str(ini)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=120)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-1000:])
    assert p.returncode == 0 and "INI_OK" in p.stdout


def test_team_churn():
    """Repeated create/run/destroy cycles on fresh teams over the same
    contexts: shm segment names, team ids and fds must all recycle
    cleanly (leak class the reference covers with create/destroy
    loops)."""
    import numpy as np

    from ucc_amd import core, dtypes

    c = core()
    n = 3
    libs = [c.Lib() for _ in range(n)]
    ctxs = [c.Context(l) for l in libs]
    for cycle in range(8):
        oob = c.LocalOob(n)
        teams = [c.team_create_post(ctxs[r], local_oob=oob, rank=r)
                 for r in range(n)]
        while True:
            sts = [c.team_create_test(t) for t in teams]
            assert all(s >= 0 for s in sts), (cycle, sts)
            if all(s == c.OK for s in sts):
                break
        arrs = [np.full(512, float(r + 1 + cycle), np.float32)
                for r in range(n)]
        outs = [np.zeros(512, np.float32) for _ in range(n)]
        reqs = [c.coll_init(teams[r], "allreduce",
                            src=arrs[r].ctypes.data,
                            dst=outs[r].ctypes.data, count=512,
                            dt=dtypes.FLOAT32) for r in range(n)]
        for rq in reqs:
            rq.post()
        spins = 0
        while any(rq.test() == c.INPROGRESS for rq in reqs):
            for ctx in ctxs:
                ctx.progress()
            spins += 1
            assert spins < 20_000_000, cycle
        exp = np.full(512, sum(range(1, n + 1)) + n * cycle, np.float32)
        for o in outs:
            np.testing.assert_allclose(o, exp)
        del reqs, teams, oob  # destroy before the next cycle


RANGE_TUNE_WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core
from ucc_amd.testing import LocalJob

job = LocalJob(2)
smap = core().score_map_str(job.teams[0])
lines = [l for l in smap.splitlines()
         if l.startswith("allreduce:host") and "@shm" in l]
# ranged update splits the entry: score 0 only below 4096, the
# 4096-inf piece keeps the original score
assert any(l.startswith("allreduce:host:0-4096:@shm") and
           l.endswith(":0") for l in lines), lines
assert any(l.startswith("allreduce:host:4096-inf:@shm") and
           not l.endswith(":0") for l in lines), lines
for count in (100, 100_000):
    arrs = [np.ones(count, np.float32),
            np.full(count, 2.0, np.float32)]
    outs = job.allreduce_np(arrs)
    for o in outs:
        np.testing.assert_allclose(o, np.full(count, 3.0, np.float32))
print("RANGE_TUNE_OK")
""" % (REPO,)


def test_tuning_string_range_split():
    """Ranged tuning entries (coll:lo-hi:mem:@alg:score) change the
    score only INSIDE the range — partially-overlapping score-map
    entries split and the outside pieces keep their score (reference
    ucc_coll_score_update_from_str semantics)."""
    env = dict(os.environ)
    env["UCC_TUNE"] = "allreduce:0-4096:host:@slotted:0"
    p = subprocess.run([sys.executable, "-c", RANGE_TUNE_WORKER],
                       env=env, capture_output=True, text=True,
                       timeout=120)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "RANGE_TUNE_OK" in p.stdout


def test_tuning_string_disjoint_and_multi():
    """Disjoint ranged entries leave the map untouched; comma-separated
    entries and ':inf' scores compose (reference DSL grammar)."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "from ucc_amd import core\n"
        "from ucc_amd.testing import LocalJob\n"
        "job = LocalJob(2)\n"
        "smap = core().score_map_str(job.teams[0])\n"
        "lines = [l for l in smap.splitlines()\n"
        "         if l.startswith('allreduce:host')]\n"
        "# disjoint range (beyond any entry start) left shm at 40\n"
        "assert any('@shm/slotted:40' in l for l in lines), lines\n"
        "# second entry boosted bcast dbt to the inf ceiling\n"
        "blines = [l for l in smap.splitlines()\n"
        "          if l.startswith('bcast:host') and '@tcp/dbt' in l]\n"
        "assert any(l.rstrip().endswith(':2147483647')\n"
        "           for l in blines), blines\n"
        "print('DSL_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    # first entry: range [1TB, inf) matches nothing that overlaps the
    # shm entry below it? shm is 0-inf so it DOES overlap -> to make a
    # truly disjoint case, clamp an entry that ends before 1TB: tcp
    # bruck allgather ends at 64k.
    env["UCC_TUNE"] = ("allgather:1099511627776-inf:@bruck:1,"
                       "bcast:@dbt:inf")
    code2 = code.replace("allreduce:host", "allgather:host").replace(
        "@shm/slotted:40", "@tcp/bruck:")
    p = subprocess.run([sys.executable, "-c", code2], env=env,
                       capture_output=True, text=True, timeout=120)
    sys.stdout.write(p.stdout[-1500:])
    sys.stderr.write(p.stderr[-1500:])
    assert p.returncode == 0 and "DSL_OK" in p.stdout
