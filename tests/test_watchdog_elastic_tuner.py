"""Failure-detection tier tests: watchdog timeout, elastic membership,
auto-tuner search/prune (reference SURVEY.md §5 aux subsystems)."""
import time

import pytest

import paddle_amd as paddle
from paddle_amd.distributed.auto_tuner import AutoTuner, Recorder, TunerConfig, prune, search_space
from paddle_amd.distributed.elastic import ElasticManager
from paddle_amd.distributed.watchdog import CommTaskManager


def test_watchdog_detects_timeout():
    fired = []
    mgr = CommTaskManager(timeout_s=0.2, poll_interval=0.1,
                          on_timeout=lambda t: fired.append(t.name))

    class NeverDone:
        def is_completed(self):
            return False

    mgr.register("allreduce_stuck", work=NeverDone())
    time.sleep(1.0)
    mgr.shutdown()
    assert "allreduce_stuck" in fired


def test_watchdog_completed_task_no_fire():
    fired = []
    mgr = CommTaskManager(timeout_s=0.2, poll_interval=0.1,
                          on_timeout=lambda t: fired.append(t.name))

    class Done:
        def is_completed(self):
            return True

    mgr.register("fast", work=Done())
    time.sleep(0.6)
    mgr.shutdown()
    assert fired == []


def test_elastic_membership(tmp_path):
    # generous margins: a loaded CI box can stall a heartbeat thread for
    # hundreds of ms, which must not look like a lost lease
    # ttl must survive multi-second scheduler stalls on a saturated box
    # (exit() removes the lease explicitly, so detection never waits on
    # the ttl -- it only guards against a's own lease lapsing)
    m1 = ElasticManager(job_id="t", host="a:1", store_dir=str(tmp_path),
                        heartbeat_interval=0.2, lease_ttl=15.0)
    m2 = ElasticManager(job_id="t", host="b:2", store_dir=str(tmp_path),
                        heartbeat_interval=0.2, lease_ttl=15.0)
    m1.register()
    m2.register()
    time.sleep(0.6)
    assert m1.hosts() == ["a:1", "b:2"]
    changes = []
    m1.watch(lambda hosts: changes.append(hosts))
    time.sleep(0.6)  # let the watch loop capture the 2-node baseline
    m2.exit()  # node b leaves
    for _ in range(40):  # poll up to 8 s instead of a fixed sleep
        if any(h == ["a:1"] for h in changes):
            break
        time.sleep(0.2)
    m1.exit()
    assert any(h == ["a:1"] for h in changes), changes


def test_auto_tuner_prune_and_search():
    cfg = TunerConfig(world_size=8, model_params_b=6.7)
    assert prune(cfg, {"dp_degree": 2, "mp_degree": 2, "pp_degree": 1,
                       "sharding_stage": 1, "micro_batch": 1}) is not None  # 2*2*1 != 8
    assert prune(cfg, {"dp_degree": 8, "mp_degree": 1, "pp_degree": 1,
                       "sharding_stage": 3, "micro_batch": 1}) is None
    space = search_space(cfg)
    assert all(t["dp_degree"] * t["mp_degree"] * t["pp_degree"] == 8 for t in space)
    assert len(space) > 0


def test_auto_tuner_records_best(tmp_path):
    cfg = TunerConfig(world_size=2, model_params_b=0.001,
                      candidates={"dp_degree": [2], "mp_degree": [1],
                                  "pp_degree": [1], "sharding_stage": [1, 3],
                                  "micro_batch": [1, 2]})
    rec = Recorder(path=str(tmp_path / "hist.jsonl"))
    tuner = AutoTuner(cfg, recorder=rec)
    tuner.run_trial = lambda t, timeout=0: 100.0 * t["micro_batch"]
    best = tuner.tune()
    assert best["trial"]["micro_batch"] == 2
    assert best["tokens_per_s"] == 200.0


def test_elastic_scale_in_out_plan(tmp_path):
    """membership scale in/out decisions (reference
    manager.py:484/:507), with debounce and min/max bounds."""
    from paddle_amd.distributed.elastic import (ElasticManager, ElasticScaler,
                                                parse_np_range)
    assert parse_np_range("2:8") == (2, 8)
    assert parse_np_range(4) == (4, 4)
    mgr = ElasticManager(job_id="t1", store_dir=str(tmp_path))
    sc = ElasticScaler(mgr, "2:4", debounce=1.0)
    # initial formation at 2 hosts
    p = sc.plan(hosts=["a", "b"], now=100.0)
    assert p.action == "none" and p.world == 2
    sc.commit(p)
    # a third host appears: debounced, then scale_out
    assert sc.plan(hosts=["a", "b", "c"], now=101.0).action == "wait"
    p = sc.plan(hosts=["a", "b", "c"], now=102.5)
    assert p.action == "scale_out" and p.world == 3
    sc.commit(p)
    # blip: host drops for < debounce then returns -> no restart
    assert sc.plan(hosts=["a", "b"], now=103.0).action == "wait"
    assert sc.plan(hosts=["a", "b", "c"], now=103.5).action == "none"
    # real loss, persists: scale_in
    assert sc.plan(hosts=["a", "c"], now=104.0).action == "wait"
    p = sc.plan(hosts=["a", "c"], now=105.5)
    assert p.action == "scale_in" and p.world == 2
    sc.commit(p)
    # below min: hold (fault-tolerance tier takes over, no shrink below min)
    sc.plan(hosts=["a"], now=106.0)
    assert sc.plan(hosts=["a"], now=108.0).action == "wait"
    # beyond max: capped
    sc.plan(hosts=list("abcdef"), now=109.0)
    p = sc.plan(hosts=list("abcdef"), now=111.0)
    assert p.action == "scale_out" and p.world == 4


def test_local_elastic_agent_relaunches(tmp_path):
    import time
    from paddle_amd.distributed.elastic import ElasticManager, LocalElasticAgent
    marker = tmp_path / "runs.log"
    prog = (f"import os; open(r'{marker}', 'a').write("
            "os.environ['WORLD_SIZE'] + ':' + os.environ['RANK'] + ':' "
            "+ os.environ['PADDLE_ELASTIC_RESTART'] + '\\n')")
    mgr = ElasticManager(job_id="t2", store_dir=str(tmp_path / "s"))
    agent = LocalElasticAgent(mgr, "1:3", ["-c", prog], debounce=0.5)
    agent.step(hosts=["a", "b"], now=1.0)          # initial world 2
    for p in agent.procs:                          # let world-2 run finish
        p.wait(timeout=20)
    agent.step(hosts=["a", "b", "c"], now=2.0)     # wait
    agent.step(hosts=["a", "b", "c"], now=3.0)     # scale_out -> world 3
    for p in agent.procs:
        p.wait(timeout=20)
    agent.shutdown()
    lines = marker.read_text().strip().splitlines()
    w2 = [l for l in lines if l.startswith("2:")]
    w3 = [l for l in lines if l.startswith("3:")]
    assert len(w2) >= 1 and len(w3) == 3, lines


def test_kill_rank_relaunch(tmp_path):
    """Fault injection (SURVEY §5 test plan): rank 1 crashes in its first
    incarnation; the agent detects the nonzero exit and relaunches the
    full worker set, which then completes."""
    from paddle_amd.distributed.elastic import ElasticManager, LocalElasticAgent
    marker = tmp_path / "runs.log"
    prog = (
        "import os, sys\n"
        f"open(r'{marker}', 'a').write("
        "os.environ['PADDLE_ELASTIC_RESTART'] + ':' + os.environ['RANK'] + '\\n')\n"
        "if os.environ['PADDLE_ELASTIC_RESTART'] == '0' and os.environ['RANK'] == '1':\n"
        "    sys.exit(1)\n")
    mgr = ElasticManager(job_id="t3", store_dir=str(tmp_path / "s"))
    agent = LocalElasticAgent(mgr, "2", ["-c", prog], debounce=0.1)
    agent.step(hosts=["a", "b"], now=1.0)          # initial launch, world 2
    for p in agent.procs:
        p.wait(timeout=20)
    plan = agent.step(hosts=["a", "b"], now=2.0)   # detects rank-1 crash
    assert plan.action == "restart"
    for p in agent.procs:
        p.wait(timeout=20)
    assert all(p.poll() == 0 for p in agent.procs)
    agent.shutdown()
    lines = marker.read_text().strip().splitlines()
    assert "0:1" in lines and "1:0" in lines and "1:1" in lines, lines
    assert agent.restarts >= 1
