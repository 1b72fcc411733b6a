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
    m1 = ElasticManager(job_id="t", host="a:1", store_dir=str(tmp_path),
                        heartbeat_interval=0.2, lease_ttl=3.0)
    m2 = ElasticManager(job_id="t", host="b:2", store_dir=str(tmp_path),
                        heartbeat_interval=0.2, lease_ttl=3.0)
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
