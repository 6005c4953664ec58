"""Topology validation tests (reference tests/test_topologies.py:7-20)."""

import pytest

from tf_yarn_amd.topologies import (ContainerKey, NodeLabel, TaskSpec,
                                    _check_general_topology,
                                    _check_ps_topology, ps_strategy_topology,
                                    single_server_topology)


def test_container_key_roundtrip():
    key = ContainerKey("worker", 3)
    assert key.to_kv_str() == "worker:3"
    assert ContainerKey.from_kv_str("worker:3") == key


def test_single_server_topology():
    topo = single_server_topology()
    assert set(topo) == {"chief", "evaluator"}
    _check_general_topology(topo)


def test_ps_strategy_topology():
    topo = ps_strategy_topology(nb_workers=2, nb_ps=1)
    assert set(topo) == {"chief", "worker", "ps", "evaluator"}
    assert topo["worker"].instances == 2
    _check_ps_topology(topo)


def test_unknown_task_type_rejected():
    with pytest.raises(ValueError, match="unknown task types"):
        _check_general_topology({"chief": TaskSpec(), "magic": TaskSpec()})


def test_exactly_one_chief():
    with pytest.raises(ValueError, match="one chief"):
        _check_general_topology({"worker": TaskSpec()})
    with pytest.raises(ValueError, match="one chief"):
        _check_general_topology({"chief": TaskSpec(instances=2)})


def test_nb_proc_bounded_by_vcores():
    with pytest.raises(ValueError, match="nb_proc_per_worker"):
        _check_general_topology(
            {"chief": TaskSpec(vcores=1, nb_proc_per_worker=2)})


def test_gpu_proc_count_bounded_by_node():
    with pytest.raises(ValueError, match="GPU training processes"):
        _check_general_topology({
            "chief": TaskSpec(vcores=8, nb_proc_per_worker=1,
                              label=NodeLabel.GPU),
            "worker": TaskSpec(vcores=8, nb_proc_per_worker=4, instances=3,
                               label=NodeLabel.GPU),
        })


def test_ps_topology_constraints():
    with pytest.raises(ValueError, match="at most one evaluator"):
        _check_ps_topology({
            "chief": TaskSpec(),
            "ps": TaskSpec(),
            "evaluator": TaskSpec(instances=2),
        })
    with pytest.raises(ValueError, match="at least one ps"):
        _check_ps_topology({
            "chief": TaskSpec(),
            "ps": TaskSpec(instances=0),
        })


def test_memory_accepts_skein_style_strings():
    """Reference TaskSpec takes Union[int, str] memory ("2 GiB") via
    skein Resources (topologies.py:64-72)."""
    assert TaskSpec(memory="2 GiB").memory == 2048
    assert TaskSpec(memory="512 MiB").memory == 512
    assert TaskSpec(memory="1 GB").memory == 1024
    assert TaskSpec(memory="48*1024" if False else 1024).memory == 1024
    assert TaskSpec(memory=2048).memory == 2048
    assert TaskSpec(memory="2048").memory == 2048


def test_evaluation_only_topology_with_custom_module():
    """The reference README's evaluation-only flow (README.md:371-380):
    run_on_yarn never validates custom-module topologies for a chief."""
    from tf_yarn_amd.topologies import _check_general_topology
    _check_general_topology({"evaluator": TaskSpec()},
                            require_chief=False)  # must not raise
    with pytest.raises(ValueError):
        _check_general_topology({"evaluator": TaskSpec()},
                                require_chief=True)
