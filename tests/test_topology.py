from rocnrdma_amd.utils.topology import (bind_rank_near_gpu, describe,
                                         parse_cpulist)


def test_parse_cpulist():
    assert parse_cpulist("0-3,8,10-11\n") == [0, 1, 2, 3, 8, 10, 11]
    assert parse_cpulist("5") == [5]
    assert parse_cpulist("") == []


def test_describe_no_gpu_ok():
    assert isinstance(describe(), list)


def test_bind_degrades_gracefully():
    # no GPU here: must be a no-op returning None, never raising
    assert bind_rank_near_gpu(0) is None
