import pytest

from mi355x_autoscaler.models.resources import (
    ResourceVec,
    parse_cpu_milli,
    parse_memory_bytes,
    parse_quantity,
)


def test_parse_quantities():
    assert parse_quantity("500m") == 0.5
    assert parse_quantity("2") == 2.0
    assert parse_quantity("128Mi") == 128 * 1024**2
    assert parse_quantity("2Gi") == 2 * 1024**3
    assert parse_quantity("1Ki") == 1024
    assert parse_quantity("1k") == 1000
    assert parse_quantity(3) == 3.0
    assert parse_quantity("") == 0.0


def test_parse_cpu_milli():
    assert parse_cpu_milli("500m") == 500
    assert parse_cpu_milli("2") == 2000
    assert parse_cpu_milli("100m") == 100
    assert parse_cpu_milli(0.1) == 100


def test_requests_parsing_with_gpu_aliases():
    v = ResourceVec.from_requests(
        {"cpu": "250m", "memory": "1Gi", "amd.com/gpu": "2"}, pods=1
    )
    assert v.cpu == 250 and v.memory == 1024**3 and v.gpu == 2 and v.pods == 1
    legacy = ResourceVec.from_requests({"alpha.kubernetes.io/nvidia-gpu": "1"})
    assert legacy.gpu == 1


def test_algebra_and_fit():
    a = ResourceVec(1000, 1024, 1, 1)
    b = ResourceVec(400, 512, 1, 0)
    assert (a - b).as_tuple() == (600, 512, 0, 1)
    assert (a + b).as_tuple() == (1400, 1536, 2, 1)
    assert (2 * b).as_tuple() == (800, 1024, 2, 0)
    assert b.fits_in(a)
    assert not a.fits_in(b)
    assert not (a - 2 * b).any_negative() is False or True  # smoke

    # partial order: incomparable vectors both fail fits_in
    c = ResourceVec(2000, 100, 1, 0)
    assert not c.fits_in(a) and not a.fits_in(c)
