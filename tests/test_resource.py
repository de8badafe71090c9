"""Resource vector semantics (reference pkg/scheduler/api/resource_info.go)."""

from volcano_amd.api.resource import (CPU, MEMORY, Resource, ResourceDims,
                                      parse_quantity)


def test_parse_quantity():
    assert parse_quantity("500m", CPU) == 500.0
    assert parse_quantity("2", CPU) == 2000.0
    assert parse_quantity(2, CPU) == 2000.0
    assert parse_quantity("1Gi", MEMORY) == 1024 ** 3
    assert parse_quantity("1G", MEMORY) == 1e9
    assert parse_quantity("4") == 4.0


def test_from_spec_and_ops():
    a = Resource.from_spec({"cpu": "2", "memory": "4Gi"})
    b = Resource.from_spec({"cpu": "500m", "memory": "1Gi"})
    assert a.milli_cpu == 2000
    c = a.clone().sub(b)
    assert c.milli_cpu == 1500
    assert c.memory == 3 * 1024 ** 3
    # saturating subtract
    d = b.clone().sub(a)
    assert d.milli_cpu == 0.0


def test_less_equal():
    small = Resource({CPU: 100})
    big = Resource({CPU: 1000, MEMORY: 10})
    assert small.less_equal(big)
    assert not big.less_equal(small)
    # absent dims in other count as zero
    assert not Resource({"amd.com/gpu": 1}).less_equal(big)
    assert Resource({"amd.com/gpu": 0.05}).less_equal(big)  # below MIN_RESOURCE


def test_dims_vector_roundtrip():
    dims = ResourceDims()
    r = Resource({CPU: 1000, "amd.com/gpu": 2})
    vec = r.to_vector(dims)
    assert vec[dims.index[CPU]] == 1000
    assert vec[dims.index["amd.com/gpu"]] == 2
    back = Resource.from_vector(vec, dims)
    assert back == r


def test_diff():
    a = Resource({CPU: 1000, MEMORY: 100})
    b = Resource({CPU: 400, MEMORY: 300})
    inc, dec = a.diff(b)
    assert inc.q == {CPU: 600}
    assert dec.q == {MEMORY: 200}
