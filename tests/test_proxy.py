"""Lazy proxy semantics (reference: pylzy/tests/proxy/test_proxy.py)."""
import pickle

import pytest

from lzy_amd.proxy import (
    is_lzy_proxy,
    materialize,
    materialized,
    proxy,
)


def test_materialize_on_touch():
    calls = []

    def make():
        calls.append(1)
        return 41

    p = proxy(make, (int,))
    assert not materialized(p)
    assert calls == []
    assert p + 1 == 42
    assert materialized(p)
    assert calls == [1]
    # cached: second touch doesn't re-materialize
    assert int(p) == 41
    assert calls == [1]


def test_arithmetic_and_comparison():
    p = proxy(lambda: 10, (int,))
    assert p * 2 == 20
    assert 2 * p == 20
    assert p > 5
    assert p <= 10
    assert -p == -10
    assert p % 3 == 1
    assert divmod(p, 3) == (3, 1)
    assert float(p) == 10.0
    assert format(p, "04d") == "0010"


def test_str_and_container():
    p = proxy(lambda: "hello", (str,))
    assert len(p) == 5
    assert p[1] == "e"
    assert "ell" in p
    assert p.upper() == "HELLO"
    assert str(p) == "hello"

    q = proxy(lambda: [1, 2, 3], (list,))
    assert list(iter(q)) == [1, 2, 3]
    q.append(4)
    assert len(q) == 4


def test_bool_and_none():
    p = proxy(lambda: None, (type(None),))
    assert materialize(p) is None

    q = proxy(lambda: 0, (int,))
    assert not bool(q)


def test_attr_forwarding():
    class Obj:
        def __init__(self):
            self.x = 5

        def double(self):
            return self.x * 2

    p = proxy(Obj, (Obj,))
    assert p.x == 5
    assert p.double() == 10
    p.x = 7
    assert p.double() == 14


def test_is_lzy_proxy():
    p = proxy(lambda: 1, (int,))
    assert is_lzy_proxy(p)
    assert not is_lzy_proxy(1)
    assert not is_lzy_proxy("x")


def test_pickle_reduces_to_value():
    p = proxy(lambda: {"a": 1}, (dict,))
    data = pickle.dumps(p)
    out = pickle.loads(data)
    assert out == {"a": 1}
    assert not is_lzy_proxy(out)


def test_hash_forwarding():
    p = proxy(lambda: "key", (str,))
    assert hash(p) == hash("key")
    assert {p: 1}[hash and "key"] == 1


def _proxy_of(typ, value):
    return proxy(lambda: value, (typ,))


def test_iteration_and_unpacking():
    p = _proxy_of(list, [1, 2, 3])
    assert list(iter(p)) == [1, 2, 3]
    a, b, c = p
    assert (a, b, c) == (1, 2, 3)
    assert 2 in p and 9 not in p


def test_isinstance_of_declared_type():
    p = _proxy_of(dict, {"k": 1})
    assert isinstance(p, dict)
    assert p["k"] == 1
    assert len(p) == 1


def test_context_manager_forwarding(tmp_path):
    f = tmp_path / "x.txt"
    f.write_text("hello")
    p = _proxy_of(object, open(f))
    with p as fh:
        assert fh.read() == "hello"


def test_format_and_repr():
    p = _proxy_of(float, 3.5)
    assert f"{p:.2f}" == "3.50"
    assert "3.5" in repr(p)


def test_callable_result():
    p = _proxy_of(object, lambda x: x * 3)
    assert p(4) == 12


def test_isinstance_none_optional():
    """Optional[T] proxy that materializes to None: isinstance follows
    the value (reference test_simple_isinstance_none)."""
    p = proxy(lambda: None, (int, type(None)))
    assert not isinstance(p, int)
    assert isinstance(p, type(None))
