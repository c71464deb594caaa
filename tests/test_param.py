"""Params system semantics."""

import pytest

from flowml.param import Param, Params, TypeConverters


class Dummy(Params):
    alpha: Param = Param("undefined", "alpha", "alpha doc", TypeConverters.toFloat)
    name: Param = Param("undefined", "name", "name doc", TypeConverters.toString)

    def __init__(self):
        super().__init__()
        self._setDefault(alpha=0.5)


def test_uid_format():
    d = Dummy()
    assert d.uid.startswith("Dummy_")
    assert len(d.uid) == len("Dummy_") + 16
    assert d.uid != Dummy().uid


def test_param_rebinding_per_instance():
    a, b = Dummy(), Dummy()
    assert a.alpha.parent == a.uid
    assert b.alpha.parent == b.uid
    assert a.alpha != b.alpha


def test_default_vs_set_distinction():
    d = Dummy()
    assert d.hasDefault(d.alpha)
    assert not d.isSet(d.alpha)
    assert d.isDefined(d.alpha)
    assert d.getOrDefault(d.alpha) == 0.5
    d.set(d.alpha, 1.25)
    assert d.isSet(d.alpha)
    assert d.getOrDefault(d.alpha) == 1.25
    d.clear(d.alpha)
    assert not d.isSet(d.alpha)
    assert d.getOrDefault(d.alpha) == 0.5


def test_type_converters():
    d = Dummy()
    d.set(d.alpha, 2)  # int -> float
    assert isinstance(d.getOrDefault(d.alpha), float)
    with pytest.raises(TypeError):
        d.set(d.name, 3.5)
    with pytest.raises(TypeError):
        TypeConverters.toInt(2.5)
    assert TypeConverters.toInt(2.0) == 2
    assert TypeConverters.toListInt([1, 2.0]) == [1, 2]


def test_copy_with_extra():
    d = Dummy()
    d.set(d.alpha, 3.0)
    c = d.copy({d.alpha: 9.0})
    assert c.getOrDefault(c.alpha) == 9.0
    assert d.getOrDefault(d.alpha) == 3.0  # original untouched
    assert c.uid == d.uid  # pyspark copy keeps uid


def test_explain_params():
    d = Dummy()
    s = d.explainParams()
    assert "alpha: alpha doc (default: 0.5)" in s
    assert "name: name doc (undefined)" in s
    d.set(d.alpha, 2.0)
    assert "current: 2.0" in d.explainParam("alpha")


def test_get_param_errors():
    d = Dummy()
    with pytest.raises(ValueError):
        d.getParam("nope")
    other = Dummy()
    with pytest.raises(ValueError):
        d._shouldOwn(other.alpha)


def test_extract_param_map():
    d = Dummy()
    d.set(d.name, "x")
    m = d.extractParamMap()
    assert m[d.name] == "x"
    assert m[d.alpha] == 0.5
