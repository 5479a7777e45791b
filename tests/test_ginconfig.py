"""Tests for the minimal gin-config implementation."""

import enum

import pytest

from tensor2robot_amd import gin


@gin.configurable
def _fn(a=1, b=2):
  return a, b


@gin.configurable
class _Model:

  def __init__(self, lr=0.1, name="m"):
    self.lr = lr
    self.name = name


@gin.configurable
def _factory(cls=None):
  return cls


@gin.configurable
def _required_fn(x=gin.REQUIRED):
  return x


class _Mode(enum.Enum):
  A = 1
  B = 2


gin.constants_from_enum(_Mode, module="test_ginconfig")


def test_basic_binding():
  gin.parse_config("_fn.a = 10")
  assert _fn() == (10, 2)
  assert _fn(a=5) == (5, 2)  # explicit kwarg wins


def test_class_binding():
  gin.parse_config("_Model.lr = 0.5\n_Model.name = 'hello'")
  m = _Model()
  assert m.lr == 0.5 and m.name == "hello"
  assert isinstance(m, _Model)


def test_reference_and_call():
  gin.parse_config("_factory.cls = @_Model\n_Model.lr = 3.0")
  cls = _factory()
  m = cls()
  assert m.lr == 3.0
  gin.clear_config()
  gin.parse_config("_factory.cls = @_Model()\n_Model.lr = 7.0")
  inst = _factory()
  assert isinstance(inst, _Model) and inst.lr == 7.0


def test_macro():
  gin.parse_config("LR = 0.25\n_Model.lr = %LR")
  assert _Model().lr == 0.25


def test_scopes():
  gin.parse_config("""
_Model.lr = 1.0
fast/_Model.lr = 9.0
""")
  assert _Model().lr == 1.0
  with gin.config_scope("fast"):
    assert _Model().lr == 9.0


def test_scoped_reference():
  gin.parse_config("""
_factory.cls = @slow/_Model()
slow/_Model.lr = 0.001
""")
  m = _factory()
  assert m.lr == 0.001


def test_collections_and_literals():
  gin.parse_config(
      "_fn.a = [1, 2.5, 'x', (3, 4), None, True]\n_fn.b = {'k': 1}")
  a, b = _fn()
  assert a == [1, 2.5, "x", (3, 4), None, True]
  assert b == {"k": 1}


def test_multiline_binding():
  gin.parse_config("""
_fn.a = [
    1,
    2,  # comment
    3,
]
""")
  assert _fn()[0] == [1, 2, 3]


def test_enum_constant():
  gin.parse_config("_fn.a = %_Mode.B")
  assert _fn()[0] is _Mode.B


def test_bare_enum_token():
  gin.parse_config("_fn.a = _Mode.A")
  assert _fn()[0] is _Mode.A


def test_required_raises():
  with pytest.raises(gin.GinError):
    _required_fn()
  gin.parse_config("_required_fn.x = 4")
  assert _required_fn() == 4


def test_external_configurable():
  wrapped = gin.external_configurable(dict, name="make_dict")
  gin.parse_config("make_dict.foo = 1")
  assert wrapped() == {"foo": 1}


def test_operative_config_str():
  gin.parse_config("_fn.a = 42")
  _fn()
  s = gin.operative_config_str()
  assert "_fn.a = 42" in s


def test_query_and_bind_parameter():
  gin.bind_parameter("_fn.b", 99)
  assert gin.query_parameter("_fn.b") == 99
  assert _fn()[1] == 99


def test_config_file(tmp_path):
  p = tmp_path / "cfg.gin"
  p.write_text("_fn.a = 'fromfile'\n")
  gin.parse_config_files_and_bindings([str(p)], "_fn.b = 'frombinding'")
  assert _fn() == ("fromfile", "frombinding")


def test_unknown_configurable_errors():
  with pytest.raises(gin.GinError):
    gin.query_parameter("nosuchthing.param")
  gin.parse_config("_fn.a = @no_such_ref")
  with pytest.raises(gin.GinError):
    _fn()
