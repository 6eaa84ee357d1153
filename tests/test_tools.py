"""Tools layer: parse_keyval, ClassRegister, Checkpoints, Context."""

import pytest
import torch

from aggregathor_amd import tools


def test_parse_keyval_basic():
    got = tools.parse_keyval(["batch-size:64", "name:foo"],
                             defaults={"batch-size": 32, "lr": 0.1})
    assert got == {"batch-size": 64, "name": "foo", "lr": 0.1}
    assert isinstance(got["batch-size"], int)


def test_parse_keyval_type_coercion_failure():
    with pytest.raises(tools.UserException):
        tools.parse_keyval(["batch-size:abc"], defaults={"batch-size": 32})


def test_parse_keyval_duplicate_key():
    with pytest.raises(tools.UserException):
        tools.parse_keyval(["a:1", "a:2"])


def test_parse_keyval_missing_separator():
    with pytest.raises(tools.UserException):
        tools.parse_keyval(["novalue"])


def test_class_register():
    reg = tools.ClassRegister("widget")
    class W:
        def __init__(self, x):
            self.x = x
    reg.register("w", W)
    assert reg.itemize() == ["w"]
    assert "w" in reg
    assert reg.instantiate("w", 3).x == 3
    with pytest.raises(tools.UserException):
        reg.register("w", W)
    with pytest.raises(tools.UserException):
        reg.instantiate("nope")


def test_checkpoints_roundtrip(tmp_path):
    ckpt = tools.Checkpoints(tmp_path)
    assert not ckpt.can_restore()
    ckpt.save({"step": 5, "w": torch.ones(3)}, 5)
    ckpt.save({"step": 20, "w": torch.full((3,), 2.0)}, 20)
    ckpt.save({"step": 100, "w": torch.full((3,), 3.0)}, 100)
    assert ckpt.can_restore()
    payload = ckpt.restore()  # latest = step 100 (numeric sort, not lexicographic)
    assert payload["step"] == 100
    assert torch.equal(payload["w"], torch.full((3,), 3.0))


def test_checkpoints_get_filters_seen(tmp_path):
    ckpt = tools.Checkpoints(tmp_path)
    ckpt.save({"step": 1}, 1)
    first = ckpt.get()
    assert len(first) == 1
    assert ckpt.get() == []
    ckpt.save({"step": 2}, 2)
    second = ckpt.get()
    assert len(second) == 1 and tools.Checkpoints.step_of(second[0]) == 2
    assert len(ckpt.get(no_filter=True)) == 2


def test_context_nesting(capsys):
    with tools.Context("outer"):
        with tools.Context("inner"):
            tools.info("hello")
    out = capsys.readouterr().out
    assert "[outer] [inner] hello" in out
