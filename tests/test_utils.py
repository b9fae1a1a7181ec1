"""Utility unit tests (convoy/util.py-analogue semantics)."""
import datetime

import pytest

from shipyard_amd import utils


def test_parse_timedelta():
    assert utils.parse_timedelta("01:02:03") == datetime.timedelta(
        hours=1, minutes=2, seconds=3)
    assert utils.parse_timedelta("2.00:00:30") == datetime.timedelta(
        days=2, seconds=30)
    assert utils.parse_timedelta(None) is None
    td = datetime.timedelta(seconds=5)
    assert utils.parse_timedelta(td) is td
    with pytest.raises(ValueError):
        utils.parse_timedelta("90 minutes")


def test_parse_size():
    assert utils.parse_size("256m") == 256 * 10 ** 6
    assert utils.parse_size("256mi") == 256 * 2 ** 20
    assert utils.parse_size("4gi") == 4 * 2 ** 30
    assert utils.parse_size(1024) == 1024
    assert utils.parse_size("1.5k") == 1500
    assert utils.parse_size(None) is None
    with pytest.raises(ValueError):
        utils.parse_size("lots")


def test_wrap_commands():
    cmd = utils.wrap_commands_in_shell(["a", "b"])
    assert cmd.startswith("/bin/bash -c") and "a; b" in cmd


def test_expand_env(monkeypatch):
    monkeypatch.setenv("SY_TEST_VAR", "xyz")
    assert utils.expand_env("$SY_TEST_VAR/d") == "xyz/d"
    assert utils.expand_env("${SY_TEST_VAR}_s") == "xyz_s"
    assert utils.expand_env("$SY_UNSET_VAR_42") == "$SY_UNSET_VAR_42"
    assert utils.expand_env("$SY_X", env={"SY_X": "o"}) == "o"


def test_base64_and_hash():
    assert utils.base64_decode_string(
        utils.base64_encode_string("héllo")) == "héllo"
    assert len(utils.hash_string("x")) == 40


def test_none_or_empty():
    assert utils.is_none_or_empty(None)
    assert utils.is_none_or_empty([])
    assert utils.is_none_or_empty("")
    assert utils.is_not_empty([1])
    assert utils.is_not_empty("a")
