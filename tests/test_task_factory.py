"""Task factory semantics (reference convoy/task_factory.py parity)."""
import pytest

from shipyard_amd.executor import task_factory as tf


def spec(factory, command="run {0} {1}"):
    return {"command": command, "task_factory": factory}


def test_parametric_product():
    out = tf.generate_tasks(spec({
        "parametric_sweep": {"product": [
            {"start": 0, "stop": 3, "step": 1},
            {"start": 10, "stop": 30, "step": 10}]},
    }))
    assert len(out) == 6
    assert out[0]["command"] == "run 0 10"
    assert out[-1]["command"] == "run 2 20"
    assert all("task_factory" not in t for t in out)


def test_parametric_product_iterables():
    out = tf.generate_tasks(spec(
        {"parametric_sweep": {"product_iterables": ["ab", "01"]}}))
    assert [t["command"] for t in out] == \
        ["run a 0", "run a 1", "run b 0", "run b 1"]


def test_parametric_zip():
    out = tf.generate_tasks(spec(
        {"parametric_sweep": {"zip": ["abc", "012"]}}))
    assert [t["command"] for t in out] == ["run a 0", "run b 1", "run c 2"]


def test_parametric_combinations():
    out = tf.generate_tasks(spec({
        "parametric_sweep": {"combinations": {
            "iterable": ["A", "B", "C"], "length": 2}}}))
    assert [t["command"] for t in out] == ["run A B", "run A C", "run B C"]


def test_parametric_permutations():
    out = tf.generate_tasks(spec({
        "parametric_sweep": {"permutations": {
            "iterable": ["A", "B"], "length": 2}}}))
    assert len(out) == 2


def test_repeat():
    out = tf.generate_tasks({"command": "noop", "task_factory":
                             {"repeat": 4}})
    assert len(out) == 4 and all(t["command"] == "noop" for t in out)


def test_random_integer_seeded():
    f = {"random": {"generate": 5, "seed": 42,
                    "integer": {"start": 0, "stop": 100, "step": 1}}}
    a = tf.generate_tasks(spec(f, command="run {0}"))
    b = tf.generate_tasks(spec(f, command="run {0}"))
    assert [t["command"] for t in a] == [t["command"] for t in b]
    assert len(a) == 5


@pytest.mark.parametrize("dist,params", [
    ("uniform", {"a": 0, "b": 1}),
    ("gauss", {"mu": 0, "sigma": 1}),
    ("beta", {"alpha": 1, "beta": 1}),
    ("exponential", {"lambda": 2}),
    ("gamma", {"alpha": 1, "beta": 1}),
    ("lognormal", {"mu": 0, "sigma": 1}),
    ("pareto", {"alpha": 1}),
    ("weibull", {"alpha": 1, "beta": 1}),
    ("triangular", {"low": 0, "high": 1}),
])
def test_random_distributions(dist, params):
    out = tf.generate_tasks(spec({
        "random": {"generate": 3, "seed": 1,
                   "distribution": {dist: params}}}, command="run {0}"))
    assert len(out) == 3


def test_file_factory(tmp_path):
    (tmp_path / "data" / "sub").mkdir(parents=True)
    (tmp_path / "data" / "a.png").write_text("x")
    (tmp_path / "data" / "b.txt").write_text("x")
    (tmp_path / "data" / "sub" / "c.png").write_text("x")
    out = tf.generate_tasks(
        {"command": "proc {file_name}",
         "task_factory": {"file": {
             "local_storage": {"remote_path": "data",
                               "include": ["*.png", "sub/*.png"]},
             "task_filepath": "file_name"}}},
        storage_root=tmp_path)
    assert sorted(t["command"] for t in out) == ["proc a.png", "proc c.png"]


def test_custom_factory(tmp_path, monkeypatch):
    mod = tmp_path / "myfactory.py"
    mod.write_text("def generate(n):\n    for i in range(int(n)):\n"
                   "        yield (i, i * 2)\n")
    monkeypatch.syspath_prepend(str(tmp_path))
    out = tf.generate_tasks(spec({
        "custom": {"module": "myfactory", "input_args": [3]}}))
    assert [t["command"] for t in out] == ["run 0 0", "run 1 2", "run 2 4"]


def test_unknown_factory_rejected():
    with pytest.raises(tf.TaskFactoryError):
        tf.generate_tasks({"command": "x", "task_factory": {"bogus": 1}})
