"""Task factories: generate task collections from one template.

Semantics-parity re-implementation of the reference's generators
(reference convoy/task_factory.py:305 `generate_task`):
  * parametric_sweep: product / product_iterables / combinations /
    permutations / zip — the parameter tuple formats the task command;
  * random: uniform/triangular/beta/exponential/gamma/gauss/lognormal/
    pareto/weibull distributions or integer randrange, `generate` count;
  * file: enumerate the local object store (the azure_storage analogue)
    and format per-file task commands;
  * repeat: N copies;
  * custom: user module exposing ``generate(*args, **kwargs)`` yielding
    parameter tuples.
"""
from __future__ import annotations

import copy
import importlib
import itertools
import random as _random
from pathlib import Path
from typing import Any, Dict, Iterator, List


class TaskFactoryError(ValueError):
    pass


def _format_command(template: dict, params) -> dict:
    out = copy.deepcopy(template)
    out.pop("task_factory", None)
    cmd = out.get("command")
    if cmd is not None:
        if isinstance(params, dict):
            out["command"] = cmd.format(**params)
        elif isinstance(params, (list, tuple)):
            out["command"] = cmd.format(*params)
        else:
            out["command"] = cmd.format(params)
    return out


def _sweep_iter(sweep: dict) -> Iterator:
    if "product" in sweep:
        ranges = [range(p["start"], p["stop"], p["step"])
                  for p in sweep["product"]]
        return itertools.product(*ranges)
    if "product_iterables" in sweep:
        return itertools.product(*sweep["product_iterables"])
    if "combinations" in sweep:
        c = sweep["combinations"]
        fn = (itertools.combinations_with_replacement
              if c.get("replacement") else itertools.combinations)
        return fn(c["iterable"], c["length"])
    if "permutations" in sweep:
        p = sweep["permutations"]
        return itertools.permutations(p["iterable"], p["length"])
    if "zip" in sweep:
        return zip(*sweep["zip"])
    raise TaskFactoryError(f"unknown parametric_sweep: {list(sweep)}")


def _random_iter(spec: dict) -> Iterator:
    n = spec["generate"]
    rng = _random.Random(spec.get("seed"))
    if "integer" in spec:
        i = spec["integer"]
        for _ in range(n):
            yield rng.randrange(i["start"], i["stop"], i["step"])
        return
    dist = spec.get("distribution") or {}
    if not dist:
        raise TaskFactoryError("random factory needs integer|distribution")
    kind, params = next(iter(dist.items()))
    fns = {
        "uniform": lambda: rng.uniform(params["a"], params["b"]),
        "triangular": lambda: rng.triangular(
            params["low"], params["high"], params.get("mode")),
        "beta": lambda: rng.betavariate(params["alpha"], params["beta"]),
        "exponential": lambda: rng.expovariate(params["lambda"]),
        "gamma": lambda: rng.gammavariate(params["alpha"], params["beta"]),
        "gauss": lambda: rng.gauss(params["mu"], params["sigma"]),
        "lognormal": lambda: rng.lognormvariate(params["mu"],
                                                params["sigma"]),
        "pareto": lambda: rng.paretovariate(params["alpha"]),
        "weibull": lambda: rng.weibullvariate(params["alpha"],
                                              params["beta"]),
    }
    if kind not in fns:
        raise TaskFactoryError(f"unknown distribution {kind}")
    for _ in range(n):
        yield fns[kind]()


def _file_iter(spec: dict, storage_root: Path) -> Iterator[dict]:
    """Enumerate object-store files (reference convoy/task_factory.py:
    348-392 `_get_storage_entities`)."""
    import fnmatch

    ls = spec.get("local_storage") or {}
    remote = ls.get("remote_path", "")
    include = ls.get("include") or []
    exclude = ls.get("exclude") or []
    base = storage_root / remote
    task_filepath = spec.get("task_filepath", "file_path")
    if not base.exists():
        return
    for p in sorted(base.rglob("*")):
        if not p.is_file():
            continue
        rel = p.relative_to(base).as_posix()
        if include and not any(fnmatch.fnmatch(rel, pat) for pat in include):
            continue
        if exclude and any(fnmatch.fnmatch(rel, pat) for pat in exclude):
            continue
        if task_filepath == "file_path":
            val = rel
        elif task_filepath == "file_path_with_container":
            val = f"{remote}/{rel}"
        elif task_filepath == "file_name":
            val = p.name
        elif task_filepath == "file_name_without_extension":
            val = p.stem
        else:
            val = rel
        yield {"file_path": str(p), "url": str(p),
               "file_path_with_container": f"{remote}/{rel}",
               "file_name": p.name,
               "file_name_without_extension": p.stem,
               task_filepath: val}


def _custom_iter(spec: dict) -> Iterator:
    mod = importlib.import_module(spec["module"], package=spec.get("package"))
    yield from mod.generate(*(spec.get("input_args") or []),
                            **(spec.get("input_kwargs") or {}))


def generate_tasks(taskspec: Dict[str, Any],
                   storage_root: Path = None) -> List[Dict[str, Any]]:
    """Expand one task template into its generated collection."""
    tf = taskspec.get("task_factory")
    if not tf:
        return [taskspec]
    out: List[Dict[str, Any]] = []
    if "parametric_sweep" in tf:
        for params in _sweep_iter(tf["parametric_sweep"]):
            out.append(_format_command(taskspec, params))
    elif "random" in tf:
        for value in _random_iter(tf["random"]):
            out.append(_format_command(taskspec, value))
    elif "file" in tf:
        if storage_root is None:
            raise TaskFactoryError(
                "file task_factory requires an object store root")
        for kw in _file_iter(tf["file"], Path(storage_root)):
            out.append(_format_command(taskspec, kw))
    elif "repeat" in tf:
        for _ in range(tf["repeat"]):
            out.append(_format_command(taskspec, ()))
    elif "custom" in tf:
        for params in _custom_iter(tf["custom"]):
            out.append(_format_command(taskspec, params))
    else:
        raise TaskFactoryError(f"unknown task_factory: {list(tf)}")
    return out
