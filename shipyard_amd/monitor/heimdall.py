"""Heimdall-analogue: Prometheus file_sd target generation.

The reference's heimdall daemon polls a monitoring table and writes
Prometheus `file_sd` JSON target files for pools and storage clusters
(reference heimdall/heimdall.py:292-608).  Locally, registrations live
in the store's kv table and targets are the node's exporters.
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, List


MONITOR_KEY = "heimdall:registrations"


def register_pool(store, pool_id: str, exporter_port: int) -> None:
    regs = _load(store)
    regs[f"pool:{pool_id}"] = {
        "type": "pool", "id": pool_id,
        "targets": [f"127.0.0.1:{exporter_port}"],
    }
    _save(store, regs)


def register_storage_cluster(store, cluster_id: str, port: int) -> None:
    regs = _load(store)
    regs[f"fs:{cluster_id}"] = {
        "type": "remotefs", "id": cluster_id,
        "targets": [f"127.0.0.1:{port}"],
    }
    _save(store, regs)


def unregister(store, key: str) -> None:
    regs = _load(store)
    regs.pop(key, None)
    _save(store, regs)


def _load(store) -> Dict[str, dict]:
    raw = store.kv_get(MONITOR_KEY)
    return json.loads(raw) if raw else {}


def _save(store, regs: Dict[str, dict]) -> None:
    store.kv_set(MONITOR_KEY, json.dumps(regs))


def write_file_sd(store, out_dir) -> List[Path]:
    """Emit one file_sd JSON per registration type (the heimdall
    `_construct_*_monitoring_targets` analogue)."""
    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    regs = _load(store)
    by_type: Dict[str, List[dict]] = {}
    for key, reg in regs.items():
        by_type.setdefault(reg["type"], []).append({
            "targets": reg["targets"],
            "labels": {"job": f"shipyard-{reg['type']}",
                       "instance_id": reg["id"]},
        })
    written = []
    for typ, targets in by_type.items():
        p = out / f"shipyard_{typ}.json"
        p.write_text(json.dumps(targets, indent=2))
        written.append(p)
    return written
