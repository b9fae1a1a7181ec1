"""Monitoring tests: exporter metric collection + heimdall file_sd."""
import json

from shipyard_amd.executor import LocalExecutor
from shipyard_amd.monitor import heimdall
from shipyard_amd.monitor.exporter import (Exporter,
                                           collect_executor_metrics,
                                           collect_gpu_metrics)


def test_collect_gpu_metrics_no_crash():
    # GPU-less host: must return [] (not raise)
    out = collect_gpu_metrics()
    assert isinstance(out, list)


def test_executor_metrics(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "m", "gpus": {"dedicated": 0}, "cpu_slots": 2,
        "node_configuration": {"rocm": {"verify": False}}}})
    ex.jobs_add({"job_specifications": [{
        "id": "jm", "tasks": [{"id": "t", "command": "true"}]}]}, "m")
    ex.run_until_idle(timeout=30)
    m = collect_executor_metrics(ex.store)
    assert m["slots_idle"] == 2
    assert m["tasks_completed"] == 1
    ex.store.close()


def test_prometheus_scrape(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "m", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}}}})
    exp = Exporter(store=ex.store)
    body = exp.scrape().decode()
    assert "shipyard_executor_metric" in body
    assert 'name="slots_idle"' in body
    ex.store.close()


def test_heimdall_file_sd(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    heimdall.register_pool(ex.store, "poolx", 9400)
    heimdall.register_storage_cluster(ex.store, "nfs1", 9100)
    files = heimdall.write_file_sd(ex.store, tmp_path / "sd")
    assert len(files) == 2
    pool_sd = json.loads((tmp_path / "sd" / "shipyard_pool.json").
                         read_text())
    assert pool_sd[0]["targets"] == ["127.0.0.1:9400"]
    assert pool_sd[0]["labels"]["instance_id"] == "poolx"
    heimdall.unregister(ex.store, "pool:poolx")
    files = heimdall.write_file_sd(ex.store, tmp_path / "sd")
    assert len(files) == 1
    ex.store.close()


def test_recurrence_spawns_instances(tmp_path):
    from shipyard_amd.executor.recurrence import JobScheduleRunner

    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "pr", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}}}})
    jobspec = {"id": "rec", "tasks": [{"id": "t", "command": "true"}],
               "recurrence": {"schedule":
                              {"recurrence_interval": "00:00:01"}}}
    r = JobScheduleRunner(ex, "pr", jobspec)
    assert r.maybe_spawn(now=1000.0) == "rec-000"
    assert r.maybe_spawn(now=1000.5) is None  # before interval
    assert r.maybe_spawn(now=1001.5) == "rec-001"
    ex.run_until_idle(timeout=30)
    states = [t["state"] for t in ex.tasks_list("rec-000")]
    assert states == ["completed"]
    ex.store.close()


def test_recurrence_through_jobs_add(tmp_path):
    """jobs_add registers a schedule (no immediate job); the scheduler
    loop materializes instances (reference JobSchedule semantics)."""
    ex = LocalExecutor(tmp_path / "rs", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "prs", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}}}})
    added = ex.jobs_add({"job_specifications": [{
        "id": "schd", "tasks": [{"id": "t", "command": "true"}],
        "recurrence": {"schedule": {"recurrence_interval": "00:00:01"}},
    }]}, "prs")
    assert added == ["schd"]
    assert ex.jobs_list() == []  # no immediate job
    assert ex.schedules_list()[0]["id"] == "schd"
    spawned = ex.process_schedules()
    assert spawned == ["schd-000"]
    ex.run_until_idle(timeout=30)
    assert ex.tasks_list("schd-000")[0]["state"] == "completed"
    assert ex.schedule_del("schd")
    assert ex.schedules_list() == []
    ex.store.close()


def test_perf_collectors_in_scrape(tmp_path):
    ex = LocalExecutor(tmp_path / "pm", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "pm", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}}}})
    ex.stores["default"].upload_bytes("seed/a.bin", b"x" * 1000)
    ex.jobs_add({"job_specifications": [{
        "id": "pj", "tasks": [{
            "id": "t", "command": "true",
            "input_data": {"local_storage": [{"remote_path": "seed"}]},
        }]}]}, "pm")
    ex.run_until_idle(timeout=30)
    body = Exporter(store=ex.store).scrape().decode()
    assert 'name="mover_transfers"' in body
    assert 'name="mover_bytes"' in body
    from shipyard_amd.monitor.exporter import collect_perf_metrics

    m = collect_perf_metrics(ex.store)
    assert m["mover_transfers"] == 1 and m["mover_bytes"] == 1000
    ex.store.close()


class TestCertsAndTLS:
    def test_self_signed_cert_and_fingerprint(self, tmp_path):
        from shipyard_amd.utils import crypto

        key, crt = crypto.generate_self_signed_cert(tmp_path, cn="unit-test")
        assert key.exists() and crt.exists()
        fp = crypto.cert_fingerprint(crt)
        assert len(fp) == 64 and all(c in "0123456789abcdef" for c in fp)
        pfx = crypto.export_pfx(key, crt, tmp_path / "b.pfx", "pw")
        assert pfx.stat().st_size > 0

    def test_exporter_tls_scrape(self, tmp_path):
        """End-to-end: exporter behind TLS, scraped over https."""
        import ssl
        import threading
        import urllib.request

        from shipyard_amd.monitor.exporter import Exporter
        from shipyard_amd.utils import crypto

        key, crt = crypto.generate_self_signed_cert(tmp_path)
        ex = Exporter(store=None, tls_cert=str(crt), tls_key=str(key))
        srv = ex.make_server(port=0)
        port = srv.server_address[1]
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        try:
            cctx = ssl.create_default_context(cafile=str(crt))
            cctx.check_hostname = False
            body = urllib.request.urlopen(
                f"https://127.0.0.1:{port}/metrics", context=cctx,
                timeout=10).read()
            assert b"shipyard" in body or body == b"" or b"#" in body
        finally:
            srv.shutdown()
            srv.server_close()

    def test_cli_cert_lifecycle(self, tmp_path):
        from click.testing import CliRunner

        from shipyard_amd.cli import cli

        r = CliRunner()
        env_root = ["--root", str(tmp_path / "root")]
        res = r.invoke(cli, ["cert", "create", "--cn", "cli-test",
                             "--pfx-password", "pw", *env_root])
        assert res.exit_code == 0, res.output
        res = r.invoke(cli, ["cert", "list", *env_root])
        assert res.exit_code == 0 and "sha256" in res.output
        res = r.invoke(cli, ["cert", "del", "--prefix", "shipyard_cert",
                             *env_root])
        assert res.exit_code == 0 and "deleted" in res.output
        res = r.invoke(cli, ["cert", "list", *env_root])
        assert res.exit_code == 0 and "sha256" not in res.output


def test_schedule_state_survives_restart(tmp_path):
    """Recurrence progress (instance counter, next_run) is durable:
    a restarted daemon continues numbering instead of re-spawning
    instance 000 (the reference delegates this to the Batch service's
    job-schedule state; here the kv record carries it)."""
    import time as _time

    ex = LocalExecutor(tmp_path / "rp", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "prp", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}}}})
    ex.jobs_add({"job_specifications": [{
        "id": "sd", "tasks": [{"id": "t", "command": "true"}],
        "recurrence": {"schedule": {"recurrence_interval": "01:00:00"}},
    }]}, "prp")
    assert ex.process_schedules() == ["sd-000"]
    ex.run_until_idle(timeout=30)
    ex.store.close()

    # "restart": a fresh executor over the same root
    ex2 = LocalExecutor(tmp_path / "rp", detect_gpus=False)
    assert ex2.process_schedules() == []       # next_run is persisted
    # one hour later the NEXT instance spawns (not a duplicate 000)
    spawned = ex2.process_schedules(now=_time.time() + 3601)
    assert spawned == ["sd-001"]
    ex2.run_until_idle(timeout=30)
    assert ex2.tasks_list("sd-001")[0]["state"] == "completed"
    ex2.store.close()
