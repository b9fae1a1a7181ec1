"""Every config family: template documents validate; strictness and
type errors are caught (the reference's validator-as-contract-test,
SURVEY.md §4)."""
import pytest
import yaml

from shipyard_amd.config import ConfigType, SchemaViolation, validate_config

VALID = {
    ConfigType.credentials: """
credentials:
  storage:
    default: {root: /tmp/x}
  registries:
    docker:
      reg.example.com: {username: u, password_secret_id: pw}
  secrets_store: {file: /tmp/sec.bin}
""",
    ConfigType.config: """
batch_shipyard:
  storage_account_settings: default
global_resources:
  docker_images: [busybox]
  local_images:
  - {name: img, source: dir}
  volumes:
    shared_data_volumes:
      sc: {volume_driver: host_dir, container_path: /mnt/x}
""",
    ConfigType.pool: """
pool_specification:
  id: p
  gpus: {dedicated: 8, low_priority: 0}
  max_tasks_per_gpu: 1
  autoscale:
    scenario:
      name: workday
      maximum_gpu_count: {dedicated: 8}
""",
    ConfigType.jobs: """
job_specifications:
- id: j
  tasks:
  - command: run
    gpus: all
    multi_instance:
      num_instances: 4
      gang: {backend: rccl, gpus_per_rank: 2}
""",
    ConfigType.fs: """
remote_fs:
  storage_clusters:
    s1: {driver: tmpfs, size: 1gi, mountpoint: /mnt/s1}
""",
    ConfigType.monitor: """
monitoring:
  exporter: {port: 9400, interval_seconds: 1.0, collectors: [gpu]}
""",
    ConfigType.federation: """
federation:
  federations:
    f1: {pools: [a, b]}
""",
    ConfigType.slurm: """
slurm:
  cluster_id: sy
  elastic_partitions:
    gpu:
      batch_pools:
        p: {max_compute_nodes: 4}
""",
}


@pytest.mark.parametrize("ctype", list(VALID), ids=[c.value for c in VALID])
def test_valid_documents(ctype):
    validate_config(ctype, yaml.safe_load(VALID[ctype]))


@pytest.mark.parametrize("ctype", list(VALID), ids=[c.value for c in VALID])
def test_unknown_key_rejected(ctype):
    doc = yaml.safe_load(VALID[ctype])
    top = doc[next(iter(doc))]
    if isinstance(top, list):
        top[0]["zz_bogus_key"] = 1
    else:
        top["zz_bogus_key"] = 1
    with pytest.raises(SchemaViolation):
        validate_config(ctype, doc)


def test_type_errors_caught():
    doc = yaml.safe_load(VALID[ConfigType.pool])
    doc["pool_specification"]["gpus"]["dedicated"] = "eight"
    with pytest.raises(SchemaViolation) as ei:
        validate_config(ConfigType.pool, doc)
    assert "expected int" in str(ei.value)


def test_enum_and_range():
    doc = yaml.safe_load(VALID[ConfigType.pool])
    doc["pool_specification"]["autoscale"]["scenario"]["name"] = "nope"
    with pytest.raises(SchemaViolation):
        validate_config(ConfigType.pool, doc)
    doc = yaml.safe_load(VALID[ConfigType.pool])
    doc["pool_specification"]["gpus"]["dedicated"] = 9
    with pytest.raises(SchemaViolation):
        validate_config(ConfigType.pool, doc)


def test_timedelta_and_size_types():
    doc = yaml.safe_load(VALID[ConfigType.jobs])
    doc["job_specifications"][0]["max_wall_time"] = "bogus"
    with pytest.raises(SchemaViolation):
        validate_config(ConfigType.jobs, doc)
    doc = yaml.safe_load(VALID[ConfigType.jobs])
    doc["job_specifications"][0]["shm_size"] = "256m"
    validate_config(ConfigType.jobs, doc)  # ok
    doc["job_specifications"][0]["shm_size"] = "xx"
    with pytest.raises(SchemaViolation):
        validate_config(ConfigType.jobs, doc)


def test_typed_accessors_for_aux_families():
    from shipyard_amd.config.settings import (federation_settings,
                                              monitoring_settings,
                                              slurm_settings)

    m = monitoring_settings(yaml.safe_load(VALID[ConfigType.monitor]))
    assert m.exporter_port == 9400 and "gpu" in m.collectors
    f = federation_settings(yaml.safe_load(VALID[ConfigType.federation]))
    assert f.federations["f1"]["pools"] == ["a", "b"]
    s = slurm_settings(yaml.safe_load(VALID[ConfigType.slurm]))
    assert s.cluster_id == "sy"
    assert s.partitions[0].batch_pools["p"]["max_compute_nodes"] == 4


def test_pool_nodes_multinode_schema():
    """Multi-node pool surface (nodes: per-host slots + ssh)."""
    validate_config(ConfigType.pool, {"pool_specification": {
        "id": "mp",
        "nodes": [
            {"id": "n0", "host": "10.0.0.4", "gpus": {"dedicated": 8},
             "ssh": {"username": "ops", "private_key": "/k"}},
            {"id": "n1", "cpu_slots": 4},
        ]}})
    with pytest.raises(SchemaViolation):
        validate_config(ConfigType.pool, {"pool_specification": {
            "id": "mp", "nodes": [{"id": "n0", "bogus": True}]}})
    with pytest.raises(SchemaViolation):  # node id required
        validate_config(ConfigType.pool, {"pool_specification": {
            "id": "mp", "nodes": [{"host": "10.0.0.4"}]}})
