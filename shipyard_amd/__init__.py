"""shipyard_amd — an MI355X-native container/batch orchestration framework.

A from-scratch re-design of Azure Batch Shipyard's capability surface
(reference: /root/reference, v3.9.1) for a single 8xMI355X node:

  * the Azure Batch service is replaced by a local executor
    (:mod:`shipyard_amd.executor`) managing pools of GPU slots;
  * multi-instance (MPI) tasks gang-launch as RCCL-over-xGMI ranks
    (:mod:`shipyard_amd.runner.gang`, :mod:`shipyard_amd.comm`);
  * the cascade image replicator and convoy/data mover gain HIP/CDNA4
    data-plane kernels (:mod:`shipyard_amd.ops`);
  * the YAML config families, strict schema validation and typed
    settings accessors mirror the reference's user surface
    (:mod:`shipyard_amd.config`).
"""

__version__ = "0.1.0"
