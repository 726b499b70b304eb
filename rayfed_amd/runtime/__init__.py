"""In-party runtime substrate.

The reference (ray-project/rayfed) runs on Ray: tasks/actors are separate
processes reached over Ray RPC and results live in the Ray object store
(SURVEY.md §1 layer L5).  This engine is built from scratch for one MI355X
node, so the substrate is redesigned around what the hardware actually needs:

- **Driver-local execution** for control-plane tasks: submitting a tiny task
  costs a queue push + future, not an inter-process RPC.  This is the main
  reason the tiny-task benchmark beats the reference (SURVEY.md §3.5 — the
  reference's per-iteration cost is dominated by Ray actor-call latency).
- **One worker process per GPU** for data-plane tasks: each worker owns a
  HIP device and a rank in the party's RCCL communicator over xGMI
  (``rayfed_amd.runtime.worker``); GPU objects stay device-resident in the
  owning worker between tasks.
- **Event-gated readiness**: task completion is a future/hipEvent, never a
  poll loop.
"""

from rayfed_amd.runtime.object_ref import ObjectRef
from rayfed_amd.runtime.executor import Executor, ActorHandle

__all__ = ["ObjectRef", "Executor", "ActorHandle"]
