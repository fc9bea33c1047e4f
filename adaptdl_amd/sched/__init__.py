"""In-process scheduling for a single 8xMI355X node (and beyond).

The reference runs Pollux as a Kubernetes control plane
(/root/reference/sched/adaptdl_sched/); here the same policy runs as an
in-process allocator thread (adaptdl_amd.sched.allocator) next to a local
job controller (adaptdl_amd.sched.controller) that implements
checkpoint-restart elasticity with worker processes, one per GPU.
"""

from adaptdl_amd.sched.allocator import LocalAllocator  # noqa: F401,E402
from adaptdl_amd.sched.controller import (  # noqa: F401,E402
    JobSpec, LocalController)
from adaptdl_amd.sched.supervisor import Supervisor  # noqa: F401,E402
