from adaptdl_amd.sched.policy.pollux import PolluxPolicy  # noqa: F401
from adaptdl_amd.sched.policy.speedup import SpeedupFunction  # noqa: F401
from adaptdl_amd.sched.policy.utils import JobInfo, NodeInfo  # noqa: F401
