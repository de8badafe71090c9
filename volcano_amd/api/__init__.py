from .resource import (CPU, GPU, MEMORY, PODS, DEFAULT_DIMS, MIN_RESOURCE,
                       Resource, ResourceDims, parse_quantity)
from .types import (ALLOCATED_STATUSES, Action, Event, JobPhase,
                    PodGroupPhase, QueueState, TaskStatus)
from .objects import (ANN_PODGROUP, ANN_PREEMPTABLE, ANN_QUEUE,
                      DEFAULT_NAMESPACE, DEFAULT_QUEUE, DEFAULT_SCHEDULER,
                      KINDS, LBL_JOB_NAME, LBL_NODEGROUP, LBL_TASK_INDEX,
                      LBL_TASK_SPEC, Command, CronJob, FlowStep, HyperNode,
                      HyperNodeMember, Job, JobFlow, JobSpec, JobStatus,
                      JobTemplate, LifecyclePolicy, MemberSelector, Node,
                      NodeShard, ObjectMeta, Pod, PodGroup, PodGroupSpec,
                      PodGroupStatus, Queue, QueueSpec, QueueStatus, Taint,
                      TaskSpec, Toleration, from_dict, to_dict)
from .info import JobInfo, NodeInfo, QueueInfo, TaskClass, TaskInfo
