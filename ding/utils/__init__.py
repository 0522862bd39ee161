from .edict import EasyDict
from .registry import (
    Registry, REGISTRIES, POLICY_REGISTRY, ENV_REGISTRY, ENV_MANAGER_REGISTRY, ENV_WRAPPER_REGISTRY,
    MODEL_REGISTRY, REWARD_MODEL_REGISTRY, WORLD_MODEL_REGISTRY, BUFFER_REGISTRY, DATASET_REGISTRY,
    SERIAL_COLLECTOR_REGISTRY, SERIAL_EVALUATOR_REGISTRY, PARALLEL_COLLECTOR_REGISTRY, LEARNER_REGISTRY,
    COMM_LEARNER_REGISTRY, COMM_COLLECTOR_REGISTRY, COMMANDER_REGISTRY, LEAGUE_REGISTRY, PLAYER_REGISTRY,
    MQ_REGISTRY, AGENT_REGISTRY, HOOK_REGISTRY, STOCHASTIC_OPTIMIZER_REGISTRY,
)
from .default_helper import (get_task_uid, 
    set_pkg_seed, lists_to_dicts, dicts_to_lists, deep_merge_dicts, deep_update, squeeze,
    one_time_warning, error_wrapper, LimitedSpaceContainer, get_shape0, split_data_generator,
    flatten_dict, RunningMeanStd,
)
from .log_helper import build_logger, pretty_print, SummaryWriter, DistributedWriter, LoggerFactory
from .time_helper import EasyTimer, WatchDog, build_time_helper, TimeWrapperTime
from .segment_tree import SegmentTree, SumSegmentTree, MinSegmentTree
from .compression_helper import (
    get_data_compressor, get_data_decompressor, dummy_compressor, dummy_decompressor,
    zlib_data_compressor, zlib_data_decompressor, lz4_data_compressor, lz4_data_decompressor,
)
from .file_helper import read_file, save_file, remove_file, read_from_file, save_to_file
from .lock_helper import LockContext, LockContextType, FcntlContext, get_rw_file_lock, get_file_lock
from .dist_helper import (
    get_rank, get_world_size, broadcast, allgather, broadcast_object_list, allreduce,
    allreduce_with_indicator, allreduce_async, reduce_data, allreduce_data, get_group,
    simple_group_split, dist_init, dist_finalize, DistContext, DDPContext, synchronize,
    is_dist_initialized,
)
from .import_helper import import_module, try_import_ceph, try_import_redis
from .scheduler_helper import Scheduler
from .normalizer_helper import DatasetNormalizer
from .fast_copy import fast_copy
from .autolog import LoggedModel, LoggedValue, TickMonitor, TimeMode, NaturalTime, TickTime, TimeProxy
from .profiler_helper import Profiler, register_profiler
from .memory_helper import SimpleMemoryProfiler
from .loader import Loader, LoaderError, is_type, to_type, interval, enum, item, dict_, collection, optional, check_only
from .misc_helpers import (
    get_vi_sequence, render, K8sLauncher, generate_slurm_script, find_free_port, node_to_partition, node_to_host,
)
from .misc_helpers import get_ip, get_pid, PropagatingThread, deprecated
from typing import Sequence as SequenceType  # reference utils/type_helper.py
from .k8s_helper import (
    get_operator_server_kwargs, exist_operator_server, pod_exec_command, OperatorServer, OrchestratorLauncher,
)
from .parity_helpers import (
    iter_mapping, override, default_get, list_split, make_key_as_identifier, remove_illegal_item,
    SingletonMetaclass, convert_easy_dict_to_dict, CloudPickleWrapper, dist_mode, to_ddp_config,
    try_import_mc, try_import_link, try_import_rediscluster, render_env, get_env_fps, find_free_port_slurm,
    DEFAULT_K8S_COLLECTOR_PORT, DEFAULT_K8S_LEARNER_PORT, DEFAULT_K8S_AGGREGATOR_SLAVE_PORT,
    DEFAULT_K8S_COORDINATOR_PORT,
)
from .misc_helpers import fps, get_vi_sequence
from .registry import REGISTRIES as registries
from .fast_copy import fast_copy as fastcopy
