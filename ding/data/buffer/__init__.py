from .buffer import Buffer, BufferedData, apply_middleware
from .deque_buffer import DequeBuffer
from .middleware import (
    PriorityExperienceReplay, clone_object, use_time_check, staleness_check, sample_range_view, padding, group_sample,
)
from .tensor_buffer import GPUPrioritizedBuffer
