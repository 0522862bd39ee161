from .buffer import (
    Buffer, BufferedData, DequeBuffer, PriorityExperienceReplay, clone_object, use_time_check, staleness_check,
    sample_range_view, padding, group_sample,
)
from .shm_buffer import ShmBuffer, ShmBufferContainer
from .storage import Storage, FileStorage
from .storage_loader import StorageLoader, FileStorageLoader
from .model_loader import ModelLoader, FileModelLoader
from .level_replay import LevelSampler
from .buffer import GPUPrioritizedBuffer
from .tensor_shipper import TrajectoryShipper
