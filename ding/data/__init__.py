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
from torch.utils.data import Dataset, DataLoader  # reference re-exports
from ding.utils.data import create_dataset, offline_data_save_type
from .buffer.buffer import apply_middleware
from .deque_buffer_wrapper import DequeBufferWrapper
from .storage import FileModelStorage
