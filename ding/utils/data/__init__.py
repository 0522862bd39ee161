from .collate_fn import default_collate, timestep_collate, diff_shape_collate, default_decollate, ttorch_collate
from .dataset import NaiveRLDataset, HDF5Dataset, D4RLDataset, D4RLTrajectoryDataset, create_dataset, offline_data_save_type, BCODataset, hdf5_save
from .dataloader import AsyncDataLoader
from .rlhf_dataset import OnlineRLDataset, OfflineRLDataset, zero_pad_sequences
