from .collate_fn import default_collate, timestep_collate, diff_shape_collate, default_decollate, ttorch_collate
