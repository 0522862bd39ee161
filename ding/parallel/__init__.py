from .grad_bucket import GradBucketAllReducer, sync_gradients_flat
