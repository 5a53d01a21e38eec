from . import h5lite, synth  # noqa: F401
from .dataset import DistributedSampler, ShardedPretrainingDataset  # noqa: F401
