from .adam import FusedAdam  # noqa: F401
from .bert_adam import SCHEDULES, BertAdam  # noqa: F401
from .clip import GradientClipper, clip_grad_norm_  # noqa: F401
from .lamb import FusedLAMB  # noqa: F401
from .schedulers import (  # noqa: F401
    ConstantWarmUpScheduler,
    CosineWarmUpScheduler,
    LinearWarmUpScheduler,
    LRScheduler,
    PolyWarmUpScheduler,
    warmup_exp_decay_exp,
)
