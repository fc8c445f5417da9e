from .gradtree import (  # noqa: F401
    destruct, accum_, dodiv_, sync_buffer, markbuffer_, getbuffer_,
    ensure_synced, check_nans, grads_of,
)
from .task_ddp import prepare_training, train, train_step, update  # noqa: F401
from .process_ddp import DDPModel, init_process_group, syncgrads_worker  # noqa: F401
from .bucketing import GradBucketer  # noqa: F401
