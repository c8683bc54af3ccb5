from .dist import (DistContext, allreduce_grads, barrier, init_from_env,  # noqa: F401
                   is_distributed)
