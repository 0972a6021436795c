from .comm import (Comm, init_comm, is_initialized, world_info)  # noqa
