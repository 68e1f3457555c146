from .parser import (ConfigParser, build_optimizer, build_lr_scheduler,  # noqa: F401
                     set_by_path, get_by_path)
