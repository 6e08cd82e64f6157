from .drivers import (asgd_thread, asgd_sync, asaga_thread, asaga_sync,  # noqa: F401
                      sgd_mllib)
