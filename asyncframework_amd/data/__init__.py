from .synthetic import synthetic_dense, synthetic_csr  # noqa: F401
from .libsvm import load_libsvm  # noqa: F401
from .shard import row_shards  # noqa: F401
