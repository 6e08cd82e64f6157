import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a ROCm GPU (run on MI355X via gpurun)")


# GPU-tier ordering: cheap kernel-numerics tests first, end-to-end engine
# tests after, subprocess/bench smokes last — so under `-x` a wedged engine
# cannot mask the kernel verdicts (round-1: one 600 s bench hang hid all 27
# remaining GPU tests from the driver).
_GPU_FILE_ORDER = [
    "test_gpu_kernels", "test_philox", "test_gpu_graph", "test_gpu_native",
    "test_gpu_resident", "test_gpu_engine", "test_dist_native",
]


def _gpu_rank(item):
    name = item.fspath.purebasename
    try:
        file_rank = _GPU_FILE_ORDER.index(name)
    except ValueError:
        file_rank = len(_GPU_FILE_ORDER)
    # subprocess-spawning tests go last within their file
    sub = 1 if "bench" in item.name or "subprocess" in item.name else 0
    return (sub, file_rank)


def pytest_collection_modifyitems(config, items):
    import torch
    if not torch.cuda.is_available():
        skip = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
        return
    # stable sort: gpu items reordered per _gpu_rank, non-gpu left in place
    gpu_items = [i for i in items if "gpu" in i.keywords]
    if gpu_items:
        ordered = sorted(gpu_items, key=_gpu_rank)
        it = iter(ordered)
        items[:] = [next(it) if "gpu" in i.keywords else i for i in items]
