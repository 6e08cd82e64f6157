"""Row sharding: 4-aligned boundaries (Philox block contract), full cover."""

from asyncframework_amd.data.shard import row_shards


def test_cover_and_alignment():
    for n, p in [(100, 3), (8_100_000, 8), (1000, 7), (17, 4), (4, 4)]:
        sh = row_shards(n, p)
        assert len(sh) == p
        assert sh[0][0] == 0 and sh[-1][1] == n
        for i in range(1, p):
            assert sh[i][0] == sh[i - 1][1]          # contiguous
            assert sh[i][0] % 4 == 0                  # aligned starts
        assert all(s <= t for s, t in sh)


def test_csr_worker_rebase():
    import torch
    from asyncframework_amd import run as runner
    from asyncframework_amd.data.synthetic import synthetic_csr
    from asyncframework_amd.engine.config import EngineConfig
    cfg = EngineConfig(d=30, N=100, num_workers=3, num_iterations=1,
                       batch_rate=0.5, device="cpu")
    data = synthetic_csr(cfg.N, cfg.d, nnz_per_row=5, seed=1)
    workers = runner.build_csr_workers(cfg, *data)
    indptr, indices, values, y = data
    total = 0
    for wk in workers:
        sh = wk.shard
        assert int(sh.indptr[0]) == 0                   # rebased
        assert int(sh.indptr[-1]) == sh.values.shape[0]  # consistent
        total += sh.n_rows
        # shard rows must reproduce the global CSR slice
        g0 = int(indptr[sh.row_start])
        assert torch.equal(sh.values,
                           values[g0:g0 + int(sh.indptr[-1])])
    assert total == cfg.N
