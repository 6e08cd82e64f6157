"""LibSVM loader (reference MLUtils.loadLibSVMFile, MLUtils.scala:71-166):
1-based indices, sparse and dense forms."""

import torch

from asyncframework_amd.data.libsvm import load_libsvm


def _write(tmp_path):
    p = tmp_path / "toy.libsvm"
    p.write_text("""1.0 1:0.5 3:1.5
-1.0 2:2.0
0.5 1:1.0 2:-1.0 4:0.25
""")
    return str(p)


def test_load_sparse(tmp_path):
    path = _write(tmp_path)
    indptr, indices, values, y = load_libsvm(path)
    assert indptr.tolist() == [0, 2, 3, 6]
    assert indices.tolist() == [0, 2, 1, 0, 1, 3]  # 1-based -> 0-based
    assert values.tolist() == [0.5, 1.5, 2.0, 1.0, -1.0, 0.25]
    assert y.tolist() == [1.0, -1.0, 0.5]


def test_load_dense(tmp_path):
    path = _write(tmp_path)
    X, y = load_libsvm(path, n_features=5, dense=True)
    assert X.shape == (3, 5)
    assert X[0, 0] == 0.5 and X[0, 2] == 1.5 and X[1, 1] == 2.0
    assert X[2, 3] == 0.25


def test_n_features_override(tmp_path):
    path = _write(tmp_path)
    indptr, indices, values, y = load_libsvm(path, n_features=10)
    assert int(indices.max()) < 10


def test_native_matches_python_fallback(tmp_path, monkeypatch):
    path = _write(tmp_path)
    from asyncframework_amd.data import libsvm as L
    nat = L.load_libsvm(path)
    # force the pure-python path
    monkeypatch.setattr(L, "_parse_native", lambda p: None)
    pyr = L.load_libsvm(path)
    for a, b in zip(nat, pyr):
        assert torch.equal(a, b)


def test_empty_and_comment_only_files(tmp_path):
    from asyncframework_amd.data.libsvm import load_libsvm
    p = tmp_path / "empty.libsvm"
    p.write_text("")
    indptr, indices, values, y = load_libsvm(str(p), n_features=4)
    assert indptr.tolist() == [0] and y.numel() == 0
    p2 = tmp_path / "hash.libsvm"
    p2.write_text("1 1:2.0\n")
    indptr2, indices2, values2, y2 = load_libsvm(str(p2), n_features=4)
    assert y2.tolist() == [1.0]
    assert indices2.tolist() == [0] and values2.tolist() == [2.0]
