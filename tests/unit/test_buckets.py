import torch

from scaffold.dist.ddp import partition_buckets


def _params(sizes):
    return [torch.nn.Parameter(torch.zeros(s)) for s in sizes]


def test_reverse_order():
    ps = _params([10, 20, 30])
    buckets = partition_buckets(ps, bucket_cap_mb=1024, first_bucket_mb=1024)
    flat = [p for b in buckets for p in b]
    assert [p.numel() for p in flat] == [30, 20, 10]


def test_first_bucket_small():
    # 1 MiB fp32 tensors = 262144 elements
    n = 262144
    ps = _params([n] * 10)
    buckets = partition_buckets(ps, bucket_cap_mb=4, first_bucket_mb=1)
    sizes_mb = [sum(p.numel() for p in b) * 4 / (1 << 20) for b in buckets]
    assert sizes_mb[0] <= 1.01
    assert all(s <= 4.01 for s in sizes_mb[1:])
    assert sum(p.numel() for b in buckets for p in b) == 10 * n


def test_single_huge_param_gets_own_bucket():
    n = 10 * (1 << 20) // 4  # 10 MiB fp32
    ps = _params([100, n, 100])
    buckets = partition_buckets(ps, bucket_cap_mb=1, first_bucket_mb=1)
    assert any(len(b) == 1 and b[0].numel() == n for b in buckets)


def test_all_params_covered_once():
    ps = _params([7, 13, 1000, 50000, 3])
    buckets = partition_buckets(ps, bucket_cap_mb=0.01, first_bucket_mb=0.01)
    ids = [id(p) for b in buckets for p in b]
    assert sorted(ids) == sorted(id(p) for p in ps)


def test_mixed_dtype_buckets():
    import torch
    ps = [torch.nn.Parameter(torch.zeros(100, dtype=torch.bfloat16)),
          torch.nn.Parameter(torch.zeros(10, dtype=torch.float32)),
          torch.nn.Parameter(torch.zeros(200, dtype=torch.bfloat16)),
          torch.nn.Parameter(torch.zeros(20, dtype=torch.float32))]
    buckets = partition_buckets(ps, bucket_cap_mb=1024, first_bucket_mb=1024)
    for b in buckets:
        dts = {p.dtype for p in b}
        assert len(dts) == 1, "bucket mixes dtypes"
    ids = sorted(id(p) for b in buckets for p in b)
    assert ids == sorted(id(p) for p in ps)
