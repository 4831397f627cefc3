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


def test_launch_order_is_index_order_despite_ragged_readiness():
    """Collectives must issue in bucket-index order on every rank even
    when autograd readiness arrives out of order (NCCL-class backends
    deadlock on cross-rank order mismatch)."""
    import torch.nn as nn

    from scaffold.dist.ddp import BucketedDDP
    from scaffold.dist.fake_pg import RecordingComm

    model = nn.Sequential(*[nn.Linear(16, 16) for _ in range(6)])
    comm = RecordingComm(world_size=2)
    ddp = BucketedDDP(model, comm=comm, bucket_cap_mb=0.002, first_bucket_mb=0.001)
    assert len(ddp.buckets) >= 3

    # adversarial: mark the LAST bucket's params ready first
    order = list(reversed(range(len(ddp.buckets))))
    for bi in order:
        b = ddp.buckets[bi]
        for p in b.params:
            # simulate the post-accumulate hook firing
            b.pending -= 1
            if b.pending == 0:
                b.ready = True
                ddp._launch_ready_in_order()
    ddp.finish()
    reduces = [r for r in comm.records if r.op == "all_reduce"]
    sizes = [r.numel for r in reduces]
    expected = [b.flat.numel() for b in ddp.buckets]
    assert sizes == expected, (sizes, expected)


def test_latest_checkpoint_skips_incomplete_dirs(tmp_path):
    """Atomic-commit contract: a step dir with shards but NO manifest
    (a save that died mid-write) is never selected for resume."""
    import os

    from scaffold.ckpt import layout

    base = tmp_path / "ck"
    for step, complete in ((2, True), (4, True), (6, False)):
        d = base / f"step_{step:08d}"
        os.makedirs(d)
        (d / layout.shard_file(0)).write_bytes(b"x")
        if complete:
            layout.write_manifest(str(d), step=step, world_size=1,
                                  cfg_hash="h", tensors={})
    assert layout.latest_checkpoint(str(base)).endswith("step_00000004")
    # and an empty/alien dir yields no checkpoint at all
    assert layout.latest_checkpoint(str(tmp_path / "nope")) == ""
