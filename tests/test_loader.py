"""Streaming-loader tests: shard disjointness/exhaustiveness, infinite
cycling, converter contract (SURVEY.md §4 unit plan item 1)."""
import numpy as np
import pyarrow as pa
import pytest
import torch

from ddlw_amd.data import make_converter
from ddlw_amd.data.loader import ShardedParquetLoader, shard_row_groups
from ddlw_amd.data.synthetic import make_synthetic_dataset


def test_shard_row_groups_disjoint_exhaustive():
    for n_groups in (1, 7, 8, 13):
        for world in (1, 2, 3, 8):
            all_groups = []
            for r in range(world):
                all_groups += shard_row_groups(n_groups, r, world)
            assert sorted(all_groups) == list(range(n_groups))


def _make_table(n=40, size=16, num_classes=4, seed=0):
    contents, labels = make_synthetic_dataset(
        n, img_height=size, img_width=size, num_classes=num_classes, seed=seed, jpeg=True
    )
    return pa.table({"content": pa.array(contents, pa.binary()), "label_idx": pa.array(labels)})


def test_converter_len_and_delete(ddlw_home):
    tbl = _make_table(40)
    conv = make_converter(tbl, row_group_rows=8)
    assert len(conv) == 40
    path = conv.path
    assert path.exists()
    conv.delete()
    assert not path.exists()


def test_loader_shards_cover_all_rows_once(ddlw_home):
    tbl = _make_table(40)
    conv = make_converter(tbl, row_group_rows=8)  # 5 row groups
    seen = []
    for r in range(2):
        with conv.make_torch_dataset(
            batch_size=4, cur_shard=r, shard_count=2, num_epochs=1, img_height=16, img_width=16
        ) as loader:
            for images, labels in loader:
                assert images.shape[1:] == (3, 16, 16)
                seen.append(labels)
    total = torch.cat(seen)
    assert len(total) == 40  # disjoint + exhaustive across the 2 shards
    conv.delete()


def test_loader_infinite_cycling(ddlw_home):
    tbl = _make_table(16)
    conv = make_converter(tbl, row_group_rows=8)
    with conv.make_torch_dataset(
        batch_size=8, cur_shard=0, shard_count=1, num_epochs=None, img_height=16, img_width=16
    ) as loader:
        it = iter(loader)
        batches = [next(it) for _ in range(5)]  # > 2 epochs worth
    assert all(b[0].shape[0] == 8 for b in batches)
    conv.delete()


@pytest.mark.gpu
def test_loader_gpu_staging(ddlw_home):
    """GPU path: pinned-buffer + side-stream H2D staging delivers the same
    batches as the CPU path."""
    tbl = _make_table(32, seed=5)
    conv = make_converter(tbl, row_group_rows=8)
    dev = torch.device("cuda:0")
    cpu_batches = []
    with conv.make_torch_dataset(batch_size=8, num_epochs=1, img_height=16, img_width=16) as loader:
        cpu_batches = [(i.clone(), l.clone()) for i, l in loader]
    with conv.make_torch_dataset(
        batch_size=8, num_epochs=1, img_height=16, img_width=16, device=dev
    ) as loader:
        gpu_batches = [(i, l) for i, l in loader]
        torch.cuda.synchronize()
    assert len(gpu_batches) == len(cpu_batches)
    for (ci, cl), (gi, gl) in zip(cpu_batches, gpu_batches):
        assert gi.is_cuda and gl.is_cuda
        assert torch.allclose(gi.cpu(), ci)
        assert torch.equal(gl.cpu(), cl)
    conv.delete()


def test_loader_label_parity(ddlw_home):
    tbl = _make_table(24, seed=3)
    conv = make_converter(tbl, row_group_rows=100)
    with conv.make_torch_dataset(
        batch_size=24, cur_shard=0, shard_count=1, num_epochs=1, img_height=16, img_width=16
    ) as loader:
        _, labels = next(iter(loader))
    expect = tbl.column("label_idx").to_pylist()
    assert labels.tolist() == expect
    conv.delete()


def test_loader_row_level_fallback_more_ranks_than_groups(tmp_path):
    """Fewer row groups than ranks: row-level round-robin keeps every shard
    non-empty, disjoint, exhaustive (the hang-prevention path)."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from ddlw_amd.data.loader import ShardedParquetLoader

    t = pa.table({"content": [b"x"] * 10, "label_idx": list(range(10))})
    pq.write_table(t, tmp_path / "p.parquet", row_group_size=10)  # ONE group
    seen = []
    for r in range(4):
        ld = ShardedParquetLoader(
            str(tmp_path), batch_size=3, cur_shard=r, shard_count=4,
            num_epochs=1, transform=lambda c: __import__("torch").zeros(1),
        )
        labels = [l for _, l in ld._iter_rows()]
        assert labels, f"rank {r} got an empty shard"
        seen.extend(labels)
    assert sorted(seen) == list(range(10))  # disjoint + exhaustive


def test_loader_process_pool_matches_thread_pool(ddlw_home):
    """The forked decode pool must deliver byte-identical batches in the
    same deterministic order as the thread path (seq reorder buffer)."""
    tbl = _make_table(40, seed=9)
    conv = make_converter(tbl, row_group_rows=8)
    kw = dict(batch_size=8, cur_shard=0, shard_count=1, num_epochs=1,
              img_height=16, img_width=16)
    with conv.make_torch_dataset(pool="thread", **kw) as loader:
        thread_batches = [(i.clone(), l.clone()) for i, l in loader]
    with conv.make_torch_dataset(pool="process", workers_count=3, **kw) as loader:
        proc_batches = [(i.clone(), l.clone()) for i, l in loader]
    assert len(proc_batches) == len(thread_batches)
    for (ti, tl), (pi, pl) in zip(thread_batches, proc_batches):
        assert torch.equal(ti, pi)
        assert torch.equal(tl, pl)
    conv.delete()


def test_loader_process_pool_infinite_cycling(ddlw_home):
    tbl = _make_table(16, seed=4)
    conv = make_converter(tbl, row_group_rows=8)
    with conv.make_torch_dataset(
        batch_size=8, num_epochs=None, img_height=16, img_width=16,
        pool="process", workers_count=2,
    ) as loader:
        it = iter(loader)
        batches = [next(it) for _ in range(5)]  # > 2 epochs worth
        it.close()
    assert all(b[0].shape[0] == 8 for b in batches)
    conv.delete()


def test_loader_process_pool_surfaces_decode_error(ddlw_home):
    """A transform raising in a worker must raise in the consumer, not hang."""
    tbl = _make_table(8, seed=1)
    conv = make_converter(tbl, row_group_rows=8)

    with conv.make_torch_dataset(
        batch_size=4, num_epochs=1, pool="process", workers_count=2,
        transform=_boom,
    ) as loader:
        with pytest.raises(RuntimeError, match="decode failed"):
            list(loader)
    conv.delete()


def _boom(content):
    # raise only inside the forked decode workers (the main-process shape
    # probe must pass so the pool actually starts)
    import multiprocessing as mp

    if mp.current_process().daemon:
        raise ValueError("bad jpeg")
    return np.zeros((4, 4, 3), np.uint8)


def test_loader_finite_multi_epoch_row_count(ddlw_home):
    """num_epochs=2 yields exactly 2x the rows, thread and process pools."""
    tbl = _make_table(24, seed=6)
    conv = make_converter(tbl, row_group_rows=8)
    for pool in ("thread", "process"):
        n = 0
        with conv.make_torch_dataset(
            batch_size=5, num_epochs=2, img_height=16, img_width=16,
            pool=pool, workers_count=2,
        ) as loader:
            for imgs, labels in loader:
                n += labels.numel()
        assert n == 48, (pool, n)
    conv.delete()
