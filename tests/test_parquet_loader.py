import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq

from maggy_amd.parallel.data import MaggyParquetDataLoader


def _write(tmp_path, rows=100, row_groups=4):
    per = rows // row_groups
    tables = []
    for g in range(row_groups):
        ids = np.arange(g * per, (g + 1) * per)
        tables.append(pa.table({
            "x": ids.astype(np.float32) * 0.5,
            "label": (ids % 2).astype(np.int64),
        }))
    path = str(tmp_path / "data.parquet")
    with pq.ParquetWriter(path, tables[0].schema) as w:
        for t in tables:
            w.write_table(t)
    return path


def test_parquet_loader_single_rank(tmp_path):
    path = _write(tmp_path)
    dl = MaggyParquetDataLoader(path, batch_size=16, rank=0, world_size=1)
    seen = []
    for batch in dl:
        assert set(batch.keys()) == {"x", "label"}
        seen.extend(batch["x"].tolist())
    assert len(seen) == 100
    # len() matches the actual iteration count (batches never span row
    # groups: 4 groups x ceil(25/16) = 8)
    assert len(dl) == 8
    assert len(list(dl)) == 8


def test_parquet_loader_sharding(tmp_path):
    path = _write(tmp_path)
    all_seen = []
    for rank in range(2):
        dl = MaggyParquetDataLoader(path, batch_size=16, rank=rank,
                                    world_size=2)
        for batch in dl:
            all_seen.extend(batch["x"].tolist())
    assert len(all_seen) == 100
    assert len(set(all_seen)) == 100  # disjoint shards


def test_parquet_loader_equal_batches_uneven_groups(tmp_path):
    """3 row groups over 2 ranks: every rank must yield the SAME number of
    batches (pad by wrapping) or lockstep DDP hangs."""
    path = _write(tmp_path, rows=90, row_groups=3)  # 30 rows per group
    loaders = [MaggyParquetDataLoader(path, batch_size=16, rank=r,
                                      world_size=2) for r in range(2)]
    counts = [len(list(dl)) for dl in loaders]
    assert counts[0] == counts[1] == len(loaders[0]) == len(loaders[1])
    # rank 0 holds groups {0, 2} (4 batches), rank 1 group {1} padded to 4
    assert counts[0] == 4


def test_parquet_loader_more_ranks_than_groups(tmp_path):
    """world_size > num_row_groups: surplus ranks wrap onto existing
    groups instead of yielding nothing."""
    path = _write(tmp_path, rows=40, row_groups=2)  # 20 rows per group
    loaders = [MaggyParquetDataLoader(path, batch_size=8, rank=r,
                                      world_size=4) for r in range(4)]
    counts = [len(list(dl)) for dl in loaders]
    assert len(set(counts)) == 1
    assert counts[0] == 3  # ceil(20/8)
