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
    assert len(dl) == np.ceil(100 / 16)


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
