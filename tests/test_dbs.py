from datetime import datetime, timedelta

from llmapigateway_amd.db import ModelRotationDB, TokensUsageDB


def test_rotation_wraparound(tmp_path):
    db = ModelRotationDB(tmp_path / "rot.db")
    seq = [db.get_next_model_index("key1", "model-a", 3) for _ in range(7)]
    assert seq == [0, 1, 2, 0, 1, 2, 0]


def test_rotation_keyed_by_api_key_and_model(tmp_path):
    db = ModelRotationDB(tmp_path / "rot.db")
    assert db.get_next_model_index("k1", "m", 4) == 0
    assert db.get_next_model_index("k2", "m", 4) == 0  # independent per api key
    assert db.get_next_model_index("k1", "m2", 4) == 0  # independent per model
    assert db.get_next_model_index("k1", "m", 4) == 1


def test_rotation_persists_across_instances(tmp_path):
    path = tmp_path / "rot.db"
    db = ModelRotationDB(path)
    db.get_next_model_index("k", "m", 5)
    db.get_next_model_index("k", "m", 5)
    db.close()
    db2 = ModelRotationDB(path)
    assert db2.get_next_model_index("k", "m", 5) == 2


def test_rotation_degenerate_total():
    db = ModelRotationDB(":memory:") if False else None
    # total_models <= 0 returns 0 without touching the DB
    import tempfile, os

    with tempfile.TemporaryDirectory() as d:
        rdb = ModelRotationDB(os.path.join(d, "r.db"))
        assert rdb.get_next_model_index("k", "m", 0) == 0


def test_usage_insert_and_count(tmp_path):
    db = TokensUsageDB(tmp_path / "usage.db")
    assert db.get_total_records_count() == 0
    assert db.insert_usage(prompt_tokens=10, completion_tokens=5, model="m1", provider="p1", cost=0.01)
    assert db.insert_usage(prompt_tokens=3, completion_tokens=7, model="m1", provider="p1")
    assert db.get_total_records_count() == 2
    recs = db.get_latest_usage_records(limit=1, offset=0)
    assert len(recs) == 1
    assert recs[0]["total_tokens"] in (10, 15)


def test_usage_aggregation_buckets(tmp_path):
    db = TokensUsageDB(tmp_path / "usage.db")
    t0 = datetime(2026, 3, 10, 14, 5)
    db.insert_usage(prompt_tokens=1, completion_tokens=1, model="m", timestamp=t0)
    db.insert_usage(prompt_tokens=2, completion_tokens=2, model="m", timestamp=t0 + timedelta(minutes=30))
    db.insert_usage(prompt_tokens=4, completion_tokens=4, model="m", timestamp=t0 + timedelta(hours=2))

    by_hour = db.get_aggregated_usage("hour")
    assert len(by_hour) == 2  # 14:00 bucket (2 recs) + 16:00 bucket
    hour_bucket = [r for r in by_hour if r["time_period"] == "2026-03-10 14:00:00"][0]
    assert hour_bucket["prompt_tokens"] == 3 and hour_bucket["count"] == 2

    by_day = db.get_aggregated_usage("day")
    assert len(by_day) == 1 and by_day[0]["total_tokens"] == 14

    by_month = db.get_aggregated_usage("month")
    assert by_month[0]["time_period"] == "2026-03"

    assert db.get_aggregated_usage("bogus") == []


def test_usage_aggregation_date_filter(tmp_path):
    db = TokensUsageDB(tmp_path / "usage.db")
    old = datetime(2020, 1, 1)
    db.insert_usage(prompt_tokens=1, model="m", timestamp=old)
    db.insert_usage(prompt_tokens=1, model="m", timestamp=datetime(2026, 1, 1))
    rows = db.get_aggregated_usage("day", start_date=datetime(2025, 1, 1))
    assert len(rows) == 1 and rows[0]["time_period"] == "2026-01-01"


def test_usage_cleanup(tmp_path):
    db = TokensUsageDB(tmp_path / "usage.db")
    db.insert_usage(prompt_tokens=1, timestamp=datetime.now() - timedelta(days=400))
    db.insert_usage(prompt_tokens=1)
    deleted = db.cleanup_old_records(days=180)
    assert deleted == 1
    assert db.get_total_records_count() == 1


def test_usage_per_model_grouping(tmp_path):
    db = TokensUsageDB(tmp_path / "usage.db")
    t = datetime(2026, 5, 1, 10)
    db.insert_usage(prompt_tokens=1, model="a", timestamp=t)
    db.insert_usage(prompt_tokens=2, model="b", timestamp=t)
    rows = db.get_aggregated_usage("day")
    assert {r["model"] for r in rows} == {"a", "b"}
