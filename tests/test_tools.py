"""Tests for tools/ (profiling post-processing)."""
import sqlite3
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def test_rocpd_to_kernel_stats_roundtrip(tmp_path):
    from tools.rocpd_to_kernel_stats import convert
    db = tmp_path / "r.db"
    c = sqlite3.connect(db)
    sfx = "deadbeef"
    c.execute(f"CREATE TABLE rocpd_info_kernel_symbol_{sfx} "
              "(id INT, display_name TEXT)")
    c.execute(f"CREATE TABLE rocpd_kernel_dispatch_{sfx} "
              "(kernel_id INT, start INT, end INT)")
    c.executemany(f"INSERT INTO rocpd_info_kernel_symbol_{sfx} VALUES (?,?)",
                  [(1, "env_step_kernel"), (2, "actor_kernel")])
    c.executemany(f"INSERT INTO rocpd_kernel_dispatch_{sfx} VALUES (?,?,?)",
                  [(1, 0, 100), (1, 200, 350), (2, 400, 450)])
    c.commit()
    out = tmp_path / "stats.csv"
    rows, tot = convert(str(db), str(out))
    assert tot == 100 + 150 + 50
    assert rows[0][0] == "env_step_kernel" and rows[0][1] == 2
    assert rows[0][2] == 250
    lines = out.read_text().strip().splitlines()
    assert lines[0].startswith('"Name"')
    assert len(lines) == 3


def test_survey_reference_scanner(tmp_path):
    from tools.survey_reference import scan
    (tmp_path / "src").mkdir()
    (tmp_path / "src/env.py").write_text(
        "class Env:\n    def step(self, a):\n        pass\n"
        "    def reset(self):\n        pass\n")
    (tmp_path / "src/train.cu").write_text("__global__ void k() {}\n")
    out = scan(tmp_path)
    assert "2 files" in out
    assert "src/env.py:2" in out
    assert "__global__" in out
    # empty dir
    empty = tmp_path / "empty"
    empty.mkdir()
    assert "EMPTY" in scan(empty)
