import csv
import json
import os

from torchbeast_amd.core.file_writer import FileWriter


def test_creates_files_and_logs(tmp_path):
    fw = FileWriter(xpid="xp1", xp_args={"lr": 0.1}, rootdir=str(tmp_path))
    fw.log({"step": 0, "loss": 1.5})
    fw.log({"step": 8, "loss": 1.0})
    fw.close()

    base = tmp_path / "xp1"
    assert (base / "meta.json").exists()
    assert (base / "logs.csv").exists()
    assert (base / "fields.csv").exists()

    meta = json.loads((base / "meta.json").read_text())
    assert meta["args"]["lr"] == 0.1
    assert meta["successful"] is True

    with open(base / "logs.csv") as f:
        rows = [r for r in csv.reader(f) if r and not r[0].startswith("#")]
    assert len(rows) == 2


def test_schema_grows_dynamically(tmp_path):
    fw = FileWriter(xpid="xp2", xp_args={}, rootdir=str(tmp_path))
    fw.log({"a": 1})
    fw.log({"a": 2, "b": 3})
    fw.close()
    fields = (tmp_path / "xp2" / "fields.csv").read_text()
    assert "b" in fields


def test_resume_continues_tick(tmp_path):
    fw = FileWriter(xpid="xp3", xp_args={}, rootdir=str(tmp_path))
    fw.log({"a": 1})
    fw.log({"a": 2})
    fw.close()

    fw2 = FileWriter(xpid="xp3", xp_args={}, rootdir=str(tmp_path))
    assert fw2._tick == 2
    fw2.log({"a": 3})
    fw2.close()


def test_latest_symlink(tmp_path):
    FileWriter(xpid="xp4", xp_args={}, rootdir=str(tmp_path)).close()
    link = tmp_path / "latest"
    assert link.is_symlink()
    assert os.path.realpath(link) == os.path.realpath(tmp_path / "xp4")
