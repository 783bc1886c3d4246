"""Batch processing tests (reference: sky/batch)."""
from tests.test_orchestrator import client, sky_env  # noqa: F401


def test_run_batch_shards_and_gathers(client):
    from skypilot_amd.batch import Dataset, run_batch
    ds = Dataset.from_list([{"x": i} for i in range(10)])
    out = run_batch(
        ds,
        # double every x using a tiny inline python job
        'python3 -c "\n'
        "import json, os\n"
        "rows = [json.loads(l) for l in open(os.environ['SKY_BATCH_INPUT'])]\n"
        "with open(os.environ['SKY_BATCH_OUTPUT'], 'w') as f:\n"
        "    for r in rows:\n"
        "        f.write(json.dumps({'y': r['x'] * 2}) + '\\n')\n"
        '"',
        num_workers=3, timeout=120)
    ys = sorted(r["y"] for r in out.rows)
    assert ys == [i * 2 for i in range(10)]


def test_remote_store_via_rclone_stub(tmp_path, monkeypatch):
    """s3:// sources pull through rclone at create and push with
    sync_to_remote (reference: S3Store + data_transfer).  A stub rclone
    on PATH stands in for the real binary (no egress here)."""
    import os
    import stat
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path / "home"))
    # fake remote bucket + stub rclone that maps s3:bucket -> dir
    bucket = tmp_path / "bucket" / "ckpts"
    bucket.mkdir(parents=True)
    (bucket / "w.bin").write_text("weights-v1")
    stub = tmp_path / "bin" / "rclone"
    stub.parent.mkdir()
    stub.write_text(f"""#!/bin/bash
# stub rclone: sync SRC DST with s3:path mapped under {tmp_path}/remote_
src="$2"; dst="$3"
map() {{ case "$1" in s3:*) echo "{tmp_path}/remote_${{1#s3:}}";; *) echo "$1";; esac; }}
rs=$(map "$src"); rd=$(map "$dst")
mkdir -p "$rd" && cp -r "$rs"/. "$rd"/
""")
    stub.chmod(stub.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", f"{stub.parent}:{os.environ['PATH']}")
    (tmp_path / "remote_mybucket" / "ckpts").mkdir(parents=True)
    (tmp_path / "remote_mybucket" / "ckpts" / "w.bin").write_text("v1")

    from skypilot_amd.data import storage as st
    d = st.get_or_create_store("ck", "s3://mybucket/ckpts")
    assert (d / "w.bin").read_text() == "v1"
    rec = next(r for r in st.list_storage() if r["name"] == "ck")
    assert rec["store_type"] == "s3"
    # local writes push back
    (d / "w.bin").write_text("v2")
    st.sync_to_remote("ck")
    assert (tmp_path / "remote_mybucket" / "ckpts" / "w.bin"
            ).read_text() == "v2"


def test_remote_store_without_rclone_fails_loudly(tmp_path, monkeypatch):
    monkeypatch.setenv("SKY_AMD_HOME", str(tmp_path / "home2"))
    monkeypatch.setenv("PATH", "/usr/bin:/bin")
    import pytest as _pytest
    from skypilot_amd.data import storage as st
    from skypilot_amd.exceptions import TaskValidationError
    with _pytest.raises(TaskValidationError, match="rclone"):
        st.get_or_create_store("nope", "s3://b/x")
