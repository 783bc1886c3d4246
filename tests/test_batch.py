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
