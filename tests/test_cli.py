"""CLI subcommands end-to-end as real processes (reference cmd/nornicdb
cobra commands: init / import / shell / decay / eval — serve is covered
by test_serve_integration.py)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(args, tmp, input_text=None, timeout=90):
    env = dict(os.environ, NORNICDB_EMBEDDER="mock", PYTHONPATH=ROOT)
    return subprocess.run(
        [sys.executable, "-m", "nornicdb_amd", *args],
        cwd=tmp, env=env, input=input_text, capture_output=True,
        text=True, timeout=timeout)


def test_init_creates_datadir(tmp_path):
    d = str(tmp_path / "db")
    r = run_cli(["init", "--data-dir", d], str(tmp_path))
    assert r.returncode == 0, r.stderr
    assert "initialized database" in r.stdout
    assert os.path.isdir(d)


def test_import_then_shell_query(tmp_path):
    d = str(tmp_path / "db")
    export = {
        "nodes": [
            {"id": "a", "labels": ["Person"], "properties": {"name": "Ada"}},
            {"id": "b", "labels": ["Person"], "properties": {"name": "Bo"}},
        ],
        "relationships": [
            {"id": "e1", "type": "KNOWS", "start": "a", "end": "b"},
        ],
    }
    f = tmp_path / "export.json"
    f.write_text(json.dumps(export))
    r = run_cli(["import", "--data-dir", d, "--file", str(f)],
                str(tmp_path))
    assert r.returncode == 0, r.stderr
    assert "imported 2 nodes, 1 relationships" in r.stdout

    # shell reads Cypher from stdin, prints tab-separated rows
    r2 = run_cli(["shell", "--data-dir", d], str(tmp_path),
                 input_text="MATCH (n:Person) RETURN n.name "
                            "ORDER BY n.name\n:quit\n")
    assert r2.returncode == 0, r2.stderr
    assert "Ada" in r2.stdout and "Bo" in r2.stdout


def test_decay_runs_cycle(tmp_path):
    d = str(tmp_path / "db")
    run_cli(["init", "--data-dir", d], str(tmp_path))
    r = run_cli(["decay", "--data-dir", d], str(tmp_path))
    assert r.returncode == 0, r.stderr
    stats = json.loads(r.stdout.strip().splitlines()[-1])
    assert isinstance(stats, dict)


def test_eval_harness(tmp_path):
    d = str(tmp_path / "db")
    # store a couple of memories first via the embedded API
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "from nornicdb_amd.db import open_db\n"
        "from nornicdb_amd.embed import create_embedder\n"
        "m = open_db(%r, embedder=create_embedder('mock', dims=64), dims=64)\n"
        "db = m.get()\n"
        "a = db.store('graph databases store relationships')\n"
        "b = db.store('vector search finds similar embeddings')\n"
        "db.embed_queue.drain()\n"
        "print(a.id); print(b.id)\n"
        "m.close()\n" % (ROOT, d))
    r0 = subprocess.run([sys.executable, "-c", code], capture_output=True,
                        text=True, timeout=90)
    assert r0.returncode == 0, r0.stderr
    ids = r0.stdout.split()
    cases = [{"query": "graph databases", "relevant": [ids[0]]}]
    f = tmp_path / "cases.json"
    f.write_text(json.dumps(cases))
    r = run_cli(["eval", "--data-dir", d, "--cases", str(f)],
                str(tmp_path))
    assert r.returncode == 0, r.stderr
    report = json.loads(r.stdout)
    assert "cases" in report or "mrr" in report or report, report
