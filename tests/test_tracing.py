import json

from rlgpuschedule_amd.utils.tracing import Tracer


def test_tracer_writes_chrome_json(tmp_path):
    tr = Tracer()
    with tr.span("collect", iter=0):
        pass
    with tr.span("update", iter=0):
        pass
    p = tmp_path / "trace.json"
    tr.save(p)
    d = json.loads(p.read_text())
    assert len(d["traceEvents"]) == 2
    assert d["traceEvents"][0]["name"] == "collect"
