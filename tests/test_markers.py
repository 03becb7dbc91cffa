"""MarkerService + MarkerCheckService (VERDICT r01 partial #48).

Reference: browser/_markerCheckService.ts — Error-severity filter + the
quick-fix code-action provider loop.
"""

from senweaver_amd.features.markers import (
    ERROR, WARNING, CodeAction, Marker, MarkerCheckService, MarkerService,
    python_lint,
)
from senweaver_amd.tools.service import ToolsService


def test_marker_store_and_severity_filter():
    ms = MarkerService()
    changes = []
    ms.on_marker_changed(changes.append)
    ms.changed("a.py", [Marker("a.py", ERROR, "bad", 3),
                        Marker("a.py", WARNING, "meh", 5)])
    assert len(ms.read()) == 2
    assert [m.message for m in ms.read(severity=ERROR)] == ["bad"]
    ms.changed("a.py", [])  # clearing removes the resource
    assert ms.read() == []
    assert changes == ["a.py", "a.py"]


def test_python_lint_provider():
    ms = python_lint("x.py", "def broken(:\n  pass\n")
    assert ms and ms[0].severity == ERROR and ms[0].startLineNumber == 1
    assert python_lint("ok.py", "a = 1\n") == []


def test_check_runs_quickfix_providers():
    store = MarkerService()
    store.changed("f.py", [Marker("f.py", ERROR, "syntax error", 2)])
    store.changed("g.py", [Marker("g.py", WARNING, "style", 1)])
    svc = MarkerCheckService(store)
    svc.register_code_action_provider(
        "python", lambda m, text: [CodeAction(f"fix line {m.startLineNumber}")])

    res = svc.check(get_text=lambda uri: "content",
                    language_of=lambda uri: "python")
    # only ERROR markers are checked; the provider's action is attached
    assert len(res) == 1
    assert res[0]["marker"].resource == "f.py"
    assert res[0]["actions"][0].title == "fix line 2"


def test_read_lint_errors_publishes_markers(tmp_path):
    (tmp_path / "bad.py").write_text("def f(:\n")
    ts = ToolsService(str(tmp_path))
    out = ts.call_tool("read_lint_errors", {"uri": "bad.py"})
    assert out.result["errors"]
    assert ts.marker_service.read(severity=ERROR)
    # fixing the file clears its markers on the next read
    (tmp_path / "bad.py").write_text("def f():\n    pass\n")
    out = ts.call_tool("read_lint_errors", {"uri": "bad.py"})
    assert out.text == "No lint errors found."
    assert ts.marker_service.read() == []
