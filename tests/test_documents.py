"""Offline document/pdf tool backends.

The reference serves these from the documentReader sidecar (port 3008,
toolsService.ts:2744+); here they are stdlib OOXML/PDF implementations
(tools/documents.py) behind the same tool names/params.
"""

import os
import zipfile
import zlib

import pytest

from senweaver_amd.tools import documents as docs
from senweaver_amd.tools.service import ToolError, ToolsService


@pytest.fixture()
def ts(tmp_path):
    return ToolsService(str(tmp_path))


def test_docx_roundtrip(ts, tmp_path):
    content = "# Report\n\nFirst paragraph with **bold** text.\nplain line"
    ts.call_tool("create_document", {"type": "word", "file_path": "r.docx",
                                     "document_data": content})
    # a real zip with the OOXML parts
    with zipfile.ZipFile(tmp_path / "r.docx") as z:
        assert "word/document.xml" in z.namelist()
        assert "[Content_Types].xml" in z.namelist()
    out = ts.call_tool("read_document", {"uri": "r.docx"})
    assert "# Report" in out.text          # heading style reconstructed
    assert "bold" in out.text and "plain line" in out.text


def test_docx_replacements_and_backup(ts, tmp_path):
    ts.call_tool("create_document", {"type": "word", "file_path": "e.docx",
                                     "document_data": "alpha beta gamma"})
    ts.call_tool("edit_document", {
        "uri": "e.docx", "backup": "true",
        "replacements": '[{"find": "beta", "replace": "delta"}]'})
    assert (tmp_path / "e.docx.bak").exists()
    out = ts.call_tool("read_document", {"uri": "e.docx"})
    assert "delta" in out.text and "beta" not in out.text


def test_replacement_target_missing_errors(ts):
    ts.call_tool("create_document", {"type": "word", "file_path": "x.docx",
                                     "document_data": "abc"})
    with pytest.raises(ToolError, match="not found"):
        ts.call_tool("edit_document", {
            "uri": "x.docx",
            "replacements": '[{"find": "zzz", "replace": "q"}]'})


def test_xlsx_roundtrip(ts):
    ts.call_tool("create_document", {"type": "excel", "file_path": "t.xlsx",
                                     "document_data": "name,qty\nwidget,3"})
    out = ts.call_tool("read_document", {"uri": "t.xlsx"})
    assert "name,qty" in out.text and "widget,3" in out.text


def test_convert_and_merge(ts, tmp_path):
    ts.call_tool("create_document", {"type": "word", "file_path": "a.docx",
                                     "document_data": "doc A"})
    (tmp_path / "b.md").write_text("doc B")
    ts.call_tool("document_convert", {"input_file": "a.docx",
                                      "output_path": "a.html", "format": "html"})
    assert "<p>doc A</p>" in (tmp_path / "a.html").read_text()
    ts.call_tool("document_merge", {"input_files": "a.docx,b.md",
                                    "output_path": "m.txt"})
    m = (tmp_path / "m.txt").read_text()
    assert "doc A" in m and "doc B" in m


def test_read_document_paging(ts, tmp_path):
    (tmp_path / "long.txt").write_text("x" * 100)
    out = ts.call_tool("read_document", {"uri": "long.txt", "start_index": 10,
                                         "max_length": 5})
    assert out.text == "xxxxx"
    assert out.result["total_chars"] == 100


def _tiny_pdf(path, text):
    """Hand-built single-page PDF with a Flate-compressed text stream."""
    stream = f"BT /F1 12 Tf 72 700 Td ({text}) Tj ET".encode()
    comp = zlib.compress(stream)
    objs = []
    objs.append(b"1 0 obj\n<< /Type /Catalog /Pages 2 0 R >>\nendobj\n")
    objs.append(b"2 0 obj\n<< /Type /Pages /Kids [3 0 R] /Count 1 >>\nendobj\n")
    objs.append(b"3 0 obj\n<< /Type /Page /Parent 2 0 R /Contents 4 0 R >>\nendobj\n")
    objs.append(b"4 0 obj\n<< /Length " + str(len(comp)).encode()
                + b" /Filter /FlateDecode >>\nstream\n" + comp
                + b"\nendstream\nendobj\n")
    body = b"%PDF-1.4\n" + b"".join(objs) + b"trailer\n<< /Root 1 0 R >>\n%%EOF\n"
    path.write_bytes(body)


def test_pdf_text_extraction(ts, tmp_path):
    _tiny_pdf(tmp_path / "doc.pdf", "Hello PDF world")
    out = ts.call_tool("read_document", {"uri": "doc.pdf"})
    assert "Hello PDF world" in out.text
    assert docs.pdf_page_count(str(tmp_path / "doc.pdf")) == 1


def test_document_extract_text(ts, tmp_path):
    ts.call_tool("create_document", {"type": "word", "file_path": "d.docx",
                                     "document_data": "extract me"})
    out = ts.call_tool("document_extract", {"input_file": "d.docx",
                                            "output_dir": "ex"})
    assert (tmp_path / "ex" / "d.txt").read_text().strip() == "extract me"


def test_network_tools_still_offline(ts):
    with pytest.raises(ToolError, match="offline"):
        ts.call_tool("web_search", {"query": "x"})
    with pytest.raises(ToolError, match="offline"):
        ts.call_tool("pdf_operation", {"operation": "merge"})


def test_api_request_loopback(ts):
    """api_request executes against a real loopback HTTP server; non-local
    targets stay structured offline errors (CustomApiService pairing)."""
    import http.server
    import threading

    class H(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            body = self.rfile.read(n)
            self.send_response(200)
            self.end_headers()
            self.wfile.write(b"echo:" + body)

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        out = ts.call_tool("api_request", {
            "url": f"http://127.0.0.1:{srv.server_port}/x",
            "method": "POST", "body": "ping"})
        assert "echo:ping" in out.text and out.result["status"] == 200
        with pytest.raises(ToolError, match="loopback"):
            ts.call_tool("api_request", {"url": "https://example.com"})
    finally:
        srv.shutdown()


def test_custom_api_service_registry(tmp_path):
    from senweaver_amd.features.customapi import (
        CustomApiField, CustomApiService, CUSTOM_API_STORAGE_KEY,
    )
    from senweaver_amd.storage import FileStorage

    store = FileStorage(str(tmp_path / "state.json"))
    svc = CustomApiService(storage=store, clock=lambda: 1234)
    api = svc.add_api("Weather", "http://127.0.0.1:9999/w", "POST",
                      "look up weather",
                      fields=[CustomApiField("city", "string", True, "城市名")],
                      response_description="JSON with temp")
    assert svc.get_api(api.id).name == "Weather"
    desc = svc.get_api_list_description()
    assert "## Weather" in desc and "必填" in desc and "api_request" in desc
    # restart-resume through the same storage key (FileStorage flushes on
    # demand, like the reference's 30s/dispose flush cycle)
    store.flush()
    svc2 = CustomApiService(storage=FileStorage(str(tmp_path / "state.json")))
    assert [a.name for a in svc2.get_enabled_apis()] == ["Weather"]
    svc2.update_api(api.id, enabled=False)
    assert svc2.get_enabled_apis() == []
    assert svc2.get_api_list_description() == ""
    svc2.delete_api(api.id)
    assert svc2.state == {"apis": []}
    assert CUSTOM_API_STORAGE_KEY == "senweaver.customApis"


def test_custom_api_description_reaches_system_message():
    """End-to-end prompt surface: CustomApiService -> the converter's
    custom_api_description slot (convertToLLMMessageService pairing)."""
    from senweaver_amd.context.pipeline import ConvertToLLMMessages
    from senweaver_amd.features.customapi import CustomApiService

    svc = CustomApiService()
    svc.add_api("Ticketing", "http://127.0.0.1:8088/t", "POST", "file a ticket")
    conv = ConvertToLLMMessages()
    msg = conv.generate_system_message(
        "agent", custom_api_description=svc.get_api_list_description())
    assert "Ticketing" in msg and "api_request" in msg
