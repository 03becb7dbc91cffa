"""Offline document backends for the document/pdf tool family.

The reference serves read_document / edit_document / create_document /
document_convert / document_extract / document_merge from an 11.6k-LoC Node
sidecar (browser/startDocumentReaderServer.cjs, port 3008).  This module is
the stdlib-only MI355X-engine equivalent: docx/xlsx through zipfile +
minimal OOXML, PDF text through zlib-decompressed content streams, and the
plain-text family (txt/md/csv/html/json) directly.  Genuinely-network tools
(web_search, fetch_url, vision) stay structured offline errors; pdf_operation
(merge/split/watermark — binary PDF rewriting) is out of the offline subset.
"""

from __future__ import annotations

import html
import io
import os
import re
import zipfile
import zlib
from typing import Dict, List, Optional

TEXT_EXTS = {".txt", ".md", ".markdown", ".csv", ".tsv", ".html", ".htm",
             ".json", ".xml", ".yaml", ".yml", ".log", ".rst"}

_W_NS = "http://schemas.openxmlformats.org/wordprocessingml/2006/main"


class DocumentError(RuntimeError):
    pass


# ---------------------------------------------------------------------------
# Readers
# ---------------------------------------------------------------------------

def _docx_text(path: str) -> str:
    with zipfile.ZipFile(path) as z:
        try:
            xml = z.read("word/document.xml").decode("utf-8", "replace")
        except KeyError:
            raise DocumentError("not a .docx (word/document.xml missing)")
    paras: List[str] = []
    for pm in re.finditer(r"<w:p[ >].*?</w:p>|<w:p/>", xml, re.S):
        chunk = pm.group(0)
        runs = re.findall(r"<w:t(?: [^>]*)?>(.*?)</w:t>", chunk, re.S)
        text = html.unescape("".join(runs))
        if re.search(r'w:val="Heading(\d)"', chunk):
            lvl = int(re.search(r'w:val="Heading(\d)"', chunk).group(1))
            text = "#" * lvl + " " + text
        paras.append(text)
    return "\n".join(paras)


def _xlsx_text(path: str) -> str:
    with zipfile.ZipFile(path) as z:
        shared: List[str] = []
        if "xl/sharedStrings.xml" in z.namelist():
            sx = z.read("xl/sharedStrings.xml").decode("utf-8", "replace")
            shared = [html.unescape(m) for m in
                      re.findall(r"<t(?: [^>]*)?>(.*?)</t>", sx, re.S)]
        sheets = sorted(n for n in z.namelist()
                        if re.match(r"xl/worksheets/sheet\d+\.xml$", n))
        out: List[str] = []
        for sn in sheets:
            xml = z.read(sn).decode("utf-8", "replace")
            for rm in re.finditer(r"<row[ >].*?</row>", xml, re.S):
                cells: List[str] = []
                for cm in re.finditer(
                        r'<c(?:\s+[^>]*?)?(?:\s+t="(\w+)")?[^>]*>(.*?)</c>',
                        rm.group(0), re.S):
                    ctype, body = cm.group(1), cm.group(2)
                    v = re.search(r"<v>(.*?)</v>", body, re.S)
                    t = re.search(r"<t(?: [^>]*)?>(.*?)</t>", body, re.S)
                    if ctype == "s" and v:
                        idx = int(v.group(1))
                        cells.append(shared[idx] if idx < len(shared) else "")
                    elif t:
                        cells.append(html.unescape(t.group(1)))
                    elif v:
                        cells.append(v.group(1))
                    else:
                        cells.append("")
                out.append(",".join(cells))
            out.append("")
        return "\n".join(out)


def _pdf_text(path: str) -> str:
    """Best-effort text extraction: decompress FlateDecode content streams
    and collect Tj/TJ show-text operators.  Covers straightforwardly
    generated PDFs; image-only or exotic-encoding PDFs yield little."""
    with open(path, "rb") as f:
        data = f.read()
    texts: List[str] = []
    for m in re.finditer(rb"stream\r?\n(.*?)\r?\nendstream", data, re.S):
        raw = m.group(1)
        try:
            raw = zlib.decompress(raw)
        except zlib.error:
            pass
        if b"BT" not in raw:
            continue
        for tm in re.finditer(rb"\((?:[^()\\]|\\.)*\)\s*Tj|\[(?:[^\]\\]|\\.)*\]\s*TJ",
                              raw, re.S):
            chunk = tm.group(0)
            for sm in re.finditer(rb"\(((?:[^()\\]|\\.)*)\)", chunk, re.S):
                s = sm.group(1)
                s = re.sub(rb"\\([nrtbf()\\])",
                           lambda g: {b"n": b"\n", b"r": b"\r", b"t": b"\t",
                                      b"b": b"\b", b"f": b"\f", b"(": b"(",
                                      b")": b")", b"\\": b"\\"}[g.group(1)], s)
                texts.append(s.decode("latin-1", "replace"))
        texts.append("\n")
    return "".join(texts).strip()


def pdf_page_count(path: str) -> int:
    with open(path, "rb") as f:
        data = f.read()
    return len(re.findall(rb"/Type\s*/Page[^s]", data))


def read_document(path: str) -> str:
    ext = os.path.splitext(path)[1].lower()
    if ext == ".docx":
        return _docx_text(path)
    if ext == ".xlsx":
        return _xlsx_text(path)
    if ext == ".pdf":
        return _pdf_text(path)
    if ext in TEXT_EXTS or ext == "":
        with open(path, "r", encoding="utf-8", errors="replace") as f:
            return f.read()
    raise DocumentError(f"unsupported document type {ext!r} "
                        "(offline subset: docx, xlsx, pdf, text formats)")


# ---------------------------------------------------------------------------
# Writers
# ---------------------------------------------------------------------------

_CONTENT_TYPES_DOCX = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Types xmlns="http://schemas.openxmlformats.org/package/2006/content-types">
<Default Extension="rels" ContentType="application/vnd.openxmlformats-package.relationships+xml"/>
<Default Extension="xml" ContentType="application/xml"/>
<Override PartName="/word/document.xml" ContentType="application/vnd.openxmlformats-officedocument.wordprocessingml.document.main+xml"/>
<Override PartName="/word/styles.xml" ContentType="application/vnd.openxmlformats-officedocument.wordprocessingml.styles+xml"/>
</Types>"""

_RELS = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">
<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/officeDocument" Target="word/document.xml"/>
</Relationships>"""

_DOC_RELS = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">
<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/styles" Target="styles.xml"/>
</Relationships>"""

_STYLES = ("""<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<w:styles xmlns:w="%s">""" % _W_NS) + "".join(
    f'<w:style w:type="paragraph" w:styleId="Heading{i}">'
    f'<w:name w:val="heading {i}"/><w:rPr><w:b/><w:sz w:val="{40 - 4 * i}"/>'
    "</w:rPr></w:style>" for i in range(1, 7)) + "</w:styles>"


def _md_to_docx_xml(content: str) -> str:
    body: List[str] = []
    for line in content.split("\n"):
        hm = re.match(r"^(#{1,6})\s+(.*)$", line)
        style = ""
        if hm:
            style = f'<w:pPr><w:pStyle w:val="Heading{len(hm.group(1))}"/></w:pPr>'
            line = hm.group(2)
        # minimal inline markdown: **bold** runs
        runs: List[str] = []
        pos = 0
        for bm in re.finditer(r"\*\*(.+?)\*\*", line):
            if bm.start() > pos:
                runs.append(f'<w:r><w:t xml:space="preserve">{html.escape(line[pos:bm.start()])}</w:t></w:r>')
            runs.append(f'<w:r><w:rPr><w:b/></w:rPr><w:t xml:space="preserve">{html.escape(bm.group(1))}</w:t></w:r>')
            pos = bm.end()
        if pos < len(line):
            runs.append(f'<w:r><w:t xml:space="preserve">{html.escape(line[pos:])}</w:t></w:r>')
        body.append(f"<w:p>{style}{''.join(runs)}</w:p>")
    return ('<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
            f'<w:document xmlns:w="{_W_NS}"><w:body>{"".join(body)}'
            "</w:body></w:document>")


def write_docx(path: str, content: str) -> None:
    with zipfile.ZipFile(path, "w", zipfile.ZIP_DEFLATED) as z:
        z.writestr("[Content_Types].xml", _CONTENT_TYPES_DOCX)
        z.writestr("_rels/.rels", _RELS)
        z.writestr("word/_rels/document.xml.rels", _DOC_RELS)
        z.writestr("word/styles.xml", _STYLES)
        z.writestr("word/document.xml", _md_to_docx_xml(content))


_CONTENT_TYPES_XLSX = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Types xmlns="http://schemas.openxmlformats.org/package/2006/content-types">
<Default Extension="rels" ContentType="application/vnd.openxmlformats-package.relationships+xml"/>
<Default Extension="xml" ContentType="application/xml"/>
<Override PartName="/xl/workbook.xml" ContentType="application/vnd.openxmlformats-officedocument.spreadsheetml.sheet.main+xml"/>
<Override PartName="/xl/worksheets/sheet1.xml" ContentType="application/vnd.openxmlformats-officedocument.spreadsheetml.worksheet+xml"/>
</Types>"""

_RELS_XLSX = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">
<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/officeDocument" Target="xl/workbook.xml"/>
</Relationships>"""

_WB = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<workbook xmlns="http://schemas.openxmlformats.org/spreadsheetml/2006/main" xmlns:r="http://schemas.openxmlformats.org/officeDocument/2006/relationships">
<sheets><sheet name="Sheet1" sheetId="1" r:id="rId1"/></sheets></workbook>"""

_WB_RELS = """<?xml version="1.0" encoding="UTF-8" standalone="yes"?>
<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">
<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/worksheet" Target="worksheets/sheet1.xml"/>
</Relationships>"""


def _col_name(i: int) -> str:
    s = ""
    i += 1
    while i:
        i, r = divmod(i - 1, 26)
        s = chr(65 + r) + s
    return s


def write_xlsx(path: str, rows: List[List[str]]) -> None:
    xrows: List[str] = []
    for ri, row in enumerate(rows, start=1):
        cells = []
        for ci, val in enumerate(row):
            ref = f"{_col_name(ci)}{ri}"
            sval = str(val)
            try:
                float(sval)
                cells.append(f'<c r="{ref}"><v>{sval}</v></c>')
            except ValueError:
                cells.append(f'<c r="{ref}" t="inlineStr"><is><t xml:space="preserve">'
                             f"{html.escape(sval)}</t></is></c>")
        xrows.append(f'<row r="{ri}">{"".join(cells)}</row>')
    sheet = ('<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
             '<worksheet xmlns="http://schemas.openxmlformats.org/spreadsheetml/2006/main">'
             f'<sheetData>{"".join(xrows)}</sheetData></worksheet>')
    with zipfile.ZipFile(path, "w", zipfile.ZIP_DEFLATED) as z:
        z.writestr("[Content_Types].xml", _CONTENT_TYPES_XLSX)
        z.writestr("_rels/.rels", _RELS_XLSX)
        z.writestr("xl/workbook.xml", _WB)
        z.writestr("xl/_rels/workbook.xml.rels", _WB_RELS)
        z.writestr("xl/worksheets/sheet1.xml", sheet)


def create_document(doc_type: str, path: str, document_data: str) -> str:
    doc_type = (doc_type or "word").lower()
    if doc_type in ("word", "docx"):
        if not path.endswith(".docx"):
            path += ".docx"
        write_docx(path, document_data or "")
        return path
    if doc_type in ("excel", "xlsx"):
        if not path.endswith(".xlsx"):
            path += ".xlsx"
        rows = [ln.split(",") for ln in (document_data or "").splitlines()]
        write_xlsx(path, rows)
        return path
    raise DocumentError(f"create_document type {doc_type!r} not in the "
                        "offline subset (word, excel)")


def edit_document(path: str, content: Optional[str],
                  replacements: Optional[List[Dict[str, str]]],
                  backup: bool = False) -> str:
    ext = os.path.splitext(path)[1].lower()
    if backup and os.path.exists(path):
        with open(path, "rb") as f:
            data = f.read()
        with open(path + ".bak", "wb") as f:
            f.write(data)
    if replacements:
        text = read_document(path)
        for rep in replacements:
            find = rep.get("find") or rep.get("old_text") or ""
            repl = rep.get("replace") or rep.get("new_text") or ""
            if find and find not in text:
                raise DocumentError(f"replacement target not found: {find[:60]!r}")
            text = text.replace(find, repl)
        content = text
    if content is None:
        raise DocumentError("edit_document needs content or replacements")
    if ext == ".docx":
        write_docx(path, content)
    elif ext in TEXT_EXTS or ext == "":
        with open(path, "w", encoding="utf-8") as f:
            f.write(content)
    else:
        raise DocumentError(f"edit_document: unsupported type {ext!r}")
    return path


def convert_document(src: str, dst: str, fmt: Optional[str] = None) -> str:
    """Text-target conversions: {docx,xlsx,pdf,md,txt,html} -> {txt,md,html,docx}."""
    fmt = (fmt or os.path.splitext(dst)[1].lstrip(".")).lower()
    text = read_document(src)
    if fmt in ("txt", "text", "md", "markdown", "csv"):
        with open(dst, "w", encoding="utf-8") as f:
            f.write(text)
    elif fmt in ("html", "htm"):
        body = "".join(f"<p>{html.escape(ln)}</p>\n" for ln in text.splitlines())
        with open(dst, "w", encoding="utf-8") as f:
            f.write(f"<!DOCTYPE html>\n<html><body>\n{body}</body></html>\n")
    elif fmt in ("docx", "word"):
        write_docx(dst, text)
    else:
        raise DocumentError(f"document_convert target {fmt!r} not in the "
                            "offline subset (txt, md, html, docx)")
    return dst


def extract_document(src: str, out_dir: str, extract_type: str = "text") -> List[str]:
    os.makedirs(out_dir, exist_ok=True)
    base = os.path.splitext(os.path.basename(src))[0]
    written: List[str] = []
    if extract_type in ("text", "all", "", None):
        txt = read_document(src)
        p = os.path.join(out_dir, base + ".txt")
        with open(p, "w", encoding="utf-8") as f:
            f.write(txt)
        written.append(p)
    if extract_type in ("images", "all") and src.lower().endswith((".docx", ".xlsx")):
        with zipfile.ZipFile(src) as z:
            for name in z.namelist():
                if re.match(r"(word|xl)/media/", name):
                    p = os.path.join(out_dir, os.path.basename(name))
                    with open(p, "wb") as f:
                        f.write(z.read(name))
                    written.append(p)
    return written


def merge_documents(srcs: List[str], dst: str) -> str:
    texts = [read_document(s) for s in srcs]
    merged = "\n\n".join(texts)
    ext = os.path.splitext(dst)[1].lower()
    if ext == ".docx":
        write_docx(dst, merged)
    else:
        with open(dst, "w", encoding="utf-8") as f:
            f.write(merged)
    return dst
