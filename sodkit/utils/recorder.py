"""Observability recorders — parity with the reference's utils/recorder.py
(TBRecorder, Timer, XLSXRecorder).  TensorBoard is an
optional dependency (absent in this image: the recorder degrades to a no-op
with a one-line notice); results tables write CSV and XLSX unconditionally —
the XLSX comes from the dependency-free writer below (an .xlsx file is a zip
of a few fixed XML parts plus one worksheet), so the reference's results.xlsx
artifact exists even with no pandas/openpyxl engine installed."""

from __future__ import annotations

import csv
import os
import zipfile
from xml.sax.saxutils import escape

_XLSX_CONTENT_TYPES = (
    '<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
    '<Types xmlns="http://schemas.openxmlformats.org/package/2006/content-types">'
    '<Default Extension="rels" ContentType="application/vnd.openxmlformats-package.relationships+xml"/>'
    '<Default Extension="xml" ContentType="application/xml"/>'
    '<Override PartName="/xl/workbook.xml" ContentType="application/vnd.openxmlformats-officedocument.spreadsheetml.sheet.main+xml"/>'
    '<Override PartName="/xl/worksheets/sheet1.xml" ContentType="application/vnd.openxmlformats-officedocument.spreadsheetml.worksheet+xml"/>'
    '</Types>')
_XLSX_RELS = (
    '<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
    '<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">'
    '<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/officeDocument" Target="xl/workbook.xml"/>'
    '</Relationships>')
_XLSX_WORKBOOK = (
    '<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
    '<workbook xmlns="http://schemas.openxmlformats.org/spreadsheetml/2006/main" '
    'xmlns:r="http://schemas.openxmlformats.org/officeDocument/2006/relationships">'
    '<sheets><sheet name="results" sheetId="1" r:id="rId1"/></sheets></workbook>')
_XLSX_WORKBOOK_RELS = (
    '<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
    '<Relationships xmlns="http://schemas.openxmlformats.org/package/2006/relationships">'
    '<Relationship Id="rId1" Type="http://schemas.openxmlformats.org/officeDocument/2006/relationships/worksheet" Target="worksheets/sheet1.xml"/>'
    '</Relationships>')


def _xlsx_cell(value) -> str:
    if isinstance(value, (int, float)) and not isinstance(value, bool):
        return f'<c t="n"><v>{value!r}</v></c>'
    return f'<c t="inlineStr"><is><t>{escape(str(value))}</t></is></c>'


def write_xlsx(path: str, rows: list[list]) -> None:
    """Write ``rows`` (first row = header) as a single-sheet .xlsx."""
    body = "".join(
        "<row>" + "".join(_xlsx_cell(v) for v in row) + "</row>" for row in rows)
    sheet = ('<?xml version="1.0" encoding="UTF-8" standalone="yes"?>'
             '<worksheet xmlns="http://schemas.openxmlformats.org/spreadsheetml/2006/main">'
             f'<sheetData>{body}</sheetData></worksheet>')
    with zipfile.ZipFile(path, "w", zipfile.ZIP_DEFLATED) as z:
        z.writestr("[Content_Types].xml", _XLSX_CONTENT_TYPES)
        z.writestr("_rels/.rels", _XLSX_RELS)
        z.writestr("xl/workbook.xml", _XLSX_WORKBOOK)
        z.writestr("xl/_rels/workbook.xml.rels", _XLSX_WORKBOOK_RELS)
        z.writestr("xl/worksheets/sheet1.xml", sheet)


class TBRecorder:
    def __init__(self, logdir: str, enabled: bool = True):
        self.writer = None
        os.makedirs(logdir, exist_ok=True)
        if not enabled:
            return
        try:
            from torch.utils.tensorboard import SummaryWriter  # type: ignore
            self.writer = SummaryWriter(logdir)
        except Exception:
            print("[sodkit] tensorboard unavailable; TB recording disabled")

    def scalar(self, tag: str, value: float, step: int) -> None:
        if self.writer is not None:
            self.writer.add_scalar(tag, value, step)

    def image(self, tag: str, img, step: int) -> None:
        """img: [C,H,W] or [H,W] tensor in [0,1] (reference TB parity:
        periodic input/pred/mask image panels)."""
        if self.writer is not None:
            if img.ndim == 2:
                img = img.unsqueeze(0)
            self.writer.add_image(tag, img.float().clamp(0, 1), step)

    def close(self) -> None:
        if self.writer is not None:
            self.writer.close()


class ResultsRecorder:
    """Per-dataset metric table -> results.csv (+ results.xlsx if possible)."""

    def __init__(self, out_dir: str):
        self.out_dir = out_dir
        self.rows: list[dict] = []

    def add(self, dataset: str, metrics: dict) -> None:
        self.rows.append({"dataset": dataset, **metrics})

    def save(self) -> None:
        if not self.rows:
            return
        os.makedirs(self.out_dir, exist_ok=True)
        keys = list(self.rows[0].keys())
        with open(os.path.join(self.out_dir, "results.csv"), "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=keys)
            w.writeheader()
            for r in self.rows:
                w.writerow(r)
        write_xlsx(os.path.join(self.out_dir, "results.xlsx"),
                   [keys] + [[r.get(k, "") for k in keys] for r in self.rows])
