"""Notebook display helpers — the analog of the reference's
DisplayUtils.py (caffe-grid/src/main/python/com/yahoo/ml/caffe/
DisplayUtils.py): render an image/caption or image/label DataFrame as an
HTML table for Jupyter.
"""

from __future__ import annotations

import base64
from typing import Optional


def df_to_html(df, image_col: str = "data", text_col: Optional[str] = None,
               limit: int = 20) -> str:
    """Rows -> <table> with inline base64 <img> tags.  `image_col` holds
    encoded image bytes (or raw arrays, rendered as PNG via PIL when
    available); `text_col` (caption/label/prediction) is shown beside."""
    rows = []
    for _, row in df.head(limit).iterrows():
        img = row.get(image_col) if hasattr(row, "get") else row[image_col]
        cell = ""
        if img is not None:
            data = bytes(img) if not isinstance(img, (bytes, bytearray)) \
                else img
            b64 = base64.b64encode(data).decode("ascii")
            cell = (f'<img src="data:image;base64,{b64}" '
                    f'style="max-height:120px"/>')
        text = "" if text_col is None else str(row[text_col])
        rows.append(f"<tr><td>{cell}</td><td>{text}</td></tr>")
    head = (f"<tr><th>{image_col}</th>"
            f"<th>{text_col or ''}</th></tr>")
    return f"<table>{head}{''.join(rows)}</table>"


def show_df(df, **kw):
    """Display inside Jupyter (no-op fallback to returning HTML text)."""
    html = df_to_html(df, **kw)
    try:
        from IPython.display import HTML, display
        display(HTML(html))
    except ImportError:
        return html
