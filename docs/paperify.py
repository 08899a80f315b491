#!/usr/bin/env python3
"""Render docs/TUTORIAL.md to docs/tutorial.html (+ docs/index.html,
byte-identical copy — the reference's Makefile does the same with its
tuto.html, Makefile:4-7; its own paperify.py is absent from the repo,
so this one is written from scratch).

Dependency-free markdown subset: ATX headings, fenced code blocks,
tables, unordered/ordered lists, blockquotes, emphasis/strong, inline
code, links, horizontal rules.  Run:  python docs/paperify.py
"""

import html
import os
import re
import shutil
import sys

HERE = os.path.dirname(os.path.abspath(__file__))


def _inline(s: str) -> str:
    s = html.escape(s, quote=False)
    # inline code first (protects everything inside)
    parts = re.split(r"(`[^`]+`)", s)
    out = []
    for p in parts:
        if p.startswith("`") and p.endswith("`") and len(p) > 1:
            out.append(f"<code>{p[1:-1]}</code>")
            continue
        p = re.sub(r"\*\*([^*]+)\*\*", r"<strong>\1</strong>", p)
        p = re.sub(r"(?<!\*)\*([^*\s][^*]*)\*(?!\*)", r"<em>\1</em>", p)
        p = re.sub(r"\[([^\]]+)\]\(([^)\s]+)\)", r'<a href="\2">\1</a>', p)
        out.append(p)
    return "".join(out)


def render(md: str, title: str) -> str:
    lines = md.split("\n")
    out = []
    i = 0
    in_list = None      # "ul" | "ol" | None
    para = []

    def flush_para():
        nonlocal para
        if para:
            out.append("<p>" + _inline(" ".join(para)) + "</p>")
            para = []

    def close_list():
        nonlocal in_list
        if in_list:
            out.append(f"</{in_list}>")
            in_list = None

    while i < len(lines):
        ln = lines[i]
        if ln.startswith("```"):
            flush_para()
            close_list()
            code = []
            i += 1
            while i < len(lines) and not lines[i].startswith("```"):
                code.append(lines[i])
                i += 1
            out.append("<pre><code>" +
                       html.escape("\n".join(code)) + "</code></pre>")
            i += 1
            continue
        m = re.match(r"^(#{1,6})\s+(.*)$", ln)
        if m:
            flush_para()
            close_list()
            lvl = len(m.group(1))
            out.append(f"<h{lvl}>{_inline(m.group(2))}</h{lvl}>")
            i += 1
            continue
        if re.match(r"^\s*\|.*\|\s*$", ln):
            flush_para()
            close_list()
            rows = []
            while i < len(lines) and re.match(r"^\s*\|.*\|\s*$", lines[i]):
                rows.append([c.strip() for c in
                             lines[i].strip().strip("|").split("|")])
                i += 1
            out.append("<table>")
            for ri, row in enumerate(rows):
                if all(re.fullmatch(r":?-{2,}:?", c) for c in row):
                    continue
                tag = "th" if ri == 0 else "td"
                out.append("<tr>" + "".join(
                    f"<{tag}>{_inline(c)}</{tag}>" for c in row) + "</tr>")
            out.append("</table>")
            continue
        m = re.match(r"^(\s*)([*+-]|\d+\.)\s+(.*)$", ln)
        if m:
            flush_para()
            kind = "ol" if m.group(2)[0].isdigit() else "ul"
            if in_list != kind:
                close_list()
                out.append(f"<{kind}>")
                in_list = kind
            item = [m.group(3)]
            i += 1
            # hanging continuation lines
            while i < len(lines) and lines[i].startswith("  ") and \
                    not re.match(r"^\s*([*+-]|\d+\.)\s+", lines[i]):
                item.append(lines[i].strip())
                i += 1
            out.append("<li>" + _inline(" ".join(item)) + "</li>")
            continue
        if re.match(r"^\s*(---+|\*\*\*+)\s*$", ln):
            flush_para()
            close_list()
            out.append("<hr/>")
            i += 1
            continue
        if ln.startswith(">"):
            flush_para()
            close_list()
            quote = []
            while i < len(lines) and lines[i].startswith(">"):
                quote.append(lines[i].lstrip("> "))
                i += 1
            out.append("<blockquote><p>" + _inline(" ".join(quote)) +
                       "</p></blockquote>")
            continue
        if not ln.strip():
            flush_para()
            close_list()
            i += 1
            continue
        para.append(ln.strip())
        i += 1
    flush_para()
    close_list()

    body = "\n".join(out)
    return f"""<!DOCTYPE html>
<html lang="en"><head><meta charset="utf-8"/>
<title>{html.escape(title)}</title>
<style>
 body {{ max-width: 46em; margin: 2em auto; padding: 0 1em;
        font: 16px/1.55 Georgia, serif; color: #222; }}
 h1,h2,h3 {{ font-family: Helvetica, Arial, sans-serif; }}
 pre {{ background: #f6f6f6; padding: .8em; overflow-x: auto;
       font-size: 13px; line-height: 1.35; }}
 code {{ font-family: Menlo, Consolas, monospace; font-size: .92em; }}
 table {{ border-collapse: collapse; margin: 1em 0; }}
 th,td {{ border: 1px solid #bbb; padding: .3em .6em;
         text-align: left; }}
 blockquote {{ color: #555; border-left: 3px solid #ccc;
              margin-left: 0; padding-left: 1em; }}
</style></head>
<body>
{body}
</body></html>
"""


def main():
    src = os.path.join(HERE, "TUTORIAL.md")
    dst = os.path.join(HERE, "tutorial.html")
    with open(src) as f:
        md = f.read()
    title = md.split("\n", 1)[0].lstrip("# ").strip()
    html_out = render(md, title)
    with open(dst, "w") as f:
        f.write(html_out)
    # the reference copies tuto.html -> index.html (byte-identical)
    shutil.copyfile(dst, os.path.join(HERE, "index.html"))
    print(f"wrote {dst} and index.html ({len(html_out)} bytes)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
