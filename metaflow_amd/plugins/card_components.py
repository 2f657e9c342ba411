"""Typed card components.

Parity target: /root/reference/metaflow/plugins/cards/card_modules/
(Markdown, Table, Image, Artifact components appended via
``current.card.append(...)``). Each component renders to self-contained
HTML; the card stays a single file with no external assets (data-URI
images), so it travels through the CAS like any artifact.
"""

import base64
import html


class CardComponent(object):
    def render(self):
        raise NotImplementedError


class Markdown(CardComponent):
    """Small-subset markdown: #/##/### headings, **bold**, *italic*,
    `code`, fenced code blocks, - lists, blank-line paragraphs."""

    def __init__(self, text):
        self.text = text

    def render(self):
        import re

        out = []
        in_code = False
        in_list = False
        for raw in str(self.text).splitlines():
            line = raw.rstrip()
            if line.strip().startswith("```"):
                if in_code:
                    out.append("</pre>")
                else:
                    out.append("<pre>")
                in_code = not in_code
                continue
            if in_code:
                out.append(html.escape(line))
                continue
            if line.startswith("- "):
                if not in_list:
                    out.append("<ul>")
                    in_list = True
                out.append("<li>%s</li>" % self._inline(line[2:]))
                continue
            if in_list:
                out.append("</ul>")
                in_list = False
            m = re.match(r"^(#{1,3})\s+(.*)$", line)
            if m:
                lvl = len(m.group(1)) + 1  # h2..h4 inside the card
                out.append("<h%d>%s</h%d>"
                           % (lvl, self._inline(m.group(2)), lvl))
            elif line:
                out.append("<p>%s</p>" % self._inline(line))
        if in_list:
            out.append("</ul>")
        if in_code:
            out.append("</pre>")
        return "\n".join(out)

    @staticmethod
    def _inline(s):
        import re

        s = html.escape(s)
        s = re.sub(r"\*\*(.+?)\*\*", r"<b>\1</b>", s)
        s = re.sub(r"\*(.+?)\*", r"<i>\1</i>", s)
        s = re.sub(r"`(.+?)`", r"<code>\1</code>", s)
        return s


class Table(CardComponent):
    def __init__(self, data, headers=None):
        self.data = data
        self.headers = headers

    def render(self):
        rows = []
        if self.headers:
            rows.append("<tr>%s</tr>" % "".join(
                "<th>%s</th>" % html.escape(str(h)) for h in self.headers))
        for row in self.data:
            rows.append("<tr>%s</tr>" % "".join(
                "<td>%s</td>" % html.escape(str(c)) for c in row))
        return "<table>%s</table>" % "\n".join(rows)


class Image(CardComponent):
    """Embed raw image bytes (or a matplotlib figure) as a data URI."""

    def __init__(self, src, label=None, fmt="png"):
        self.label = label
        if hasattr(src, "savefig"):  # matplotlib figure
            import io

            buf = io.BytesIO()
            src.savefig(buf, format=fmt)
            src = buf.getvalue()
        self.data = src
        self.fmt = fmt

    def render(self):
        b64 = base64.b64encode(self.data).decode()
        img = ('<img src="data:image/%s;base64,%s" '
               'style="max-width:100%%"/>' % (self.fmt, b64))
        if self.label:
            img += "<p><i>%s</i></p>" % html.escape(str(self.label))
        return img


class Artifact(CardComponent):
    """Pretty-printed repr of any python object (truncated)."""

    def __init__(self, obj, name=None, max_chars=4096):
        self.obj = obj
        self.name = name
        self.max_chars = max_chars

    def render(self):
        try:
            import pprint

            text = pprint.pformat(self.obj, width=100)
        except Exception:
            text = repr(self.obj)
        if len(text) > self.max_chars:
            text = text[:self.max_chars] + "\n... [truncated]"
        head = ("<b><code>%s</code></b>" % html.escape(self.name)
                if self.name else "")
        return "%s<pre>%s</pre>" % (head, html.escape(text))
