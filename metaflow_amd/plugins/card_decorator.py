"""@card: per-task HTML report stored in the task datastore.

Parity target: /root/reference/metaflow/plugins/cards/ (3.9 kLoC) scoped to
the useful core: a self-contained HTML card per task with task metadata,
artifact summaries, logs, and (on GPU boxes) the telemetry the gpu_monitor
sidecar sampled. ``current.card.append(html_or_text)`` adds user content;
the card is rendered in task_finished and readable via
``Task.card_html`` / the `card` CLI.
"""

import html
import time

from ..decorators import StepDecorator, make_step_decorator

_CARD_TEMPLATE = """<!DOCTYPE html>
<html><head><meta charset="utf-8"><title>{title}</title>
<style>
body {{ font-family: -apple-system, Segoe UI, sans-serif; margin: 2rem;
       background: #fafafa; color: #222; }}
h1 {{ font-size: 1.3rem; }} h2 {{ font-size: 1.05rem; margin-top: 1.5rem; }}
table {{ border-collapse: collapse; width: 100%; background: #fff; }}
td, th {{ border: 1px solid #ddd; padding: 6px 10px; font-size: 0.9rem;
          text-align: left; }}
th {{ background: #f0f0f0; }}
pre {{ background: #1e1e1e; color: #d4d4d4; padding: 1rem;
       overflow-x: auto; font-size: 0.8rem; }}
.badge {{ display: inline-block; padding: 2px 10px; border-radius: 10px;
          color: #fff; background: {badge}; font-size: 0.85rem; }}
</style></head><body>
<h1>{title} <span class="badge">{status}</span></h1>
<table>
<tr><th>pathspec</th><td>{pathspec}</td></tr>
<tr><th>attempt</th><td>{attempt}</td></tr>
<tr><th>generated</th><td>{ts}</td></tr>
</table>
<h2>Artifacts</h2>
<table><tr><th>name</th><th>type</th><th>size (bytes)</th><th>sha</th></tr>
{artifact_rows}
</table>
{user_sections}
</body></html>
"""


class CardBuilder(object):
    def __init__(self):
        self.sections = []
        self._refresh_cb = None

    def refresh(self):
        """Write the card NOW with the sections appended so far
        (reference: CardCreator async refresh — long-running tasks can
        publish progress before task_finished renders the final card)."""
        if self._refresh_cb is not None:
            self._refresh_cb()

    def append(self, content, title=None):
        """Append a section: a card component (card_components.Markdown/
        Table/Image/Artifact), raw HTML (str starting with '<'), or
        plain text."""
        self.sections.append((title, content))

    def extend(self, contents):
        for c in contents:
            self.append(c)

    def render_sections(self):
        out = []
        for title, content in self.sections:
            if title:
                out.append("<h2>%s</h2>" % html.escape(str(title)))
            if hasattr(content, "render"):
                try:
                    out.append(content.render())
                except Exception as ex:  # a bad component can't kill a card
                    out.append("<pre>[component error: %s]</pre>"
                               % html.escape(repr(ex)))
                continue
            c = str(content)
            if c.lstrip().startswith("<"):
                out.append(c)
            else:
                out.append("<pre>%s</pre>" % html.escape(c))
        return "\n".join(out)


def render_card(task_datastore, step_name, ok, builder=None):
    rows = []
    for name in sorted(task_datastore.artifact_names()):
        if name.startswith("_"):
            continue
        info = task_datastore.artifact_info(name) or {}
        rows.append(
            "<tr><td>%s</td><td>%s</td><td>%s</td><td><code>%s</code>"
            "</td></tr>"
            % (html.escape(name), html.escape(str(info.get("type", ""))),
               info.get("size", ""), (info.get("sha") or "")[:16]))
    return _CARD_TEMPLATE.format(
        title="%s" % step_name,
        status="OK" if ok else "FAILED",
        badge="#2e7d32" if ok else "#c62828",
        pathspec=task_datastore.pathspec,
        attempt=task_datastore.attempt,
        ts=time.strftime("%Y-%m-%d %H:%M:%S UTC", time.gmtime()),
        artifact_rows="\n".join(rows),
        user_sections=builder.render_sections() if builder else "",
    )


def _telemetry_section(jsonl_path):
    """Summarize the gpu_monitor sidecar's samples into an HTML table."""
    import json as _json
    import os as _os

    if not _os.path.isfile(jsonl_path):
        return None
    mem, busy, n = [], [], 0
    try:
        with open(jsonl_path) as f:
            for line in f:
                try:
                    rec = _json.loads(line)
                except ValueError:
                    continue
                n += 1
                if "mem_used_gb" in rec:
                    mem.append(rec["mem_used_gb"])
                smi = rec.get("rocm_smi") or {}
                for dev in smi.values():
                    if isinstance(dev, dict):
                        use = dev.get("GPU use (%)")
                        if use is not None:
                            try:
                                busy.append(float(use))
                            except (TypeError, ValueError):
                                pass
    except OSError:
        return None
    if not n:
        return None
    rows = ["<tr><th>samples</th><td>%d</td></tr>" % n]
    if mem:
        rows.append("<tr><th>HBM used (GB)</th><td>max %.1f / mean %.1f"
                    "</td></tr>" % (max(mem), sum(mem) / len(mem)))
    if busy:
        rows.append("<tr><th>GPU busy (%%)</th><td>max %.0f / mean %.0f"
                    "</td></tr>" % (max(busy), sum(busy) / len(busy)))
    return "<table>%s</table>" % "".join(rows)


def rocprof_stats_section(stats_dir, top=15):
    """Kernel-time breakdown table from a rocprofv3 --stats output dir
    (any *kernel_stats.csv under it). Used by the runtime to splice a
    profile section into the task card when @card(profile=True)."""
    import csv
    import glob
    import os as _os

    files = sorted(glob.glob(_os.path.join(stats_dir, "**",
                                           "*kernel_stats.csv"),
                             recursive=True))
    if not files:
        return None
    rows = []
    with open(files[-1]) as f:
        reader = csv.DictReader(f)
        for rec in reader:
            keys = {k.lower().replace("_", "").replace(" ", ""): k
                    for k in rec}
            name = rec.get(keys.get("name", ""), "")
            pct = rec.get(keys.get("percentage", ""), "")
            total = rec.get(keys.get("totaldurationns", ""), "")
            calls = rec.get(keys.get("calls", ""), "")
            try:
                total_ms = float(total) / 1e6
            except (TypeError, ValueError):
                total_ms = 0.0
            rows.append((total_ms, pct, calls, name))
    rows.sort(reverse=True)
    out = ["<table><tr><th>total ms</th><th>%</th><th>calls</th>"
           "<th>kernel</th></tr>"]
    for total_ms, pct, calls, name in rows[:top]:
        try:
            pct_s = "%.1f" % float(pct)
        except (TypeError, ValueError):
            pct_s = str(pct)
        out.append("<tr><td>%.1f</td><td>%s</td><td>%s</td>"
                   "<td><code>%s</code></td></tr>"
                   % (total_ms, pct_s, html.escape(str(calls)),
                      html.escape(str(name)[:120])))
    out.append("</table>")
    return "\n".join(out)


class CardDecorator(StepDecorator):
    name = "card"
    # profile=True reruns the task under `rocprofv3 --kernel-trace
    # --stats` (when available on the box) and the scheduler splices the
    # kernel-time breakdown into the card after the task exits;
    # gpu_telemetry=True samples rocm-smi/HBM via the gpu_monitor
    # sidecar for the card's telemetry section.
    defaults = {"id": "default", "profile": False, "gpu_telemetry": True}
    allow_multiple = True

    def runtime_step_cli(self, args, retry_count, max_user_code_retries,
                         ubf_context):
        if not self.attributes.get("profile"):
            return
        import shutil
        import tempfile

        rocprof = shutil.which("rocprofv3")
        if rocprof is None:
            return
        out_dir = tempfile.mkdtemp(prefix="mfx_rocprof_")
        # --output-format csv: rocprofv3's default output is a SQLite
        # results.db; the stats CSV the splicer parses needs asking for
        args["cmd"] = [rocprof, "--kernel-trace", "--stats",
                       "--output-format", "csv", "-d", out_dir,
                       "--"] + args["cmd"]
        args["env"]["MFX_ROCPROF_OUT"] = out_dir
        args["env"].setdefault("TMPDIR", "/tmp")

    def task_pre_step(self, step_name, task_datastore, metadata, run_id,
                      task_id, flow, graph, retry_count,
                      max_user_code_retries, ubf_context, inputs):
        from ..current import current

        self._builder = CardBuilder()
        self._ds = task_datastore
        self._telemetry_path = None
        self._gpu_sidecar = None
        card_id = self.attributes.get("id", "default")
        if self.attributes.get("gpu_telemetry", True):
            try:
                import torch

                if torch.cuda.is_available():
                    import tempfile

                    from ..sidecar import SidecarSubProcess

                    fd, path = tempfile.mkstemp(prefix="mfx_gpu_card_",
                                                suffix=".jsonl")
                    import os as _os

                    _os.close(fd)
                    self._telemetry_path = path
                    self._gpu_sidecar = SidecarSubProcess(
                        "gpu_monitor", {"out_path": path})
            except Exception:
                self._gpu_sidecar = None

        def _refresh(builder=self._builder, ds=task_datastore,
                     step=step_name, cid=card_id):
            try:
                ds.save_metadata(
                    "card_%s" % cid,
                    {"html": render_card(ds, step, True, builder)})
            except Exception:
                pass

        self._builder._refresh_cb = _refresh
        current._update_env({"card": self._builder})

    def task_finished(self, step_name, flow, graph, is_task_ok, retry_count,
                      max_user_code_retries):
        try:
            if self._gpu_sidecar is not None:
                try:
                    self._gpu_sidecar.terminate()
                except Exception:
                    pass
            if self._telemetry_path:
                section = _telemetry_section(self._telemetry_path)
                if section:
                    self._builder.append(section, title="GPU telemetry")
            html_doc = render_card(self._ds, step_name, is_task_ok,
                                   self._builder)
            self._ds.save_metadata(
                "card_%s" % self.attributes.get("id", "default"),
                {"html": html_doc})
        except Exception:
            pass


card = make_step_decorator(CardDecorator)


def get_card(task_datastore, card_id="default"):
    meta = task_datastore.load_metadata("card_%s" % card_id)
    if meta is None:
        return None
    doc = meta["html"]
    # splice in the post-exit rocprof section when the scheduler saved
    # one (@card(profile=True): the stats CSV only exists after the task
    # process — and rocprofv3 around it — exited)
    prof = task_datastore.load_metadata("card_profile")
    if prof and prof.get("html") and "</body>" in doc:
        doc = doc.replace(
            "</body>",
            "<h2>Kernel-time breakdown (rocprofv3)</h2>\n%s\n</body>"
            % prof["html"])
    return doc
